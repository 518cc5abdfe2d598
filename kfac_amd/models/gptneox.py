"""GPT-NeoX-style decoder LM (125M default) for the LM benchmark.

Matches the BASELINE.json config "GPT-NeoX 125M (kfac/gpt_neox
Linear-only layers) synthetic seq=2048": K-FAC preconditions the MLP
linears (named mlp.dense_h_to_4h / mlp.dense_4h_to_h like GPT-NeoX);
attention, embeddings and the LM head are skip-listed by the caller.
Attention uses torch's fused scaled_dot_product_attention.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class MLP(nn.Module):
    def __init__(self, d: int, ffn: int):
        super().__init__()
        self.dense_h_to_4h = nn.Linear(d, ffn)
        self.dense_4h_to_h = nn.Linear(ffn, d)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dense_4h_to_h(F.gelu(self.dense_h_to_4h(x)))


class Attention(nn.Module):
    def __init__(self, d: int, n_heads: int):
        super().__init__()
        self.n_heads = n_heads
        self.query_key_value = nn.Linear(d, 3 * d)
        self.dense = nn.Linear(d, d)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, s, d = x.shape
        qkv = self.query_key_value(x)
        q, k, v = qkv.chunk(3, dim=-1)
        hd = d // self.n_heads

        def heads(t: torch.Tensor) -> torch.Tensor:
            return t.view(b, s, self.n_heads, hd).transpose(1, 2)

        out = F.scaled_dot_product_attention(
            heads(q), heads(k), heads(v), is_causal=True,
        )
        out = out.transpose(1, 2).reshape(b, s, d)
        return self.dense(out)


class Block(nn.Module):
    def __init__(self, d: int, n_heads: int, ffn: int):
        super().__init__()
        self.input_layernorm = nn.LayerNorm(d)
        self.attention = Attention(d, n_heads)
        self.post_attention_layernorm = nn.LayerNorm(d)
        self.mlp = MLP(d, ffn)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.attention(self.input_layernorm(x))
        return x + self.mlp(self.post_attention_layernorm(x))


class GPTNeoXModel(nn.Module):
    """Decoder-only LM; (batch, seq) int tokens -> (batch, seq, vocab)."""

    def __init__(
        self,
        vocab: int = 50304,
        d: int = 768,
        n_layers: int = 12,
        n_heads: int = 12,
        ffn: int = 3072,
        max_seq: int = 2048,
    ):
        super().__init__()
        self.embed_in = nn.Embedding(vocab, d)
        self.embed_pos = nn.Embedding(max_seq, d)
        self.layers = nn.ModuleList(
            Block(d, n_heads, ffn) for _ in range(n_layers)
        )
        self.final_layer_norm = nn.LayerNorm(d)
        self.embed_out = nn.Linear(d, vocab, bias=False)
        # weight tying -> 125.3M parameters total at the default config
        self.embed_out.weight = self.embed_in.weight

    def forward(self, tokens: torch.Tensor) -> torch.Tensor:
        b, s = tokens.shape
        pos = torch.arange(s, device=tokens.device)
        x = self.embed_in(tokens) + self.embed_pos(pos)[None]
        for layer in self.layers:
            x = layer(x)
        return self.embed_out(self.final_layer_norm(x))


def gptneox_125m(vocab: int = 50304) -> GPTNeoXModel:
    """~125M-parameter configuration."""
    return GPTNeoXModel(vocab=vocab, d=768, n_layers=12, n_heads=12, ffn=3072)


# skip-list for K-FAC: precondition MLP linears only (reference LM
# example skips embedding/decoder/attention, torch_language_model.py:162-167)
KFAC_SKIP_LAYERS = ['.*attention.*', 'embed.*', '.*embed_out.*']
