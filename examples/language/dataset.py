"""Token-stream datasets for the LM example.

Offline image: synthetic token streams by default; a plain-text corpus
file can be tokenized bytewise when present (parity role of reference
examples/language/dataset.py without the torchtext download).
"""

from __future__ import annotations

import os

import torch


def synthetic_batch(
    vocab: int,
    batch: int,
    seq: int,
    device: torch.device,
    seed: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    """(input, target) token batch, reproducible per seed."""
    g = torch.Generator().manual_seed(seed)
    data = torch.randint(0, vocab, (seq + 1, batch), generator=g)
    return data[:-1].to(device), data[1:].reshape(-1).to(device)


def load_corpus(path: str, vocab: int = 256) -> torch.Tensor | None:
    """Bytewise-tokenize a text file into one long tensor (or None)."""
    if not os.path.isfile(path):
        return None
    with open(path, 'rb') as f:
        data = f.read()
    return torch.frombuffer(bytearray(data), dtype=torch.uint8).long() % vocab


def corpus_batch(
    corpus: torch.Tensor,
    batch: int,
    seq: int,
    step: int,
    device: torch.device,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Sequential (input, target) batch from a tokenized corpus."""
    n = corpus.numel() - 1
    starts = (
        (step * batch + torch.arange(batch)) * seq
    ) % (n - seq)
    x = torch.stack([corpus[s : s + seq] for s in starts], dim=1)
    y = torch.stack([corpus[s + 1 : s + seq + 1] for s in starts], dim=1)
    return x.to(device), y.reshape(-1).to(device)
