"""Mocks for tensor-parallel modules (reference testing/gpt_neox.py:16-52).

Registration matches on *class name*, so plain nn.Linear subclasses with
the Megatron names suffice.
"""

from __future__ import annotations

import torch


class ColumnParallelLinear(torch.nn.Linear):
    """Output-sharded linear (weight shard: [out/mp, in])."""


class RowParallelLinear(torch.nn.Linear):
    """Input-sharded linear (weight shard: [out, in/mp])."""


class ParallelMLP(torch.nn.Module):
    """Two sharded linears, mp=1 shapes (for dp-only tests)."""

    def __init__(self, in_dim: int = 10, hidden: int = 16, out_dim: int = 4):
        super().__init__()
        self.dense_h_to_4h = ColumnParallelLinear(in_dim, hidden)
        self.dense_4h_to_h = RowParallelLinear(hidden, out_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dense_4h_to_h(torch.relu(self.dense_h_to_4h(x)))


class FullMLP(torch.nn.Module):
    """Unsharded twin of ShardedParallelMLP (single-process reference)."""

    def __init__(self, in_dim: int = 10, hidden: int = 16, out_dim: int = 4):
        super().__init__()
        self.dense_h_to_4h = torch.nn.Linear(in_dim, hidden, bias=True)
        self.dense_4h_to_h = torch.nn.Linear(hidden, out_dim, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dense_4h_to_h(torch.relu(self.dense_h_to_4h(x)))


class _CopyToModelParallel(torch.autograd.Function):
    """Identity forward; allreduce backward (megatron f operator)."""

    @staticmethod
    def forward(ctx, x, group):  # type: ignore[override]
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        import torch.distributed as dist

        grad = grad.clone()
        dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromModelParallel(torch.autograd.Function):
    """Allreduce forward; identity backward (megatron g operator)."""

    @staticmethod
    def forward(ctx, x, group):  # type: ignore[override]
        import torch.distributed as dist

        x = x.clone()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        return grad, None


class ShardedParallelMLP(torch.nn.Module):
    """mp-sharded twin of FullMLP, built FROM a FullMLP's weights.

    Column layer holds rows [rank*h/mp, (rank+1)*h/mp) of W1 and the
    matching bias shard; row layer holds the same slice of W2's columns
    (bias-free: megatron adds the row bias after the reduce). Forward
    reproduces the megatron f/g collective pattern so autograd produces
    the sharded gradients a real TP model would.
    """

    def __init__(
        self,
        full: FullMLP,
        rank: int,
        mp_world: int,
        group: 'torch.distributed.ProcessGroup',
    ):
        super().__init__()
        hidden = full.dense_h_to_4h.out_features
        assert hidden % mp_world == 0
        shard = hidden // mp_world
        sl = slice(rank * shard, (rank + 1) * shard)
        self.dense_h_to_4h = ColumnParallelLinear(
            full.dense_h_to_4h.in_features, shard, bias=True,
        )
        self.dense_4h_to_h = RowParallelLinear(
            shard, full.dense_4h_to_h.out_features, bias=False,
        )
        with torch.no_grad():
            self.dense_h_to_4h.weight.copy_(full.dense_h_to_4h.weight[sl])
            self.dense_h_to_4h.bias.copy_(full.dense_h_to_4h.bias[sl])
            self.dense_4h_to_h.weight.copy_(full.dense_4h_to_h.weight[:, sl])
        self.group = group

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = _CopyToModelParallel.apply(x, self.group)
        h = torch.relu(self.dense_h_to_4h(x))
        partial = self.dense_4h_to_h(h)
        return _ReduceFromModelParallel.apply(partial, self.group)
