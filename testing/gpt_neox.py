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
