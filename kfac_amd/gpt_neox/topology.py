"""3D parallel topology description (DeepSpeed-free).

Replaces the reference's dependency on DeepSpeed's
``PipeModelDataParallelTopology`` (kfac/gpt_neox/assignment.py:10-17) with
a self-contained mixed-radix rank layout: axes ordered (pipe, data,
model) with pipe most significant and model fastest-varying — the same
layout DeepSpeed uses, so group lists are interchangeable.
"""

from __future__ import annotations

import dataclasses


@dataclasses.dataclass(frozen=True)
class Coord:
    """Coordinates of a rank in the 3D grid."""

    pipe: int
    data: int
    model: int


class PipeModelDataTopology:
    """Mixed-radix (pipe, data, model) rank topology."""

    def __init__(self, num_pp: int, num_mp: int, num_dp: int) -> None:
        """Init topology.

        Args:
            num_pp: pipeline-parallel size (most significant axis).
            num_mp: model/tensor-parallel size (fastest-varying axis).
            num_dp: data-parallel size.
        """
        if num_pp < 1 or num_mp < 1 or num_dp < 1:
            raise ValueError('all parallelism degrees must be >= 1')
        self.num_pp = num_pp
        self.num_mp = num_mp
        self.num_dp = num_dp

    def world_size(self) -> int:
        """Total ranks."""
        return self.num_pp * self.num_dp * self.num_mp

    def get_coord(self, rank: int) -> Coord:
        """Decompose a rank into (pipe, data, model) coordinates."""
        if not 0 <= rank < self.world_size():
            raise ValueError(f'rank {rank} outside world {self.world_size()}')
        model = rank % self.num_mp
        t = rank // self.num_mp
        data = t % self.num_dp
        pipe = t // self.num_dp
        return Coord(pipe=pipe, data=data, model=model)

    def get_rank(self, pipe: int, data: int, model: int) -> int:
        """Compose a rank from coordinates."""
        return (pipe * self.num_dp + data) * self.num_mp + model

    def get_axis_comm_lists(self, axis: str) -> list[list[int]]:
        """Rank groups varying only along ``axis``.

        Same contract as DeepSpeed's topology: e.g. axis='data' returns
        one list per (pipe, model) pair containing the ranks that differ
        only in their data coordinate.
        """
        groups: list[list[int]] = []
        if axis == 'data':
            for p in range(self.num_pp):
                for m in range(self.num_mp):
                    groups.append(
                        [self.get_rank(p, d, m) for d in range(self.num_dp)],
                    )
        elif axis == 'model':
            for p in range(self.num_pp):
                for d in range(self.num_dp):
                    groups.append(
                        [self.get_rank(p, d, m) for m in range(self.num_mp)],
                    )
        elif axis == 'pipe':
            for d in range(self.num_dp):
                for m in range(self.num_mp):
                    groups.append(
                        [self.get_rank(p, d, m) for p in range(self.num_pp)],
                    )
        else:
            raise ValueError(f'unknown axis {axis!r}')
        return groups
