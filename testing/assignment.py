"""LazyAssignment: trivial WorkAssignment for layer-math tests.

Every rank is both the inverse worker and a gradient worker for every
layer (reference testing/assignment.py:10-87); decouples placement from
layer math so the preconditioner control flow can be exercised without a
real KAISA grid. ``broadcast`` toggles the broadcast branches.
"""

from __future__ import annotations

import torch.distributed as dist

from kfac_amd.assignment import WorkAssignment


class LazyAssignment(WorkAssignment):
    """Every rank does everything; groups are the global group."""

    def __init__(self, rank: int = 0, broadcast: bool = False):
        self.rank = rank
        self.broadcast = broadcast

    def broadcast_gradients(self) -> bool:
        return self.broadcast

    def broadcast_inverses(self) -> bool:
        return self.broadcast

    def get_layers(self) -> tuple[str, ...]:
        return ()

    def get_factors(self, layer: str) -> tuple[str, ...]:
        return ('A', 'G')

    def inv_worker(self, layer: str, factor: str) -> int:
        return self.rank

    def is_grad_worker(self, layer: str) -> bool:
        return True

    def src_grad_worker(self, layer: str) -> int:
        return self.rank

    def factor_group(self, layer: str, factor: str) -> dist.ProcessGroup | None:
        return None

    def grad_worker_group(self, layer: str) -> dist.ProcessGroup | None:
        return None

    def grad_receiver_group(self, layer: str) -> dist.ProcessGroup | None:
        return None
