"""KAISA work assignment: who inverts which factor, who preconditions.

Parity surface with reference kfac/assignment.py:30-471. The KAISA model
(parameterized by ``grad_worker_fraction``) arranges the world as a
G x (W/G) grid, G = number of gradient workers per layer:

  - gradient-worker groups = the grid *columns* (ranks with stride W/G):
    every layer is served by exactly one column (the one containing its
    inverse worker); members of that column precondition the layer's
    gradient themselves and receive the eigendecomposition broadcast.
  - gradient-receiver groups = the grid *rows* (contiguous rank blocks):
    a row intersects every column in exactly ONE rank — that rank is the
    row's src_grad_worker for layers owned by the column.

On one 8-GPU MI355X node, xGMI is 7 point-to-point links per GPU, so both
broadcast patterns (inverse -> column, precon grad -> row) are single-hop:
any sub-group broadcast is per-link bound, which is why HYBRID-OPT's
smaller broadcast groups cost no extra latency versus NVSwitch fabrics.

Invariants (property-tested in tests/test_assignment.py):
  * columns partition the world; rows partition the world;
  * |column| = G, |row| = W/G;
  * |column ∩ row| = 1 for every column/row pair;
  * every layer's inverse workers lie inside its column.
"""

from __future__ import annotations

import dataclasses
from abc import ABCMeta
from abc import abstractmethod
from typing import Callable

import torch.distributed as dist


class WorkAssignment(metaclass=ABCMeta):
    """Abstract work-assignment interface (reference assignment.py:30-118)."""

    def __repr__(self) -> str:
        layer_strs = []
        for layer in self.get_layers():
            invs = {
                factor: self.inv_worker(layer, factor)
                for factor in self.get_factors(layer)
            }
            layer_strs.append(
                f'  layer="{layer}": '
                f'is_grad_worker={self.is_grad_worker(layer)}, '
                f'src_grad_worker={self.src_grad_worker(layer)}, '
                f'inv_workers={invs}',
            )
        body = ',\n'.join(layer_strs)
        return f'{self.__class__.__name__}(\n{body}\n)'

    @abstractmethod
    def broadcast_gradients(self) -> bool:
        """True if preconditioned gradients must be broadcast."""
        raise NotImplementedError

    @abstractmethod
    def broadcast_inverses(self) -> bool:
        """True if inverses must be broadcast to other grad workers."""
        raise NotImplementedError

    @abstractmethod
    def get_layers(self) -> tuple[str, ...]:
        """All assigned layer names."""
        raise NotImplementedError

    @abstractmethod
    def get_factors(self, layer: str) -> tuple[str, ...]:
        """Factor names for a layer."""
        raise NotImplementedError

    @abstractmethod
    def inv_worker(self, layer: str, factor: str) -> int:
        """Rank that computes this factor's inverse."""
        raise NotImplementedError

    @abstractmethod
    def is_grad_worker(self, layer: str) -> bool:
        """True if this rank preconditions this layer's gradient."""
        raise NotImplementedError

    @abstractmethod
    def src_grad_worker(self, layer: str) -> int:
        """Rank that sends this rank the preconditioned gradient."""
        raise NotImplementedError

    @abstractmethod
    def factor_group(self, layer: str, factor: str) -> dist.ProcessGroup | None:
        """Group over which this factor is allreduced."""
        raise NotImplementedError

    @abstractmethod
    def grad_worker_group(self, layer: str) -> dist.ProcessGroup | None:
        """Group for the inverse broadcast (inv worker -> grad workers)."""
        raise NotImplementedError

    @abstractmethod
    def grad_receiver_group(self, layer: str) -> dist.ProcessGroup | None:
        """Group for the precon-grad broadcast (src worker -> receivers)."""
        raise NotImplementedError


@dataclasses.dataclass
class _Group:
    """Rank set + its (possibly None = global) process group handle."""

    ranks: frozenset[int]
    group: dist.ProcessGroup | None


class KAISAAssignment(WorkAssignment):
    """KAISA grid assignment (reference assignment.py:121-471)."""

    def __init__(
        self,
        work: dict[str, dict[str, float]],
        *,
        local_rank: int,
        world_size: int,
        grad_worker_fraction: float,
        group_func: Callable[[list[int]], dist.ProcessGroup | None],
        colocate_factors: bool = True,
    ) -> None:
        """Init KAISAAssignment.

        Args:
            work: {layer: {factor: cost}} load model (n^3 or n^2).
            local_rank: this process's global rank.
            world_size: world size.
            grad_worker_fraction: G/W; W*fraction must be an integer
                (G = max(1, W*fraction)) dividing W.
            group_func: factory mapping a rank list to a process group
                (all ranks must call it identically and in the same order).
            colocate_factors: place both factors of a layer on one rank
                (required for prediv_eigenvalues).
        """
        if not 0 <= grad_worker_fraction <= 1:
            raise ValueError(
                f'grad_worker_fraction must be in [0, 1]. '
                f'Got {grad_worker_fraction}.',
            )
        if local_rank < 0:
            raise ValueError('local_rank must be >= 0')
        if world_size <= 0:
            raise ValueError('world_size must be > 0')
        if local_rank >= world_size:
            raise ValueError(
                f'local_rank={local_rank} larger than world_size={world_size}',
            )
        grad_workers_f = max(1, world_size * grad_worker_fraction)
        if grad_workers_f != int(grad_workers_f):
            raise ValueError(
                'world_size*grad_worker_fraction must produce an integer '
                f'value. Found {world_size}*{grad_worker_fraction}'
                f'={grad_workers_f}.',
            )
        grad_workers = int(grad_workers_f)
        if world_size % grad_workers != 0:
            raise ValueError(
                'world_size must be an integer multiple of the gradient '
                'worker count',
            )

        self.local_rank = local_rank
        self.world_size = world_size
        self.grad_worker_fraction = grad_worker_fraction
        self.grad_workers = grad_workers
        self.group_func = group_func
        self.colocate_factors = colocate_factors

        columns = self.partition_grad_workers(world_size, grad_workers)
        rows = self.partition_grad_receivers(world_size, grad_workers)

        # One process group per distinct rank set, keyed by membership
        # (not size — see the reference's latent bug, distributed.py:376-378).
        group_cache: dict[frozenset[int], dist.ProcessGroup | None] = {}
        for ranks in sorted(columns | rows, key=lambda s: sorted(s)):
            group_cache[ranks] = group_func(sorted(ranks))

        self._inv_assignments = self.greedy_assignment(
            work,
            [sorted(ranks) for ranks in sorted(columns, key=lambda s: sorted(s))],
            world_size,
            colocate_factors,
        )

        self._grad_worker_groups: dict[str, _Group] = {}
        self._grad_receiver_groups: dict[str, _Group] = {}
        my_row = next(r for r in rows if local_rank in r)
        for layer, factors in self._inv_assignments.items():
            inv_worker = next(iter(factors.values()))
            col = next(c for c in columns if inv_worker in c)
            self._grad_worker_groups[layer] = _Group(col, group_cache[col])
            self._grad_receiver_groups[layer] = _Group(my_row, group_cache[my_row])

    # -- grid construction -------------------------------------------------

    @staticmethod
    def partition_grad_workers(
        world_size: int,
        grad_workers: int,
    ) -> set[frozenset[int]]:
        """Columns of the KAISA grid: rank sets with stride W/G.

        Example (W=8, G=2): {{0,4},{1,5},{2,6},{3,7}}.
        """
        if world_size <= 0:
            raise ValueError('world_size must be > 0')
        if grad_workers <= 0 or world_size % grad_workers != 0:
            raise ValueError(
                'world_size must be an integer multiple of the gradient '
                'worker count',
            )
        stride = world_size // grad_workers
        return {
            frozenset(range(c, world_size, stride)) for c in range(stride)
        }

    @staticmethod
    def partition_grad_receivers(
        world_size: int,
        grad_workers: int,
    ) -> set[frozenset[int]]:
        """Rows of the KAISA grid: contiguous rank blocks of size W/G.

        Example (W=8, G=2): {{0,1,2,3},{4,5,6,7}}.
        """
        if world_size <= 0:
            raise ValueError('world_size must be > 0')
        if grad_workers <= 0 or world_size % grad_workers != 0:
            raise ValueError(
                'world_size must be an integer multiple of the gradient '
                'worker count',
            )
        size = world_size // grad_workers
        return {
            frozenset(range(r * size, (r + 1) * size))
            for r in range(grad_workers)
        }

    # -- load balancing ----------------------------------------------------

    @staticmethod
    def greedy_assignment(
        work: dict[str, dict[str, float]],
        worker_groups: list[list[int]],
        world_size: int,
        colocate_factors: bool,
    ) -> dict[str, dict[str, int]]:
        """Greedy lowest-load assignment of factor work.

        Layers are taken in descending total-cost order; each layer goes
        to the currently least-loaded worker group; within the group its
        factors go to the least-loaded rank (both together if
        ``colocate_factors``). Deterministic given identical inputs, so
        every rank computes the same assignment without communication
        (reference assignment.py:227-319).
        """
        loads = [0.0] * world_size
        assignments: dict[str, dict[str, int]] = {
            layer: {factor: -1 for factor in factors}
            for layer, factors in work.items()
        }
        totals = {layer: sum(f.values()) for layer, f in work.items()}
        # Stable tie-break on name keeps the order identical across ranks.
        ordered = sorted(totals, key=lambda l: (-totals[l], l))

        for layer in ordered:
            group = min(
                worker_groups,
                key=lambda g: sum(loads[i] for i in g),
            )
            if colocate_factors:
                rank = min(group, key=lambda i: loads[i])
                loads[rank] += totals[layer]
                for factor in work[layer]:
                    assignments[layer][factor] = rank
            else:
                factors = sorted(
                    work[layer].items(),
                    key=lambda kv: (kv[1], kv[0]),
                    reverse=True,
                )
                for factor, cost in factors:
                    rank = min(group, key=lambda i: loads[i])
                    loads[rank] += cost
                    assignments[layer][factor] = rank

        for layer in assignments:
            for factor in assignments[layer]:
                assert assignments[layer][factor] >= 0
        return assignments

    # -- WorkAssignment interface -------------------------------------------

    def broadcast_gradients(self) -> bool:
        """True unless COMM-OPT (every rank is a grad worker)."""
        return self.grad_workers < self.world_size

    def broadcast_inverses(self) -> bool:
        """True unless MEM-OPT (only the inv worker preconditions)."""
        return self.grad_workers > 1

    def get_layers(self) -> tuple[str, ...]:
        return tuple(self._inv_assignments.keys())

    def get_factors(self, layer: str) -> tuple[str, ...]:
        return tuple(self._inv_assignments[layer].keys())

    def inv_worker(self, layer: str, factor: str) -> int:
        return self._inv_assignments[layer][factor]

    def is_grad_worker(self, layer: str) -> bool:
        return self.local_rank in self._grad_worker_groups[layer].ranks

    def src_grad_worker(self, layer: str) -> int:
        """The single rank in (my row ∩ the layer's column)."""
        inter = (
            self._grad_worker_groups[layer].ranks
            & self._grad_receiver_groups[layer].ranks
        )
        assert len(inter) == 1
        return next(iter(inter))

    def factor_group(self, layer: str, factor: str) -> dist.ProcessGroup | None:
        """Factors are contributed by every data-parallel rank: global group."""
        return None

    def grad_worker_group(self, layer: str) -> dist.ProcessGroup | None:
        return self._grad_worker_groups[layer].group

    def grad_receiver_group(self, layer: str) -> dist.ProcessGroup | None:
        return self._grad_receiver_groups[layer].group
