"""Pipeline-parallel-aware work assignment.

Parity with reference kfac/gpt_neox/assignment.py:20-236, minus the
DeepSpeed dependency: any object with ``get_axis_comm_lists(axis)``,
``get_coord(rank)`` and ``world_size()`` works as the topology (see
kfac_amd.gpt_neox.topology.PipeModelDataTopology).

Work is balanced only across the pipe-parallel peers (ranks that own the
same layers); the strategy is hard-coded MEM-OPT: only the inverse
worker preconditions (no inverse broadcast), preconditioned gradients
are broadcast over the data-parallel group every step.
"""

from __future__ import annotations

from typing import Any

import torch.distributed as dist

from kfac_amd.assignment import WorkAssignment
from kfac_amd.gpt_neox.mpu import get_group_with_rank


class GPTNeoXAssignment(WorkAssignment):
    """MEM-OPT assignment balanced across same-pipe-stage ranks."""

    def __init__(
        self,
        work: dict[str, dict[str, float]],
        *,
        local_rank: int,
        topology: Any,
        data_parallel_group: dist.ProcessGroup | None,
        model_parallel_group: dist.ProcessGroup | None,
    ) -> None:
        """Init GPTNeoXAssignment.

        Args:
            work: {layer: {factor: cost}} for the layers THIS pipe stage
                owns (costs identical across the stage's ranks).
            local_rank: this process's global rank.
            topology: 3D topology (duck-typed; see module docstring).
            data_parallel_group: DP group of this rank.
            model_parallel_group: TP group of this rank.
        """
        for attr in ('get_axis_comm_lists', 'get_coord', 'world_size'):
            if not hasattr(topology, attr):
                raise TypeError(
                    f'topology must provide {attr}(); got {type(topology)}',
                )

        self.local_rank = local_rank
        self.data_parallel_group = data_parallel_group
        self.model_parallel_group = model_parallel_group

        self.data_parallel_groups = topology.get_axis_comm_lists('data')
        self.model_parallel_groups = topology.get_axis_comm_lists('model')
        self.pipe_parallel_groups = topology.get_axis_comm_lists('pipe')

        self.data_parallel_peers = get_group_with_rank(
            local_rank, self.data_parallel_groups,
        )
        self.model_parallel_peers = get_group_with_rank(
            local_rank, self.model_parallel_groups,
        )
        self.pipe_parallel_rank = topology.get_coord(local_rank).pipe
        # Ranks holding the same layers as us: only they matter for
        # balancing this stage's factor work.
        self.pipe_parallel_peers = [
            r
            for r in range(topology.world_size())
            if topology.get_coord(r).pipe == self.pipe_parallel_rank
        ]

        # Reuse an existing group when the peer set coincides with it.
        # A None mp/dp group means size-1 (no parallelism on that axis)
        # and must NOT be reused: group=None is the GLOBAL group to
        # torch.distributed, and with pp > 1 a factor allreduce over the
        # global group would cross pipeline stages that own different
        # layers (mismatched collectives).
        if (
            set(self.pipe_parallel_peers) == set(self.model_parallel_peers)
            and self.model_parallel_group is not None
        ):
            self.pipe_parallel_peer_group = self.model_parallel_group
        elif (
            set(self.pipe_parallel_peers) == set(self.data_parallel_peers)
            and self.data_parallel_group is not None
        ):
            self.pipe_parallel_peer_group = self.data_parallel_group
        else:
            # dist.new_group is collective over the WORLD: every rank
            # must create every stage's peer group, in the same order,
            # keeping its own. (Per-stage peer lists differ by rank, so
            # a single new_group(self.pipe_parallel_peers) call would
            # violate the collective contract.)
            self.pipe_parallel_peer_group = None
            if dist.is_initialized():
                num_stages = 1 + max(
                    topology.get_coord(r).pipe
                    for r in range(topology.world_size())
                )
                for stage in range(num_stages):
                    peers = [
                        r
                        for r in range(topology.world_size())
                        if topology.get_coord(r).pipe == stage
                    ]
                    group = dist.new_group(peers)
                    if stage == self.pipe_parallel_rank:
                        self.pipe_parallel_peer_group = group
                assert self.pipe_parallel_peer_group is not None
            # else: serial/unit-test construction — no groups to make,
            # and every collective is a no-op without an initialized
            # world

        # Greedy lowest-load balance over the pipe peers (colocated
        # factors: MEM-OPT needs A and G on one rank).
        loads = [0.0 for _ in self.pipe_parallel_peers]
        self._inv_assignments = {
            layer: {factor: -1 for factor in factors}
            for layer, factors in work.items()
        }
        ordered = sorted(
            ((layer, sum(f.values())) for layer, f in work.items()),
            key=lambda item: (item[1], item[0]),
            reverse=True,
        )
        for layer, cost in ordered:
            idx = loads.index(min(loads))
            worker = self.pipe_parallel_peers[idx]
            for factor in self._inv_assignments[layer]:
                self._inv_assignments[layer][factor] = worker
            loads[idx] += cost

    def broadcast_gradients(self) -> bool:
        """MEM-OPT: preconditioned grads are broadcast every step."""
        return True

    def broadcast_inverses(self) -> bool:
        """MEM-OPT: no inverse broadcast."""
        return False

    def get_layers(self) -> tuple[str, ...]:
        return tuple(self._inv_assignments.keys())

    def get_factors(self, layer: str) -> tuple[str, ...]:
        return tuple(self._inv_assignments[layer].keys())

    def inv_worker(self, layer: str, factor: str) -> int:
        return self._inv_assignments[layer][factor]

    def factor_worker(self, layer: str, factor: str) -> int:
        """Primary rank: the member of the inv worker's DP group that is
        also a TP peer of this rank (gathers factors from TP shards).
        """
        inv_ranks = set(self._inv_assignments[layer].values())
        assert len(inv_ranks) == 1
        inv_rank = inv_ranks.pop()
        dp_ranks = get_group_with_rank(inv_rank, self.data_parallel_groups)
        workers = set(dp_ranks) & set(self.model_parallel_peers)
        assert len(workers) == 1
        return workers.pop()

    def is_grad_worker(self, layer: str) -> bool:
        """True if this rank's TP group contains the inv worker.

        Every TP peer of the inv worker must enter preconditioned_grad
        (gather/scatter protocol), so membership is by TP group.
        """
        return (
            len(
                set(self._inv_assignments[layer].values())
                & set(self.model_parallel_peers),
            )
            == 1
        )

    def src_grad_worker(self, layer: str) -> int:
        """The DP peer of this rank inside the inv worker's TP group."""
        ranks = list(self._inv_assignments[layer].values())
        assert ranks.count(ranks[0]) == len(ranks)
        src_rank = ranks[0]
        mp_ranks = get_group_with_rank(src_rank, self.model_parallel_groups)
        src = set(self.data_parallel_peers) & set(mp_ranks)
        assert len(src) == 1
        return src.pop()

    def factor_group(self, layer: str, factor: str) -> dist.ProcessGroup | None:
        """Ignored: GPTNeoXKFACEigenLayer routes reductions itself."""
        return None

    def grad_worker_group(self, layer: str) -> dist.ProcessGroup | None:
        raise NotImplementedError(
            'The GPT-NeoX assignment strategy only supports MEM-OPT and '
            'therefore should not be performing inverse factor '
            'communication.',
        )

    def grad_receiver_group(self, layer: str) -> dist.ProcessGroup | None:
        return self.data_parallel_group
