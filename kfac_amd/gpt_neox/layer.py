"""Tensor-parallel-aware K-FAC eigen layer.

Parity with reference kfac/gpt_neox/layer.py:18-311: sharded activations
(input parallelism) or output-grads (output parallelism) are gathered to
the layer's primary rank before factor accumulation; factor reductions
route per parallelism (sharded-dim factor over the data-parallel group by
the primary only, unsharded factor over the pipe-peer group by all); the
precondition step gathers weight-grad shards to the primary, runs the
Kronecker chain there, and scatters shards back.

MI355X difference: the scatter is a TRUE ``dist.scatter`` (RCCL supports
send/recv point-to-point over xGMI) instead of the reference's
reduce_scatter-with-zero-contributions emulation (layer.py:281-307),
which moved mp_world_size x the bytes over the links.
"""

from __future__ import annotations

from typing import Any
from typing import Literal

import torch
import torch.distributed as dist

from kfac_amd.distributed import get_rank
from kfac_amd.distributed import get_world_size
from kfac_amd.gpt_neox.mpu import gather_from_model_parallel_region
from kfac_amd.gpt_neox.mpu import split_tensor_along_dim
from kfac_amd.layers.eigen import KFACEigenLayer
from kfac_amd.layers.modules import ModuleHelper


class GPTNeoXKFACEigenLayer(KFACEigenLayer):
    """Model-parallel-aware eigen layer."""

    # The grouped 4-launch precondition fast path bypasses the
    # gather/scatter protocol, so it must not be used for TP layers.
    grouped_precondition = False

    def __init__(
        self,
        module: ModuleHelper,
        *,
        parallelism: Literal['input', 'output'],
        model_parallel_group: dist.ProcessGroup | None,
        data_parallel_group: dist.ProcessGroup | None | int = -1,
        pipe_parallel_peer_group: dist.ProcessGroup | None | int = -1,
        primary_rank: int | None = None,
        **kwargs: Any,
    ) -> None:
        """Init GPTNeoXKFACEigenLayer.

        Args:
            module: module helper (GPTNeoXLinearModuleHelper).
            parallelism: 'input' (RowParallelLinear) or 'output'
                (ColumnParallelLinear) sharding.
            model_parallel_group: TP group of this rank.
            data_parallel_group: DP group (set later if -1).
            pipe_parallel_peer_group: same-pipe-stage rank group.
            primary_rank: rank that gathers/computes/scatters for this
                layer (set post-assignment).
            **kwargs: forwarded to KFACEigenLayer.
        """
        self.parallelism = parallelism
        self.primary_rank = primary_rank
        self.model_parallel_group = model_parallel_group
        self.data_parallel_group = data_parallel_group
        self.pipe_parallel_peer_group = pipe_parallel_peer_group
        super().__init__(module=module, **kwargs)

    def _check_groups(self) -> None:
        if self.primary_rank is None:
            raise RuntimeError('primary rank has not been set yet.')
        valid = (dist.ProcessGroup, type(None))
        if not isinstance(self.data_parallel_group, valid) or not isinstance(
            self.pipe_parallel_peer_group,
            valid,
        ):
            raise RuntimeError(
                'data_parallel_group or pipe_parallel_peer_group has not '
                'been set yet.',
            )

    # -- factor accumulation ------------------------------------------------

    def save_layer_input(self, input_: list[torch.Tensor]) -> None:
        """Gather sharded input to the primary rank, then accumulate."""
        if self.primary_rank is None:
            raise RuntimeError('primary rank has not been set yet.')
        if self.parallelism == 'input':
            a = gather_from_model_parallel_region(
                input_[0],
                dst=self.primary_rank,
                model_parallel_group=self.model_parallel_group,
            )
            if a is not None:
                super().save_layer_input([a])
        else:
            super().save_layer_input(input_)

    def save_layer_grad_output(
        self,
        grad_output: tuple[torch.Tensor, ...],
    ) -> None:
        """Gather sharded output-grad to the primary rank, then accumulate."""
        if self.primary_rank is None:
            raise RuntimeError('primary rank has not been set yet.')
        if self.parallelism == 'output':
            g = gather_from_model_parallel_region(
                grad_output[0],
                dst=self.primary_rank,
                model_parallel_group=self.model_parallel_group,
            )
            if g is not None:
                super().save_layer_grad_output((g,))
        else:
            super().save_layer_grad_output(grad_output)

    # -- factor reduction -----------------------------------------------------

    def reduce_a_factor(self, group: dist.ProcessGroup | None = None) -> None:
        """Route the A reduction by parallelism (reference layer.py:61-93).

        The sharded-dim factor exists only on primary ranks, which
        average it over the data-parallel group; the unsharded factor is
        averaged over the pipe-peer group by everyone.
        """
        self._check_groups()
        if self.parallelism == 'input':
            if get_rank() != self.primary_rank:
                return
            super().reduce_a_factor(self.data_parallel_group)  # type: ignore[arg-type]
        elif self.parallelism == 'output':
            super().reduce_a_factor(self.pipe_parallel_peer_group)  # type: ignore[arg-type]
        else:
            raise AssertionError('Unreachable.')

    def reduce_g_factor(self, group: dist.ProcessGroup | None = None) -> None:
        """Route the G reduction by parallelism (reference layer.py:95-127)."""
        self._check_groups()
        if self.parallelism == 'input':
            super().reduce_g_factor(self.pipe_parallel_peer_group)  # type: ignore[arg-type]
        elif self.parallelism == 'output':
            if get_rank() != self.primary_rank:
                return
            super().reduce_g_factor(self.data_parallel_group)  # type: ignore[arg-type]
        else:
            raise AssertionError('Unreachable.')

    # -- precondition -----------------------------------------------------------

    def preconditioned_grad(self, damping: float = 0.001) -> None:
        """Gather shards -> precondition on primary -> scatter shards back.

        Every rank in the model-parallel group must enter.
        """
        if self.primary_rank is None:
            raise RuntimeError('primary rank has not been set yet.')
        rank = get_rank()
        # None means "no model parallelism" (size 1), matching
        # gather_from_model_parallel_region and the module helper.
        mp_world = (
            1
            if self.model_parallel_group is None
            else get_world_size(self.model_parallel_group)
        )

        if rank == self.primary_rank and (
            self.qa is None
            or self.qg is None
            or (not self.prediv_eigenvalues and self.da is None)
            or (not self.prediv_eigenvalues and self.dg is None)
            or (self.prediv_eigenvalues and self.dgda is None)
        ):
            raise RuntimeError(
                'Eigendecompositions for both A and G have not been computed',
            )

        grad_partition = self.module.get_weight_grad()
        shard_dim = -1 if self.parallelism == 'input' else 0
        grad = gather_from_model_parallel_region(
            grad_partition,
            dst=self.primary_rank,
            model_parallel_group=self.model_parallel_group,
            dim=shard_dim,
        )

        bias_grad: torch.Tensor | None = None
        bias_grad_partition: torch.Tensor | None = None
        if self.module.has_bias():
            bias_grad_partition = self.module.get_bias_grad()
            if self.parallelism == 'output':
                # bias is sharded only under output parallelism
                bias_grad = gather_from_model_parallel_region(
                    bias_grad_partition,
                    dst=self.primary_rank,
                    model_parallel_group=self.model_parallel_group,
                    dim=0,
                )
            else:
                bias_grad = bias_grad_partition

        weight_grads: list[torch.Tensor] | None = None
        bias_grads: list[torch.Tensor] | None = None
        if grad is not None:
            # This rank holds the full gradient: run the Kronecker chain.
            from kfac_amd import ops

            grad_shape = grad.size()
            if self.module.has_bias():
                assert bias_grad is not None
                bias_shape = bias_grad.size()
                grad = torch.cat([grad, bias_grad.view(-1, 1)], 1)
            grad_type = grad.dtype
            if self.prediv_eigenvalues:
                full = ops.precond_eigen(grad, self.qa, self.qg, dgda=self.dgda)
            else:
                full = ops.precond_eigen(
                    grad,
                    self.qa,
                    self.qg,
                    da=self.da,
                    dg=self.dg,
                    damping=damping,
                )
            full = full.to(grad_type)
            if self.module.has_bias():
                weight_grad = full[:, :-1].reshape(grad_shape)
                bias_grad = full[:, -1:].reshape(bias_shape).contiguous()
            else:
                weight_grad = full.reshape(grad_shape)
            weight_grads = list(
                split_tensor_along_dim(
                    weight_grad,
                    mp_world,
                    dim=shard_dim,
                    contiguous_split_chunks=True,
                ),
            )
            if self.module.has_bias() and self.parallelism == 'output':
                assert bias_grad is not None
                bias_grads = list(
                    split_tensor_along_dim(
                        bias_grad, mp_world, dim=0,
                        contiguous_split_chunks=True,
                    ),
                )

        if mp_world > 1:
            # True scatter over xGMI p2p (RCCL send/recv).
            recv = torch.empty_like(grad_partition)
            dist.scatter(
                recv,
                weight_grads if rank == self.primary_rank else None,
                src=self.primary_rank,
                group=self.model_parallel_group,
            )
            grad_partition = recv
        else:
            assert weight_grads is not None
            grad_partition = weight_grads[0]

        if self.module.has_bias():
            assert bias_grad_partition is not None
            if mp_world > 1:
                if self.parallelism == 'output':
                    recv_b = torch.empty_like(bias_grad_partition)
                    dist.scatter(
                        recv_b,
                        bias_grads if rank == self.primary_rank else None,
                        src=self.primary_rank,
                        group=self.model_parallel_group,
                    )
                    bias_grad = recv_b
                else:
                    if rank != self.primary_rank:
                        bias_grad = torch.empty_like(bias_grad_partition)
                    assert bias_grad is not None and bias_grad.is_contiguous()
                    dist.broadcast(
                        bias_grad,
                        src=self.primary_rank,
                        group=self.model_parallel_group,
                    )
            assert bias_grad is not None
            self.grad = torch.cat([grad_partition, bias_grad.view(-1, 1)], 1)
        else:
            self.grad = grad_partition
