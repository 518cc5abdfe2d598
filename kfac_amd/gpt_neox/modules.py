"""Module helper for tensor-parallel (sharded) Linear layers.

Parity with reference kfac/gpt_neox/modules.py:13-63: factor shapes are
scaled by the model-parallel world size along the sharded dimension
because factors are computed from *gathered* activations/gradients.
"""

from __future__ import annotations

from typing import Literal

import torch
import torch.distributed as dist

from kfac_amd.layers.modules import LinearModuleHelper


class GPTNeoXLinearModuleHelper(LinearModuleHelper):
    """Helper for Column/RowParallelLinear-style sharded Linear modules."""

    def __init__(
        self,
        module: torch.nn.Module,
        model_parallel_group: dist.ProcessGroup | None,
        parallelism: Literal['input', 'output'],
    ):
        """Init helper.

        Args:
            module: the sharded linear module (weight: [out, in_shard] for
                'input' parallelism, [out_shard, in] for 'output').
            model_parallel_group: TP process group (None -> size 1).
            parallelism: which side of the layer is sharded.
        """
        self.module = module
        self.model_parallel_group = model_parallel_group
        self.model_parallel_world_size = (
            1
            if model_parallel_group is None
            else dist.get_world_size(model_parallel_group)
        )
        self.parallelism = parallelism

    @property
    def a_factor_shape(self) -> tuple[int, int]:
        """A covers the FULL input dim (gathered activations)."""
        dim1 = self.module.weight.size(1)  # type: ignore[operator]
        if self.parallelism == 'input':
            n = dim1 * self.model_parallel_world_size + int(self.has_bias())
        else:
            n = dim1 + int(self.has_bias())
        return (n, n)

    @property
    def g_factor_shape(self) -> tuple[int, int]:
        """G covers the FULL output dim (gathered output-grads)."""
        dim0 = self.module.weight.size(0)  # type: ignore[operator]
        if self.parallelism == 'output':
            n = dim0 * self.model_parallel_world_size
        else:
            n = dim0
        return (n, n)
