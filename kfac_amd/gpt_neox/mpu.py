"""Model-parallel communication utilities.

Parity with reference kfac/gpt_neox/mpu.py:9-133. On one MI355X node
every model-parallel peer is one xGMI hop away, so the gather-to-dst is
implemented as an all_gather (single-hop from every peer) + concat on
the destination, matching the reference's choice.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def gather_from_model_parallel_region(
    tensor: torch.Tensor,
    dst: int,
    model_parallel_group: dist.ProcessGroup | None,
    fp32_allreduce: bool = False,
    dim: int = -1,
) -> torch.Tensor | None:
    """Gather model-parallel shards into a full tensor on rank ``dst``.

    Returns the gathered tensor on ``dst``, None elsewhere.
    """
    world_size = (
        1
        if model_parallel_group is None
        else dist.get_world_size(model_parallel_group)
    )
    if world_size == 1:
        return tensor

    dt = tensor.dtype
    if dt == torch.bfloat16 and fp32_allreduce:
        tensor = tensor.float()
    tensor = tensor.contiguous()

    tensor_list = [torch.empty_like(tensor) for _ in range(world_size)]
    dist.all_gather(tensor_list, tensor, group=model_parallel_group)

    if dist.get_rank() == dst:
        output = torch.cat(tensor_list, dim=dim).contiguous()
        if dt == torch.bfloat16 and fp32_allreduce:
            output = output.bfloat16()
        return output
    return None


def get_group_with_rank(rank: int, groups: list[list[int]]) -> list[int]:
    """First group in ``groups`` containing ``rank``.

    Raises:
        ValueError: if no group contains the rank.
    """
    for group in groups:
        if rank in group:
            return group
    raise ValueError(f'Rank {rank} was not in any of the groups.')


def split_tensor_along_dim(
    tensor: torch.Tensor,
    num_partitions: int,
    dim: int,
    contiguous_split_chunks: bool = False,
) -> tuple[torch.Tensor, ...]:
    """Split a tensor into equal partitions along ``dim``.

    Raises:
        ValueError: if the dim size is not divisible by num_partitions.
    """
    dim_size = tensor.size()[dim]
    if dim_size % num_partitions != 0:
        raise ValueError(
            f'Tensor dim {dim} (size={dim_size}) is not divisible '
            f'into {num_partitions} parts.',
        )
    chunks = torch.split(tensor, dim_size // num_partitions, dim=dim)
    if contiguous_split_chunks:
        return tuple(c.contiguous() for c in chunks)
    return tuple(chunks)
