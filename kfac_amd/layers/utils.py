"""Covariance / reshape utilities (parity: reference kfac/layers/utils.py).

Kept as a user-visible API for compatibility; the hot path uses the fused
accumulating ops in kfac_amd.ops instead.
"""

from __future__ import annotations

import torch

from kfac_amd.ops.reference import append_bias_ones  # noqa: F401 (re-export)


def get_cov(
    a: torch.Tensor,
    b: torch.Tensor | None = None,
    scale: float | None = None,
) -> torch.Tensor:
    """Empirical second moment a^T @ (b or a) / scale, symmetrized.

    Reference kfac/layers/utils.py:18-59.
    """
    if len(a.shape) != 2:
        raise ValueError(
            f'Input tensor must have 2 dimensions. Got tensor with shape '
            f'{a.shape}',
        )
    if b is not None and a.shape != b.shape:
        raise ValueError(
            f'Input tensors must have same shape. Got tensors of shape '
            f'{a.shape} and {b.shape}.',
        )
    if scale is None:
        scale = a.size(0)
    if b is None:
        cov = a.t() @ (a / scale)
        return (cov + cov.t()) / 2.0
    return a.t() @ (b / scale)


def reshape_data(
    data_list: list[torch.Tensor],
    batch_first: bool = True,
    collapse_dims: bool = False,
) -> torch.Tensor:
    """Concat tensors along the batch dim; optionally collapse to 2D.

    Reference kfac/layers/utils.py:62-83.
    """
    d = torch.cat(data_list, dim=int(not batch_first))
    if collapse_dims and len(d.shape) > 2:
        d = d.view(-1, d.shape[-1])
    return d
