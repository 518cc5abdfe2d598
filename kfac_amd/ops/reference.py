"""Pure-torch reference implementations of every K-FAC compute op.

These are the numerics ground truth for the HIP kernels in ``csrc/`` and the
execution path on CPU. Semantics match the reference package
(kfac/layers/utils.py:18-59, kfac/layers/modules.py:123-192) but are fused
differently: the covariance op takes an explicit ``coeff``/``beta`` pair so
the EMA factor update (reference kfac/layers/base.py:375-405) and the
conv spatial scaling fold into a single accumulation epilogue — on MI355X
the HIP kernel does all of this in one pass over HBM.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def append_bias_ones(tensor: torch.Tensor) -> torch.Tensor:
    """Append a column of ones to the last dim (reference utils.py:8-15)."""
    shape = list(tensor.shape[:-1]) + [1]
    return torch.cat([tensor, tensor.new_ones(shape)], dim=-1)


def cov_linear(
    a: torch.Tensor,
    *,
    bias: bool,
    out: torch.Tensor,
    beta: float,
    coeff: float,
) -> torch.Tensor:
    """out = beta * out + coeff * (a'^T @ a'), a' = [a, 1] if bias.

    ``a`` is 2D (rows = batch*seq, cols = in/out features), any float dtype;
    ``out`` is square fp32 and is updated in place. The product is
    symmetrized, matching reference get_cov (utils.py:55-57).
    """
    a = a.reshape(-1, a.shape[-1])
    if bias:
        a = append_bias_ones(a)
    a32 = a.to(out.dtype)
    cov = a32.t() @ a32
    cov = (cov + cov.t()).mul_(0.5 * coeff)
    if beta == 0.0:
        out.copy_(cov)
    else:
        out.mul_(beta).add_(cov)
    return out


def extract_patches(
    x: torch.Tensor,
    kernel_size: tuple[int, int],
    stride: tuple[int, int],
    padding: tuple[int, int],
) -> torch.Tensor:
    """im2col: (N, C, H, W) -> (N, oh, ow, C*kh*kw).

    Matches reference Conv2dModuleHelper._extract_patches
    (kfac/layers/modules.py:210-237).
    """
    if padding[0] + padding[1] > 0:
        x = F.pad(x, (padding[1], padding[1], padding[0], padding[0]))
    x = x.unfold(2, kernel_size[0], stride[0])
    x = x.unfold(3, kernel_size[1], stride[1])
    x = x.permute(0, 2, 3, 1, 4, 5).contiguous()
    return x.view(x.size(0), x.size(1), x.size(2), -1)


def cov_conv_a(
    x: torch.Tensor,
    *,
    kernel_size: tuple[int, int],
    stride: tuple[int, int],
    padding: tuple[int, int],
    bias: bool,
    out: torch.Tensor,
    beta: float,
    coeff_scale: float = 1.0,
) -> torch.Tensor:
    """A-factor contribution of a Conv2d input.

    Equivalent to reference get_a_factor (modules.py:170-178):
    patches / spatial_size, append ones (also divided), cov with
    scale = N*oh*ow. Folded here into a single coefficient
    ``coeff_scale / (M * s^2)`` applied to the raw patch product.
    """
    patches = extract_patches(x, kernel_size, stride, padding)
    s = patches.size(1) * patches.size(2)
    a = patches.view(-1, patches.size(-1))
    m = a.size(0)
    if bias:
        a = append_bias_ones(a)
    coeff = coeff_scale / (m * s * s)
    return cov_linear(a, bias=False, out=out, beta=beta, coeff=coeff)


def cov_conv_g(
    g: torch.Tensor,
    *,
    out: torch.Tensor,
    beta: float,
    coeff_scale: float = 1.0,
) -> torch.Tensor:
    """G-factor contribution of a Conv2d output-gradient (NCHW).

    Reference get_g_factor (modules.py:180-192): NCHW -> rows (N*oh*ow, O),
    divided by spatial size, cov with scale = rows.
    """
    s = g.size(2) * g.size(3)
    rows = g.permute(0, 2, 3, 1).reshape(-1, g.size(1))
    m = rows.size(0)
    coeff = coeff_scale / (m * s * s)
    return cov_linear(rows, bias=False, out=out, beta=beta, coeff=coeff)


def precond_eigen(
    grad: torch.Tensor,
    qa: torch.Tensor,
    qg: torch.Tensor,
    *,
    dgda: torch.Tensor | None = None,
    da: torch.Tensor | None = None,
    dg: torch.Tensor | None = None,
    damping: float = 0.0,
) -> torch.Tensor:
    """Kronecker-eigenbasis preconditioning (reference eigen.py:350-385).

    v1 = QG^T @ grad @ QA; v2 = v1 * dGdA (or v1 / (outer(dG, dA)+damping));
    out = QG @ v2 @ QA^T, cast back to grad dtype.
    """
    g32 = grad.to(qa.dtype)
    v1 = qg.t() @ g32 @ qa
    if dgda is not None:
        v2 = v1 * dgda
    else:
        assert da is not None and dg is not None
        v2 = v1 / (torch.outer(dg, da) + damping)
    return (qg @ v2 @ qa.t()).to(grad.dtype)


def precond_inverse(
    grad: torch.Tensor,
    a_inv: torch.Tensor,
    g_inv: torch.Tensor,
) -> torch.Tensor:
    """Explicit-inverse preconditioning: G^-1 @ grad @ A^-1 (inverse.py:215-234)."""
    g32 = grad.to(g_inv.dtype)
    return (g_inv @ g32 @ a_inv.to(g_inv.dtype)).to(grad.dtype)


def eigh(x: torch.Tensor, *, clamp: bool = True) -> tuple[torch.Tensor, torch.Tensor]:
    """Symmetric eigendecomposition in fp32 (reference eigen.py:309-344).

    Returns (d, Q) with eigenvalues ascending; eigenvalues clamped >= 0.
    """
    dt = torch.float64 if x.dtype == torch.float64 else torch.float32
    d, q = torch.linalg.eigh(x.to(dt))
    if clamp:
        d = torch.clamp(d, min=0.0)
    # rocSOLVER can return the eigenvector matrix as a transposed view;
    # downstream kernels and collectives need contiguous storage.
    return d.contiguous(), q.contiguous()


def inv_damped(x: torch.Tensor, damping: float) -> torch.Tensor:
    """(x + damping*I)^-1 via Cholesky in fp32 (reference inverse.py:186-213).

    Falls back to LU inverse if the damped matrix is not positive definite.
    """
    dt = torch.float64 if x.dtype == torch.float64 else torch.float32
    xd = x.to(dt).clone()
    torch.diagonal(xd).add_(damping)
    try:
        chol = torch.linalg.cholesky(xd)
        return torch.cholesky_inverse(chol)
    except Exception:  # singular / not PD: LU fallback
        return torch.linalg.inv(xd)


def triu_pack(x: torch.Tensor) -> torch.Tensor:
    """Pack the upper triangle of a square matrix into a flat vector.

    Wire format for symmetric factors (reference distributed.py:422-446);
    halves bytes on xGMI.
    """
    if x.dim() != 2 or x.size(0) != x.size(1):
        raise ValueError(f'Expected square 2D tensor, got {tuple(x.shape)}')
    n = x.size(0)
    idx = torch.triu_indices(n, n, device=x.device)
    return x[idx[0], idx[1]].contiguous()


def triu_unpack(v: torch.Tensor, n: int) -> torch.Tensor:
    """Inverse of triu_pack: rebuild the full symmetric matrix."""
    out = v.new_empty((n, n))
    idx = torch.triu_indices(n, n, device=v.device)
    out[idx[0], idx[1]] = v
    out.t()[idx[0], idx[1]] = v
    return out
