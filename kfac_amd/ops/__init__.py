"""Device-dispatched K-FAC compute ops.

Every hot op in the K-FAC pipeline goes through this module:

- On MI355X (ROCm CUDA device) the ops call the in-tree HIP extension
  ``kfac_amd._kfaccore`` (hand-written CDNA4/gfx950 kernels: fused
  im2col+SYRK+EMA covariance, fused Kronecker precondition, fused kl-clip
  reduction, triu pack/unpack). If the extension is missing on a GPU the
  ops raise instead of silently falling back to eager torch — set
  ``KFAC_AMD_ALLOW_EAGER=1`` only for debugging.
- On CPU the pure-torch implementations in ``kfac_amd.ops.reference`` run;
  they are also the numerics ground truth the HIP kernels are tested
  against (tests/test_ops_gpu.py).

Dense eigendecomposition and Cholesky inverse go through torch.linalg
(rocSOLVER/hipSOLVER on ROCm) for large factors; small factors use the
batched device path when available.
"""

from __future__ import annotations

import os
from typing import Any

import torch

from kfac_amd.ops import reference as ref

_EXT: Any = None
_EXT_TRIED = False


def _load_ext() -> Any:
    """Import the in-tree HIP extension (kfac_amd/_kfaccore*.so).

    KFAC_AMD_FORCE_EAGER=1 disables the extension entirely — used only to
    benchmark the torch-eager baseline (the reference implementation's
    op-for-op algorithm) on the same hardware.
    """
    global _EXT, _EXT_TRIED
    if os.environ.get('KFAC_AMD_FORCE_EAGER', '0') == '1':
        return None
    if not _EXT_TRIED:
        _EXT_TRIED = True
        try:
            from kfac_amd import _kfaccore  # type: ignore[attr-defined]

            _EXT = _kfaccore
        except ImportError:
            _EXT = None
    return _EXT


def extension_available() -> bool:
    """True if the HIP extension is importable."""
    return _load_ext() is not None


def _require_ext(op: str) -> Any:
    ext = _load_ext()
    if ext is None:
        if (
            os.environ.get('KFAC_AMD_ALLOW_EAGER', '0') == '1'
            or os.environ.get('KFAC_AMD_FORCE_EAGER', '0') == '1'
        ):
            return None
        raise RuntimeError(
            f'kfac_amd op {op!r} called on a GPU tensor but the HIP '
            'extension kfac_amd._kfaccore is not built. Run '
            'python -c "import __graft_entry__; __graft_entry__.build()" '
            'or set KFAC_AMD_ALLOW_EAGER=1 (debug only).',
        )
    return ext


def cov_linear(
    a: torch.Tensor,
    *,
    bias: bool,
    out: torch.Tensor,
    beta: float,
    coeff: float,
) -> torch.Tensor:
    """out = beta*out + coeff * ([a,1]^T [a,1]); fused SYRK + EMA epilogue."""
    if a.is_cuda:
        ext = _require_ext('cov_linear')
        if ext is not None:
            a2 = a.reshape(-1, a.shape[-1])
            ext.cov_linear(a2, out, bias, beta, coeff)
            return out
    return ref.cov_linear(a, bias=bias, out=out, beta=beta, coeff=coeff)


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
    """Fused im2col + SYRK + EMA: A-factor contribution of a conv input."""
    if x.is_cuda:
        ext = _require_ext('cov_conv_a')
        if ext is not None:
            ext.cov_conv_a(
                x.contiguous(),
                out,
                kernel_size[0],
                kernel_size[1],
                stride[0],
                stride[1],
                padding[0],
                padding[1],
                bias,
                beta,
                coeff_scale,
            )
            return out
    return ref.cov_conv_a(
        x,
        kernel_size=kernel_size,
        stride=stride,
        padding=padding,
        bias=bias,
        out=out,
        beta=beta,
        coeff_scale=coeff_scale,
    )


def cov_conv_g(
    g: torch.Tensor,
    *,
    out: torch.Tensor,
    beta: float,
    coeff_scale: float = 1.0,
) -> torch.Tensor:
    """Fused NCHW-transpose + SYRK + EMA: G-factor contribution of a conv."""
    if g.is_cuda:
        ext = _require_ext('cov_conv_g')
        if ext is not None:
            ext.cov_conv_g(g.contiguous(), out, beta, coeff_scale)
            return out
    return ref.cov_conv_g(g, out=out, beta=beta, coeff_scale=coeff_scale)


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
    """QG^T @ grad @ QA -> elementwise -> QG @ v @ QA^T, one fused chain."""
    if grad.is_cuda:
        ext = _require_ext('precond_eigen')
        if ext is not None:
            # the device kernels and the xf32 library chain are fp32;
            # non-fp32 inv_dtype state is cast here (no-op by default)
            qa = qa.to(torch.float32)
            qg = qg.to(torch.float32)
            big = (
                chain_flops(grad.size(0), grad.size(1))
                > CHAIN_FLOPS_XF32_THRESHOLD
            )
            if dgda is not None:
                dgda = dgda.to(torch.float32)
                if big:
                    return precond_eigen_xf32(grad, qa, qg, dgda)
                return ext.precond_eigen_fused(
                    grad.contiguous(),
                    qa.contiguous(),
                    qg.contiguous(),
                    dgda.contiguous(),
                )
            assert da is not None and dg is not None
            da = da.to(torch.float32)
            dg = dg.to(torch.float32)
            if big:
                from kfac_amd.ops import blocked

                g32 = grad.to(torch.float32)
                with blocked.gemm_engine(True):
                    v1 = (qg.transpose(-1, -2) @ g32) @ qa
                    v2 = v1 / (torch.outer(dg, da) + damping)
                    out = (qg @ v2) @ qa.transpose(-1, -2)
                return out.to(grad.dtype)
            return ext.precond_eigen(
                grad.contiguous(),
                qa.contiguous(),
                qg.contiguous(),
                dg.contiguous(),
                da.contiguous(),
                float(damping),
            )
    return ref.precond_eigen(
        grad, qa, qg, dgda=dgda, da=da, dg=dg, damping=damping,
    )


# Layers whose chain GEMMs exceed this flop count run on the hipBLASLt
# xf32 path (305 TF at n=4608 vs ~70-135 TF for the in-house split
# kernel at skinny shapes, gpurun_out/gemm_rates.txt); smaller layers
# stay in the grouped single-launch chain where launch count dominates.
CHAIN_FLOPS_XF32_THRESHOLD = 2.0e10


def chain_flops(m: int, n: int) -> float:
    """Flops of the 4-GEMM Kronecker chain for an (m, n) gradient."""
    return 4.0 * m * n * (m + n)


def precond_eigen_xf32(
    grad: torch.Tensor,
    qa: torch.Tensor,
    qg: torch.Tensor,
    dgda: torch.Tensor,
) -> torch.Tensor:
    """Kronecker precondition chain on the hipBLASLt xf32 engine.

    Same math as ``precond_eigen_fused`` (prediv form); used for large
    layers where library GEMM throughput beats the grouped launch
    saving.  ~4.5e-6 relative accuracy (bf16x3 internally), same class
    as the in-house split path.
    """
    from kfac_amd.ops import blocked

    g32 = grad.to(torch.float32)
    with blocked.gemm_engine(True):
        v1 = (qg.transpose(-1, -2) @ g32) @ qa
        v2 = v1 * dgda
        out = (qg @ v2) @ qa.transpose(-1, -2)
    return out.to(grad.dtype)


def precond_eigen_grouped(
    grads: list[torch.Tensor],
    qas: list[torch.Tensor],
    qgs: list[torch.Tensor],
    dgdas: list[torch.Tensor],
) -> list[torch.Tensor]:
    """Whole eigen precondition chain for all layers in 4 kernel launches.

    GPU-only (requires the HIP extension); the per-layer path is the
    fallback for CPU or non-prediv configurations.
    """
    ext = _require_ext('precond_eigen_grouped')
    assert ext is not None
    return ext.precond_eigen_grouped(grads, qas, qgs, dgdas)


def precond_apply_grouped(
    weight_grads: list[torch.Tensor],
    bias_grads: list[torch.Tensor],
    qas: list[torch.Tensor],
    qgs: list[torch.Tensor],
    dgdas: list[torch.Tensor],
    kl_clip: float,
    lr: float,
    accum_init: torch.Tensor | None = None,
) -> torch.Tensor:
    """Fused COMM-OPT precondition + kl-clip + in-place grad update.

    ~9 kernel launches for the whole model; returns the applied scale
    (1-elem device tensor). GPU-only.  ``accum_init`` carries kl-clip
    dot contributions from layers preconditioned outside the call
    (the xf32-routed large layers) so the scale covers both sets.
    """
    ext = _require_ext('precond_apply_grouped')
    assert ext is not None
    return ext.precond_apply_grouped(
        weight_grads, bias_grads, qas, qgs, dgdas, kl_clip, lr, accum_init,
    )


def precond_inverse(
    grad: torch.Tensor,
    a_inv: torch.Tensor,
    g_inv: torch.Tensor,
) -> torch.Tensor:
    """G^-1 @ grad @ A^-1."""
    if grad.is_cuda:
        ext = _require_ext('precond_inverse')
        if ext is not None:
            return ext.precond_inverse(
                grad.contiguous(), a_inv.contiguous(), g_inv.contiguous(),
            )
    return ref.precond_inverse(grad, a_inv, g_inv)


def kl_clip_accum(
    accum: torch.Tensor,
    precon: torch.Tensor,
    grad: torch.Tensor,
) -> None:
    """accum (0-dim fp32, device) += sum(precon * grad). No host sync.

    Replaces the reference's per-layer ``.sum().item()`` host round-trips
    (base_preconditioner.py:411-435) with a device-side scalar accumulation.
    """
    if precon.is_cuda:
        ext = _require_ext('kl_clip_accum')
        if ext is not None:
            ext.kl_clip_accum(accum, precon.contiguous(), grad.contiguous())
            return
    accum.add_((precon.to(torch.float32) * grad.to(torch.float32)).sum())


def grad_scale_from_accum(
    accum: torch.Tensor,
    kl_clip: float,
    lr: float,
) -> torch.Tensor:
    """scale = min(1, sqrt(kl_clip / |accum * lr^2|)) as a device scalar."""
    s = (accum * (lr * lr)).abs()
    return torch.clamp(torch.sqrt(kl_clip / torch.clamp(s, min=1e-30)), max=1.0)


def eigh(x: torch.Tensor, *, clamp: bool = True) -> tuple[torch.Tensor, torch.Tensor]:
    """fp32 symmetric eigendecomposition; eigenvalues clamped >= 0."""
    return ref.eigh(x, clamp=clamp)


def eigh_batched(stack: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Batched fp32 symmetric eigendecomposition of a (B, n, n) stack.

    On CUDA: n <= 64 routes to the hand-written one-wave-per-matrix LDS
    Jacobi kernel (eigenvalues unsorted — K-FAC is order-invariant);
    larger n routes to rocSOLVER syevd.  This is the COLD path: the
    preconditioner's inverse phase uses the warm-started block-Jacobi
    solver (ops/warm_eigh.py) whenever the previous phase's eigenbasis
    is available.  (Round 1 wrapped syevd in a hipGraph replay; that
    was removed in round 2 after it was shown unsound — syevd's
    tridiagonal iteration launches a data-dependent kernel sequence,
    see profiles/jacobi_warm.md.)  Eigenvalues are NOT clamped here;
    callers clamp >= 0 (reference eigen.py:321,344).
    """
    if stack.is_cuda:
        ext = _load_ext()
        if ext is not None:
            # Near-diagonal screen: G factors of layers whose gradient
            # contributions vanished are EXACTLY (decayed-identity)
            # scalar multiples of I — dense solvers can even fail to
            # converge on their denormal off-diagonals.  diag + I is
            # the exact answer (error <= off-mass <= 1e-7 ||F||).
            d = stack.diagonal(dim1=-2, dim2=-1)
            off = torch.linalg.norm(
                (stack - torch.diag_embed(d)).reshape(stack.size(0), -1),
                dim=-1,
            )
            tn = torch.linalg.norm(
                stack.reshape(stack.size(0), -1), dim=-1,
            ).clamp_min(1e-30)
            if bool((off <= 1e-7 * tn).all()):
                n = stack.size(1)
                eye = torch.eye(
                    n, dtype=stack.dtype, device=stack.device,
                ).expand_as(stack).contiguous()
                return d.clone(), eye
            if stack.size(1) <= 64:
                return ext.syevj_small(stack.contiguous(), 20, 1e-5)
            w, vt = ext.syevd_batched(stack.contiguous())
            return w, vt.transpose(1, 2)
    return torch.linalg.eigh(stack)


def inv_damped(x: torch.Tensor, damping: float) -> torch.Tensor:
    """(x + damping I)^-1 in fp32."""
    return ref.inv_damped(x, damping)


def refine_inverse(
    m: torch.Tensor,
    x0: torch.Tensor,
    *,
    max_iters: int = 4,
    tol: float = 1e-6,
) -> tuple[torch.Tensor, bool]:
    """Newton-Schulz refinement of an approximate inverse of SPD ``m``.

    X <- X (2I - M X), quadratic convergence while ||I - M X0|| < 1.
    The INVERSE method's analog of the warm eigensolver: K-FAC factors
    are slowly-drifting EMAs, so the previous phase's damped inverse is
    an excellent X0 and 1-2 adaptive iterations (2 GEMMs each — the
    residual check reuses the M X product) replace a fresh
    factorize+invert whose rocSOLVER panel chain measured only 2-4 TF
    at these shapes vs ~300 TF for GEMMs (profiles/qdwh_bench.md).
    Returns ``(x, ok)`` where ``ok`` certifies
    ||M X - I||_F <= tol * sqrt(n); callers MUST fall back to the exact
    inverse when ``ok`` is False (cold start, damping change, or drift
    too large — a bad X0 DIVERGES quadratically and can never pass the
    certificate).
    """
    n = m.size(-1)
    gate = tol * (float(n) ** 0.5)
    eye = torch.eye(n, dtype=m.dtype, device=m.device)
    x = x0
    ok = False
    for it in range(max_iters + 1):
        r = m @ x
        resid = float(torch.linalg.norm(r - eye))
        if resid <= gate:
            ok = True
            break
        if it == max_iters or not (resid < float(n)):
            # out of budget, or far outside the convergence basin
            break
        x = x @ (2.0 * eye - r)
    # the inverse of an SPD matrix is symmetric; enforce exactly so the
    # triu wire format remains valid
    x = 0.5 * (x + x.transpose(-1, -2))
    return x, ok


def refine_inverse_batched(
    m: torch.Tensor,
    x0: torch.Tensor,
    *,
    max_iters: int = 4,
    tol: float = 1e-6,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Batched Newton-Schulz inverse refinement over a (B, n, n) stack.

    Same contract as :func:`refine_inverse` but vectorized over the
    batch (the grouped inverse phase refines every same-size factor in
    a handful of bmm launches); returns ``(x, ok_mask)`` with a
    per-matrix certificate — callers exact-solve the ``~ok`` subset.
    """
    bsz, n, _ = m.shape
    gate = tol * (float(n) ** 0.5)
    eye = torch.eye(n, dtype=m.dtype, device=m.device).expand_as(m)
    x = x0
    ok = torch.zeros(bsz, dtype=torch.bool, device=m.device)
    for it in range(max_iters + 1):
        r = m @ x
        resid = torch.linalg.norm((r - eye).reshape(bsz, -1), dim=-1)
        # NaN/Inf from a diverging (bad-start) matrix compares False
        ok = resid <= gate
        if it == max_iters or bool(ok.all()):
            break
        x = x @ (2.0 * eye - r)
    x = 0.5 * (x + x.transpose(-1, -2))
    return x, ok


def triu_pack(x: torch.Tensor) -> torch.Tensor:
    """Symmetric-matrix wire format: upper triangle as a flat vector."""
    if x.is_cuda:
        ext = _require_ext('triu_pack')
        if ext is not None:
            return ext.triu_pack(x.contiguous())
    return ref.triu_pack(x)


def triu_unpack(v: torch.Tensor, n: int, out: torch.Tensor | None = None) -> torch.Tensor:
    """Rebuild the full symmetric matrix from its packed upper triangle."""
    if v.is_cuda:
        ext = _require_ext('triu_unpack')
        if ext is not None:
            res = ext.triu_unpack(v.contiguous(), n)
            if out is not None:
                out.copy_(res)
                return out
            return res
    res = ref.triu_unpack(v, n)
    if out is not None:
        out.copy_(res)
        return out
    return res
