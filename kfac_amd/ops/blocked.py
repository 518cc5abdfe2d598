"""Batched blocked Cholesky / triangular inverse on the GEMM engines.

rocSOLVER's potrf / potrs / trsm run at 2-4 TF on MI355X for K-FAC's
batched shapes (latency-bound panel chains, measured in
profiles/qdwh_bench.md) while hipBLASLt GEMMs reach 125 TF (fp32) and
~305 TF (xf32 = bf16x3 split precision, ~4.5e-6 relative accuracy).
These routines restructure the factorizations so that everything O(n^3)
is a batched GEMM and only the O(n * nb^2) diagonal-block work runs in
the hand-written LDS kernel (csrc/chol.hip, one workgroup per matrix).

Engine selection: ``tf32=True`` runs the trailing/apply GEMMs on the
split-precision path.  The polar iterations with large Halley "c" keep
fp32 (kappa(I + c X^2) ~ c makes split precision unsafe there); the
well-conditioned tail iterations and all CholQR/apply GEMMs use tf32.

Used by the QDWH eigensolver (ops/qdwh.py) and the INVERSE compute
method's batched (F + damping I)^{-1} (replacing reference
kfac/layers/inverse.py:186-213).
"""

from __future__ import annotations

import contextlib
from typing import Iterator

import torch

NB = 128


@contextlib.contextmanager
def gemm_engine(tf32: bool) -> Iterator[None]:
    """Scope the hipBLASLt xf32 (bf16x3) path on or off.

    ``allow_tf32`` is PROCESS-GLOBAL state: flipping it from the async
    inverse worker thread while the main thread trains could silently
    run a user's fp32 matmuls at reduced precision.  Off the main
    thread this is therefore a no-op — the async phase's GEMMs run on
    the exact-fp32 engine (2x slower, but that work is overlapped with
    training steps by design).
    """
    import threading

    if threading.current_thread() is not threading.main_thread():
        yield
        return
    prev = torch.backends.cuda.matmul.allow_tf32
    torch.backends.cuda.matmul.allow_tf32 = tf32
    try:
        yield
    finally:
        torch.backends.cuda.matmul.allow_tf32 = prev


def _ext():
    from kfac_amd import ops

    return ops._load_ext()


def potrf_batched(
    a: torch.Tensor,
    *,
    tf32: bool = True,
    keep_dinv: bool = False,
) -> torch.Tensor | tuple[torch.Tensor, list[torch.Tensor]]:
    """Lower Cholesky of a batch of SPD matrices, in one of two modes.

    GPU with extension: blocked right-looking factorization — LDS kernel
    for each diagonal block (which also yields the block's triangular
    inverse), batched GEMMs for the panel (L21 = A21 L11^{-T}) and
    symmetric trailing update.  Otherwise: torch.linalg.cholesky.

    Returns L (lower triangular, same shape); with ``keep_dinv`` also
    the list of diagonal-block inverses for the follow-up triangular
    inverse (saves re-inverting them).

    The input is not modified; the factorization works on a clone.
    """
    ext = _ext()
    if not (a.is_cuda and ext is not None):
        l = torch.linalg.cholesky(a)
        return (l, []) if keep_dinv else l
    bsz, n, _ = a.shape
    w = a.clone()
    dinvs: list[torch.Tensor] = []
    with gemm_engine(tf32):
        for j in range(0, n, NB):
            m = min(NB, n - j)
            dinv = torch.empty(bsz, NB, NB, dtype=a.dtype, device=a.device)
            ext.chol_diag_inv(w, dinv, j, m)
            dinvs.append(dinv)
            if j + m < n:
                a21 = w[:, j + m :, j : j + m]
                # L21 = A21 @ L11^{-T}
                l21 = a21 @ dinv[:, :m, :m].transpose(-1, -2)
                a21.copy_(l21)
                # trailing update (full symmetric form; only the lower
                # triangle is read by later steps)
                w[:, j + m :, j + m :] -= l21 @ l21.transpose(-1, -2)
    l = torch.tril(w)
    return (l, dinvs) if keep_dinv else l


def trinv_batched(
    l: torch.Tensor,
    dinvs: list[torch.Tensor] | None = None,
    *,
    tf32: bool = True,
) -> torch.Tensor:
    """Inverse of a batch of lower-triangular matrices.

    Block recurrence with the diagonal-block inverses from
    :func:`potrf_batched`: row block i of T = L^{-1} is
    ``T[i, :i] = -Dinv_i (L[i, :i] @ T[:i, :i])`` — one batched GEMM
    pair per block row, all O(n^3) on the GEMM engine.
    """
    ext = _ext()
    if not (l.is_cuda and ext is not None):
        return torch.linalg.solve_triangular(
            l, torch.eye(
                l.size(-1), dtype=l.dtype, device=l.device,
            ).expand_as(l).contiguous(), upper=False,
        )
    bsz, n, _ = l.shape
    t = torch.zeros_like(l)
    nblocks = (n + NB - 1) // NB
    if dinvs is None:
        # invert the diagonal blocks with the LDS kernel (on a copy —
        # the kernel also factors, so feed it D D^T whose factor is D).
        dinvs = []
        for bi in range(nblocks):
            j = bi * NB
            m = min(NB, n - j)
            d = l[:, j : j + m, j : j + m]
            work = (d @ d.transpose(-1, -2)).contiguous()
            dinv = torch.empty(bsz, NB, NB, dtype=l.dtype, device=l.device)
            ext.chol_diag_inv(work, dinv, 0, m)
            dinvs.append(dinv)
    with gemm_engine(tf32):
        for bi in range(nblocks):
            j = bi * NB
            m = min(NB, n - j)
            t[:, j : j + m, j : j + m] = dinvs[bi][:, :m, :m]
            if bi > 0:
                acc = l[:, j : j + m, :j] @ t[:, :j, :j]
                t[:, j : j + m, :j] = -dinvs[bi][:, :m, :m] @ acc
    return t


def spd_solve_right(
    x: torch.Tensor,
    z: torch.Tensor,
    *,
    tf32_chol: bool = True,
    tf32_apply: bool = True,
) -> torch.Tensor:
    """Compute X @ Z^{-1} for SPD Z via the blocked factorization:
    Z = L L^T  =>  X Z^{-1} = ((X L^{-T}) L^{-1}), with L^{-1} formed
    explicitly so both applications are plain batched GEMMs."""
    l, dinvs = potrf_batched(z, tf32=tf32_chol, keep_dinv=True)
    t = trinv_batched(l, dinvs or None, tf32=tf32_chol)
    with gemm_engine(tf32_apply):
        return (x @ t.transpose(-1, -2)) @ t


def spd_inverse_batched(
    a: torch.Tensor,
    damping: float = 0.0,
    *,
    tf32: bool = True,
) -> torch.Tensor:
    """Batched (A + damping I)^{-1} for SPD A: the INVERSE compute
    method's factor inversion (K7; reference inverse.py:186-213).

    A^{-1} = L^{-T} L^{-1} = T^T T with T = L^{-1} — Cholesky, blocked
    triangular inverse, one syrk-shaped GEMM; symmetrized exactly.
    """
    z = a
    if damping != 0.0:
        z = a.clone()
        z.diagonal(dim1=-2, dim2=-1).add_(damping)
    ext = _ext()
    if not (a.is_cuda and ext is not None):
        chol = torch.linalg.cholesky(z)
        return torch.cholesky_inverse(chol)
    l, dinvs = potrf_batched(z, tf32=tf32, keep_dinv=True)
    t = trinv_batched(l, dinvs or None, tf32=tf32)
    with gemm_engine(tf32):
        inv = t.transpose(-1, -2) @ t
    return 0.5 * (inv + inv.transpose(-1, -2))
