"""Two-stage symmetric eigensolver — round-2 groundwork.

Stage 1 (implemented here, torch ops): full symmetric -> symmetric BAND
matrix of bandwidth ``b`` via blocked Householder panels — geqrf on each
sub-panel and compact-WY two-sided application via ormqr. This is the
GEMM-dominated 4/3 n^3 part that replaces rocSOLVER sytrd's
memory-bound latrd chain (the measured inverse-phase bottleneck, see
docs/eigh_two_stage_plan.md); on GPU, geqrf/ormqr dispatch to rocSOLVER's
blocked implementations and the bulk of the flops are plain GEMMs that
round 2 moves onto the split-precision MFMA path.

Stage 2+3 (band -> tridiagonal -> eigenpairs): validated here through
LAPACK's banded solver (scipy eig_banded) on CPU — numerically exact but
CPU-bound (measured 15 s at n=4608, so NOT a production path); round 2
replaces it with a batched bulge-chasing kernel + rocsolver_sstedc.

The module exists so the stage-1 math and the pipeline plumbing
(band layout, Q1 accumulation order, eigenvector back-transform) are
already validated end-to-end (tests/test_two_stage.py) before the
kernels are written.
"""

from __future__ import annotations

import torch

__all__ = [
    'reduce_to_band',
    'reduce_to_band_batched',
    'apply_q1',
    'apply_q1_batched',
    'band_to_tridiag',
    'eigh_two_stage_cpu',
    'eigh_two_stage_self',
]


def reduce_to_band(
    a: torch.Tensor,
    band: int,
) -> tuple[torch.Tensor, list[tuple[int, torch.Tensor, torch.Tensor]]]:
    """Reduce a symmetric matrix to symmetric banded form.

    Returns ``(B, panels)`` where ``B`` is symmetric with bandwidth
    ``band`` (``B[i, j] == 0`` for ``|i - j| > band``), ``panels`` is the
    list of ``(row0, geqrf_a, geqrf_tau)`` Householder panels, and
    ``A == Q1 @ B @ Q1.T`` with ``Q1`` the product of the panel
    reflectors (apply with :func:`apply_q1`).

    The per-panel work is one tall-skinny QR (geqrf) plus two ormqr
    applications to the trailing submatrix — i.e. (I - V T V^T)^T S
    (I - V T V^T), which LAPACK/rocSOLVER evaluate as blocked GEMMs.
    """
    if a.dim() != 2 or a.size(0) != a.size(1):
        raise ValueError(f'expected square matrix, got {tuple(a.shape)}')
    if band < 1:
        raise ValueError('band must be >= 1')
    n = a.size(0)
    b = a.clone()
    panels: list[tuple[int, torch.Tensor, torch.Tensor]] = []
    j = 0
    while j + band < n:
        r0 = j + band
        ncols = min(band, n - r0)  # never wider than the rows below
        panel = b[r0:, j : j + ncols].contiguous()
        qr_a, tau = torch.geqrf(panel)
        k = min(panel.size(0), panel.size(1))
        block = torch.zeros_like(panel)
        block[:k, :] = torch.triu(qr_a[:k, :])
        b[r0:, j : j + ncols] = block
        b[j : j + ncols, r0:] = block.t()
        # Q^T from the left over ALL columns right of the panel and Q
        # from the right over all those rows — with a ragged panel
        # (ncols < band) the strip j+ncols..r0-1 is not yet banded and
        # MUST be transformed too, or the result is not a similarity.
        rest = slice(j + ncols, n)
        b[r0:, rest] = torch.ormqr(
            qr_a, tau, b[r0:, rest].contiguous(), left=True, transpose=True,
        )
        b[rest, r0:] = torch.ormqr(
            qr_a, tau, b[rest, r0:].contiguous(), left=False, transpose=False,
        )
        s = b[r0:, r0:]
        b[r0:, r0:] = 0.5 * (s + s.t())  # exact-symmetry hygiene
        b[r0:, j + ncols : r0] = b[j + ncols : r0, r0:].t()
        panels.append((r0, qr_a, tau))
        j += ncols
    return b, panels


def apply_q1(
    panels: list[tuple[int, torch.Tensor, torch.Tensor]],
    x: torch.Tensor,
) -> torch.Tensor:
    """Compute ``Q1 @ x`` for the band reduction's accumulated basis.

    ``Q1 = Q_p1 @ Q_p2 @ ... @ Q_pk`` (panel order), so the panels apply
    right-to-left; each panel only touches rows ``row0:``.
    """
    out = x.clone()
    for row0, qr_a, tau in reversed(panels):
        out[row0:] = torch.ormqr(
            qr_a, tau, out[row0:].contiguous(), left=True, transpose=False,
        )
    return out


def eigh_two_stage_cpu(
    a: torch.Tensor,
    band: int = 32,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Full two-stage eigendecomposition (CPU validation pipeline).

    Stage 1: :func:`reduce_to_band`; stages 2+3: LAPACK banded solver.
    Returns ``(w, v)`` with the torch.linalg.eigh convention (ascending
    eigenvalues, eigenvectors in columns).
    """
    import numpy as np
    from scipy.linalg import eig_banded

    dt = a.dtype
    b_mat, panels = reduce_to_band(a.to(torch.float64), band)
    n = a.size(0)
    ab = np.zeros((band + 1, n))
    bm = b_mat.numpy()
    for i in range(band + 1):
        ab[i, : n - i] = np.diagonal(bm, -i)
    w, v = eig_banded(ab, lower=True)
    vec = apply_q1(panels, torch.from_numpy(v))
    return torch.from_numpy(w).to(dt), vec.to(dt)


def _givens(f: float, g: float) -> tuple[float, float]:
    """Rotation (c, s) with [[c, s], [-s, c]]^T [f, g]^T = [r, 0]^T."""
    import math

    if g == 0.0:
        return 1.0, 0.0
    r = math.hypot(f, g)
    return f / r, g / r


def band_to_tridiag(
    b_mat: torch.Tensor,
    band: int,
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Stage 2: symmetric band matrix -> tridiagonal via bulge chasing.

    Pure-torch reference of the rotation SCHEDULE the round-2 HIP kernel
    parallelizes (batch-first, wavefront within a matrix): for each
    column j the out-of-tridiagonal band entries are annihilated bottom-
    up with Givens rotations; each rotation spills a bulge one band
    further down, chased off the matrix in strides of ``band``.

    Returns ``(d, e, q2)``: diagonal, subdiagonal, and the accumulated
    orthogonal transform with ``B == q2 @ T @ q2.T``. Dense O(n^3)
    bookkeeping — a validation oracle, not a production path.
    """
    a = b_mat.clone()
    n = a.size(0)
    q2 = torch.eye(n, dtype=a.dtype, device=a.device)

    def rot(p: int, q: int, c: float, s: float) -> None:
        rp = c * a[p, :] + s * a[q, :]
        rq = -s * a[p, :] + c * a[q, :]
        a[p, :], a[q, :] = rp, rq
        cp = c * a[:, p] + s * a[:, q]
        cq = -s * a[:, p] + c * a[:, q]
        a[:, p], a[:, q] = cp, cq
        gp = c * q2[:, p] + s * q2[:, q]
        gq = -s * q2[:, p] + c * q2[:, q]
        q2[:, p], q2[:, q] = gp, gq

    for j in range(n - 2):
        hi = min(j + band, n - 1)
        for i in range(hi, j + 1, -1):
            if float(a[i, j]) == 0.0:
                continue
            # zero B[i, j] against pivot B[i-1, j]
            c, s = _givens(float(a[i - 1, j]), float(a[i, j]))
            rot(i - 1, i, c, s)
            a[i, j] = 0.0
            a[j, i] = 0.0
            # chase the bulge (r+band, r-1) down in strides of band
            r = i
            while r + band < n and float(a[r + band, r - 1]) != 0.0:
                c, s = _givens(
                    float(a[r + band - 1, r - 1]), float(a[r + band, r - 1]),
                )
                rot(r + band - 1, r + band, c, s)
                a[r + band, r - 1] = 0.0
                a[r - 1, r + band] = 0.0
                r += band
    d = torch.diagonal(a, 0).clone()
    e = torch.diagonal(a, -1).clone()
    return d, e, q2


def eigh_two_stage_self(
    a: torch.Tensor,
    band: int = 32,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Fully self-implemented two-stage pipeline (validation oracle):
    stage 1 blocked band reduction, stage 2 bulge chasing, stage 3
    tridiagonal eigensolve, eigenvectors back-transformed Q1 Q2 Z."""
    dt = a.dtype
    a64 = a.to(torch.float64)
    b_mat, panels = reduce_to_band(a64, band)
    d, e, q2 = band_to_tridiag(b_mat, band)
    n = a.size(0)
    tri = (
        torch.diag(d)
        + torch.diag(e, -1)
        + torch.diag(e, 1)
    )
    w, z = torch.linalg.eigh(tri)
    vec = apply_q1(panels, q2 @ z)
    return w.to(dt), vec.to(dt)


def reduce_to_band_batched(
    a: torch.Tensor,
    band: int,
) -> tuple[torch.Tensor, list[tuple[int, torch.Tensor, torch.Tensor]]]:
    """Batched band reduction of a (B, n, n) stack of symmetric matrices.

    Same panel schedule as :func:`reduce_to_band` but every geqrf/ormqr
    is BATCHED over the group — the form the inverse phase needs (K-FAC
    eigendecomposes same-size factor groups: 3x4608, 24x3072, ...) and
    the fix for the measured per-panel launch latency
    (docs/eigh_two_stage_plan.md: 72 sequential panels at b=64 were
    10x slower than 36 at b=128; batching multiplies the work per
    launch by the group size instead).
    """
    if a.dim() != 3 or a.size(1) != a.size(2):
        raise ValueError(f'expected (B, n, n), got {tuple(a.shape)}')
    n = a.size(1)
    b = a.clone()
    panels: list[tuple[int, torch.Tensor, torch.Tensor]] = []
    j = 0
    while j + band < n:
        r0 = j + band
        ncols = min(band, n - r0)
        panel = b[:, r0:, j : j + ncols].contiguous()
        qr_a, tau = torch.geqrf(panel)
        k = min(panel.size(1), panel.size(2))
        block = torch.zeros_like(panel)
        block[:, :k, :] = torch.triu(qr_a[:, :k, :])
        b[:, r0:, j : j + ncols] = block
        b[:, j : j + ncols, r0:] = block.transpose(1, 2)
        rest = slice(j + ncols, n)
        b[:, r0:, rest] = torch.ormqr(
            qr_a, tau, b[:, r0:, rest].contiguous(),
            left=True, transpose=True,
        )
        b[:, rest, r0:] = torch.ormqr(
            qr_a, tau, b[:, rest, r0:].contiguous(),
            left=False, transpose=False,
        )
        s = b[:, r0:, r0:]
        b[:, r0:, r0:] = 0.5 * (s + s.transpose(1, 2))
        b[:, r0:, j + ncols : r0] = b[:, j + ncols : r0, r0:].transpose(1, 2)
        panels.append((r0, qr_a, tau))
        j += ncols
    return b, panels


def apply_q1_batched(
    panels: list[tuple[int, torch.Tensor, torch.Tensor]],
    x: torch.Tensor,
) -> torch.Tensor:
    """Batched ``Q1 @ x`` for :func:`reduce_to_band_batched` output."""
    out = x.clone()
    for row0, qr_a, tau in reversed(panels):
        out[:, row0:] = torch.ormqr(
            qr_a, tau, out[:, row0:].contiguous(),
            left=True, transpose=False,
        )
    return out
