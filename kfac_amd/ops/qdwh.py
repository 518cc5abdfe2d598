"""Spectral divide-and-conquer eigensolver via QDWH polar iterations.

Replaces rocSOLVER ``syevd`` for K-FAC's large factors (reference op:
``torch.linalg.eigh`` at /root/reference/kfac/layers/eigen.py:309-344).

Why this algorithm on MI355X
----------------------------
rocSOLVER syevd is dominated by the one-stage tridiagonalization
(``latrd``): a serial BLAS-2 panel chain that streams the full trailing
matrix once per column — ~4 n^3 bytes of HBM traffic per matrix and tens
of thousands of small launches (measured round 1:
profiles/eigh_strategies.md).  The classical fix (two-stage band
reduction + bulge chasing) was prototyped in round 1
(kfac_amd/ops/two_stage_eigh.py) but its stage-2 back-transform applies
~n^2/(2b) Givens/Householder transforms to the eigenvector matrix —
~2 n^3 rotation-bound (not GEMM-bound) flops that neither MFMA nor LDS
tiling can rescue without a research-grade blocked-WY accumulation.

QDWH-eig (Nakatsukasa & Higham, SIAM J. Sci. Comput. 2013) instead
splits the spectrum with the matrix sign function computed by
dynamically-weighted Halley iterations: every step is a batched
GEMM / Cholesky / triangular-solve — exactly the ops this extension
already runs at MFMA rates — and K-FAC's same-size factor groups
(e.g. 24 x 3072 for GPT-NeoX) batch through every stage.  Subproblems
are padded to a uniform size and solved by the existing batched leaf
solvers (syevd for mid sizes, the one-wave LDS Jacobi for n <= 64).

Numerical contract: K-FAC adds damping ~1e-3 to the eigenvalues, so the
acceptance gate is reconstruction ||V diag(w) V^T - A|| <= ~1e-4 ||A||
and orthogonality ||V^T V - I|| <= ~1e-4.  Eigenvalues within ~l0*||A||
of a split point may be exchanged between the two sides; the analysis in
docs/eigh_qdwh_design.md shows the induced reconstruction error is
bounded by the exchange window, which is far below the damping.
"""

from __future__ import annotations

import math

import torch

__all__ = [
    'halley_coefficients',
    'polar_sign',
    'split_spectrum',
    'eigh_qdwh',
]

# Lower bound assumed for min_i |lambda_i - sigma| / alpha at the start
# of the sign iteration.  Eigenvalues closer than ~L0*alpha to the split
# point may be assigned to either side (bounded error, see module doc).
_L0 = 1e-5
# Cap on the Halley weighting parameter "c": kappa(I + c X^2) ~ c, so
# this bounds the conditioning of the per-iteration Cholesky solve to
# what fp32 handles with ~1e-3 headroom.  Capping only slows (never
# breaks) convergence for |lambda| below the capped threshold.
_C_MAX = 3.0e4


def _h(ll: float) -> float:
    """Dynamic Halley coefficient a = h(l) (QDWH recurrence)."""
    l2 = ll * ll
    gamma = (4.0 * (1.0 - l2) / (l2 * l2)) ** (1.0 / 3.0)
    s = math.sqrt(1.0 + gamma)
    return s + 0.5 * math.sqrt(8.0 - 4.0 * gamma + 8.0 * (2.0 - l2) / (l2 * s))


def _l_for_cmax(c_max: float) -> float:
    """The l whose dynamic coefficients give c(l) == c_max (bisection)."""
    lo, hi = 1e-12, 0.999
    for _ in range(200):
        mid = math.sqrt(lo * hi)
        a = _h(mid)
        b = (a - 1.0) ** 2 / 4.0
        c = a + b - 1.0
        if c > c_max:
            lo = mid
        else:
            hi = mid
    return hi


def halley_coefficients(
    l0: float = _L0,
    c_max: float = _C_MAX,
    tol: float = 1e-7,
    max_iters: int = 12,
) -> list[tuple[float, float, float]]:
    """Precompute the (a, b, c) schedule for the capped dynamic QDWH.

    Pure host-side math — the schedule depends only on (l0, c_max), so
    the device iteration runs a fixed launch sequence with no host
    synchronization.  Iterates until the map has converged (l ~ 1 and
    a ~ 3, the pure-Halley fixed point, for ``tol``-level closeness).
    """
    l_cap = _l_for_cmax(c_max)
    coeffs: list[tuple[float, float, float]] = []
    ll = l0
    for _ in range(max_iters):
        l_eff = max(ll, l_cap)
        a = _h(l_eff)
        b = (a - 1.0) ** 2 / 4.0
        c = a + b - 1.0
        coeffs.append((a, b, c))
        ll = ll * (a + b * ll * ll) / (1.0 + c * ll * ll)
        ll = min(ll, 1.0)
        # Converged when the lower edge has reached ~1: two more pure
        # Halley polishing steps then stop.
        if 1.0 - ll < tol:
            coeffs.append((3.0, 1.0, 3.0))
            coeffs.append((3.0, 1.0, 3.0))
            break
    return coeffs


_COEFFS_CACHE: dict[tuple[float, float], list[tuple[float, float, float]]] = {}


def _coeffs(l0: float, c_max: float) -> list[tuple[float, float, float]]:
    key = (l0, c_max)
    if key not in _COEFFS_CACHE:
        _COEFFS_CACHE[key] = halley_coefficients(l0, c_max)
    return _COEFFS_CACHE[key]


def polar_sign(
    x: torch.Tensor,
    l0: float = _L0,
    c_max: float = _C_MAX,
) -> torch.Tensor:
    """Matrix sign of a batch of SYMMETRIC matrices, pre-scaled to
    spectral radius <= 1.

    ``x`` is (B, n, n) symmetric with ||x||_2 <= 1; returns U ~ sign(x)
    (symmetric orthogonal up to the convergence tolerance).  Iteration:

        X <- (b/c) X + (a - b/c) * X (I + c X^2)^{-1}

    with the capped dynamic (a, b, c) schedule.  All batched.

    On GPU the inner solve runs the blocked-Cholesky engine
    (ops/blocked.py: LDS diagonal-block kernel + batched GEMMs) instead
    of rocSOLVER potrf/potrs (measured 2-4 TF).  Iterations with large
    "c" (kappa(Z) ~ c) stay on exact-fp32 GEMMs; once c <= 100 the
    well-conditioned tail runs the hipBLASLt xf32 split-precision path
    (~2.5x fp32 throughput at ~4.5e-6 relative accuracy).
    """
    from kfac_amd.ops import blocked

    use_blocked = x.is_cuda and blocked._ext() is not None
    eye = torch.eye(x.size(-1), dtype=x.dtype, device=x.device)
    for a, b, c in _coeffs(l0, c_max):
        if use_blocked:
            wide = c <= 100.0
            with blocked.gemm_engine(wide):
                z = torch.baddbmm(eye, x, x, beta=1.0, alpha=c)
            y = blocked.spd_solve_right(
                x, z, tf32_chol=wide, tf32_apply=wide,
            )
        else:
            z = torch.baddbmm(eye, x, x, beta=1.0, alpha=c)
            w = torch.linalg.cholesky(z)
            # y = Z^{-1} X == (X Z^{-1})^T; Z, X symmetric and the
            # iterate is symmetrized below, so the transpose is free.
            y = torch.cholesky_solve(x, w)
        x = (b / c) * x + (a - b / c) * y
        x = 0.5 * (x + x.transpose(-1, -2))
    return x


def _chol_qr(y: torch.Tensor, ridge: float = 0.0) -> torch.Tensor:
    """Orthonormalize the columns of each (n, k) matrix in the batch via
    Cholesky-QR.  Valid when kappa(y)^2 * eps < 1; callers run two
    passes (CholQR2) with a projector re-application in between.  A
    relative ``ridge`` keeps the Gram positive definite when the sketch
    may be ill-conditioned (first pass of a complement basis); the
    second pass scrubs the ridge-induced error."""
    from kfac_amd.ops import blocked

    use_blocked = y.is_cuda and blocked._ext() is not None
    with blocked.gemm_engine(use_blocked):
        g = y.transpose(-1, -2) @ y
    if ridge > 0.0:
        scale = torch.diagonal(g, dim1=-2, dim2=-1).mean(
            dim=-1, keepdim=True,
        )
        g = g + (ridge * scale).unsqueeze(-1) * torch.eye(
            g.size(-1), dtype=g.dtype, device=g.device,
        )
    if use_blocked:
        l, dinvs = blocked.potrf_batched(g, tf32=True, keep_dinv=True)
        t = blocked.trinv_batched(l, dinvs or None, tf32=True)
        with blocked.gemm_engine(True):
            # Y R^{-1} with R = L^T:  Y L^{-T} = Y T^T
            return y @ t.transpose(-1, -2)
    r = torch.linalg.cholesky(g, upper=True)
    return torch.linalg.solve_triangular(r, y, upper=True, left=False)


def _range_basis(
    u: torch.Tensor,
    omega: torch.Tensor,
    k: int,
    refinements: int = 2,
) -> torch.Tensor:
    """Orthonormal basis (n, k) of the dominant-k subspace of
    P = (I - U)/2 via randomized subspace iteration.

    ``refinements`` extra P-applications control the leak of converged
    (p = 1) directions past the cut when partially-converged directions
    (p ~ 0.5 — eigenvalues inside the sign iteration's window at the
    split point) sit near rank k: leak ~ (1/2)^(2*(1+refinements)).
    The production shift choice (max gap of the previous phase's
    spectrum) keeps such directions away from the cut; the refinements
    are defense in depth for stale hints.
    """
    # Pass-1 ridge sized for a SQUARE Gaussian sketch: kappa(Y) ~ 2k with
    # a heavy tail, so the fp32 Gram can be numerically indefinite at
    # production sizes (observed at k ~ 1500).  A 1e-3-relative ridge
    # keeps the Cholesky alive and bounds kappa of the resulting basis
    # to ~sqrt(ridge)/sigma_min; the projector refinements + later
    # passes scrub the ridge-induced error completely.
    y = 0.5 * (omega[:, :k] - u @ omega[:, :k])
    q = _chol_qr(y.unsqueeze(0), ridge=1e-3).squeeze(0)
    for _ in range(refinements):
        y = 0.5 * (q - u @ q)
        q = _chol_qr(y.unsqueeze(0), ridge=1e-6).squeeze(0)
    return _chol_qr(q.unsqueeze(0)).squeeze(0)


def _complement_basis(
    u: torch.Tensor,
    omega: torch.Tensor,
    m: int,
    q1: torch.Tensor,
) -> torch.Tensor:
    """Orthonormal basis (n, m) of the orthogonal complement of ``q1``.

    Seeded with (I + U)/2 omega so the sketch is well aligned with the
    complement, but — deliberately — NOT refined through the projector:
    if eigenvalues sit inside the sign iteration's convergence window,
    range((I+U)/2) does not cover the full complement and a projector
    refinement collapses the missing directions (rank loss).  Any
    orthonormal completion of q1 is exactly as good: the discarded
    coupling Q2^T A Q1 depends only on q1's range accuracy.
    """

    def proj_out(y: torch.Tensor) -> torch.Tensor:
        return y - q1 @ (q1.transpose(-1, -2) @ y)

    y = 0.5 * (omega[:, :m] + u @ omega[:, :m])
    q = _chol_qr(proj_out(y).unsqueeze(0), ridge=1e-3).squeeze(0)
    q = _chol_qr(proj_out(q).unsqueeze(0), ridge=1e-6).squeeze(0)
    return _chol_qr(proj_out(q).unsqueeze(0)).squeeze(0)


def split_spectrum(
    a: torch.Tensor,
    sigma: torch.Tensor,
    generator: torch.Generator | None = None,
) -> tuple[list[torch.Tensor], list[torch.Tensor], list[torch.Tensor], list[torch.Tensor]]:
    """Split each symmetric matrix in the batch at its shift ``sigma``.

    Args:
        a: (B, n, n) symmetric.
        sigma: (B,) split shifts.
        generator: RNG for the randomized range finder.

    Returns:
        (a_lo, q_lo, a_hi, q_hi): per-matrix lists — ``q_lo[i]`` is an
        (n, k_i) orthonormal basis of the (eigenvalue < sigma_i)
        subspace and ``a_lo[i] = q_lo[i]^T a[i] q_lo[i]``; similarly hi.
    """
    bsz, n, _ = a.shape
    x = a - sigma.view(-1, 1, 1) * torch.eye(
        n, dtype=a.dtype, device=a.device,
    )
    # Upper bound on ||X||_2: symmetric => max absolute row sum.
    alpha = x.abs().sum(dim=-1).max(dim=-1).values.clamp_min(1e-30)
    u = polar_sign(x / alpha.view(-1, 1, 1))
    # rank of the lower subspace: P = (I - U)/2, k = round(tr(P))
    tr_u = torch.diagonal(u, dim1=-2, dim2=-1).sum(-1)
    ks = torch.round((n - tr_u) / 2.0).long().clamp(1, n - 1)
    ks_host = ks.tolist()  # one host sync per split level

    from kfac_amd.ops import blocked

    omega = torch.randn(n, n, dtype=a.dtype, device=a.device, generator=generator)
    a_lo: list[torch.Tensor] = []
    q_lo: list[torch.Tensor] = []
    a_hi: list[torch.Tensor] = []
    q_hi: list[torch.Tensor] = []
    use_wide = a.is_cuda and blocked._ext() is not None
    with blocked.gemm_engine(use_wide):
        for i in range(bsz):
            k = int(ks_host[i])
            q1 = _range_basis(u[i], omega, k)
            q2 = _complement_basis(u[i], omega, n - k, q1)
            t1 = a[i] @ q1
            m1 = q1.transpose(-1, -2) @ t1
            t2 = a[i] @ q2
            m2 = q2.transpose(-1, -2) @ t2
            a_lo.append(0.5 * (m1 + m1.transpose(-1, -2)))
            q_lo.append(q1)
            a_hi.append(0.5 * (m2 + m2.transpose(-1, -2)))
            q_hi.append(q2)
    return a_lo, q_lo, a_hi, q_hi


def _median_shift(a: torch.Tensor, hint: torch.Tensor | None) -> torch.Tensor:
    """Split-point estimate per matrix.

    With a ``hint`` (previous phase's eigenvalues — K-FAC factors change
    slowly between phases, so layers pass their cached spectrum), choose
    the midpoint of the LARGEST GAP among the middle 50% of hint values:
    balanced split AND a shift that keeps eigenvalues out of the sign
    iteration's convergence window.  Without a hint, fall back to the
    median of the sorted diagonal — balance is approximate and a cluster
    could in principle sit at the shift (bounded, documented error); the
    K-FAC integration always has hints after the first phase.
    """
    if hint is not None:
        h, _ = torch.sort(hint, dim=-1)
        n = h.size(-1)
        lo = n // 4
        hi = max(lo + 2, (3 * n) // 4)
        window = h[..., lo:hi]
        gaps = window[..., 1:] - window[..., :-1]
        gi = gaps.argmax(dim=-1, keepdim=True)
        left = torch.gather(window, -1, gi).squeeze(-1)
        right = torch.gather(window, -1, gi + 1).squeeze(-1)
        return 0.5 * (left + right)
    d, _ = torch.sort(torch.diagonal(a, dim1=-2, dim2=-1), dim=-1)
    return d[..., d.size(-1) // 2]


def eigh_qdwh(
    stack: torch.Tensor,
    *,
    leaf_size: int = 512,
    max_levels: int = 3,
    leaf_fn=None,
    shift_hint: torch.Tensor | None = None,
    generator: torch.Generator | None = None,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Batched symmetric eigendecomposition by spectral divide-and-conquer.

    Args:
        stack: (B, n, n) symmetric fp32.
        leaf_size: subproblems at or below this size go to ``leaf_fn``.
        max_levels: maximum number of split levels.
        leaf_fn: ``f(stack) -> (w, v)`` batched dense eigensolver for the
            leaves (default torch.linalg.eigh; on GPU the caller passes
            ops.eigh_batched which routes to syevd / the LDS Jacobi).
        shift_hint: (B, n) previous eigenvalues for split-point choice.
        generator: RNG for the randomized range finders.

    Returns:
        (w, v): (B, n) ascending eigenvalues, (B, n, n) eigenvectors in
        columns — torch.linalg.eigh convention.
    """
    if leaf_fn is None:
        leaf_fn = torch.linalg.eigh
    bsz, n, _ = stack.shape
    if n <= leaf_size or max_levels <= 0:
        return leaf_fn(stack)

    # Work items: (matrix index, basis Q mapping subproblem -> original
    # coordinates, subproblem matrix, eigenvalue hint)
    jobs = [
        (
            i,
            None,
            stack[i],
            None if shift_hint is None else shift_hint[i],
        )
        for i in range(bsz)
    ]
    for _level in range(max_levels):
        if all(j[2].size(-1) <= leaf_size for j in jobs):
            break
        next_jobs = []
        # group splittable jobs by size for batching
        by_size: dict[int, list[int]] = {}
        for idx, job in enumerate(jobs):
            sz = job[2].size(-1)
            if sz > leaf_size:
                by_size.setdefault(sz, []).append(idx)
            else:
                next_jobs.append(job)
        for sz, idxs in by_size.items():
            sub = torch.stack([jobs[i][2] for i in idxs])
            hints = None
            if all(jobs[i][3] is not None for i in idxs):
                hints = torch.stack([jobs[i][3] for i in idxs])
            sigma = _median_shift(sub, hints)
            a_lo, q_lo, a_hi, q_hi = split_spectrum(sub, sigma, generator)
            from kfac_amd.ops import blocked as _blocked

            _wide = sub.is_cuda and _blocked._ext() is not None
            for j, i in enumerate(idxs):
                mat_i, q_parent, _, hint_i = jobs[i]
                for a_c, q_c, lo in (
                    (a_lo[j], q_lo[j], True),
                    (a_hi[j], q_hi[j], False),
                ):
                    if q_parent is None:
                        q_full = q_c
                    else:
                        with _blocked.gemm_engine(_wide):
                            q_full = q_parent @ q_c
                    h_c = None
                    if hint_i is not None:
                        s = float(sigma[j])
                        sel = hint_i < s if lo else hint_i >= s
                        hh = hint_i[sel]
                        # hint is advisory; pad/trim to subproblem size
                        k = q_c.size(-1)
                        if hh.numel() >= 1:
                            if hh.numel() > k:
                                hh = hh[:k] if lo else hh[-k:]
                            elif hh.numel() < k:
                                hh = torch.cat(
                                    [hh, hh[-1:].expand(k - hh.numel())],
                                )
                            h_c = hh
                    next_jobs.append((mat_i, q_full, a_c, h_c))
        jobs = next_jobs

    # Solve leaves, batched per size.
    w_out = stack.new_empty(bsz, n)
    v_out = stack.new_empty(bsz, n, n)
    fill: dict[int, int] = {i: 0 for i in range(bsz)}
    by_size = {}
    for idx, job in enumerate(jobs):
        by_size.setdefault(job[2].size(-1), []).append(idx)
    from kfac_amd.ops import blocked as _blk

    for sz, idxs in by_size.items():
        sub = torch.stack([jobs[i][2] for i in idxs])
        w_leaf, v_leaf = leaf_fn(sub)
        _wide = sub.is_cuda and _blk._ext() is not None
        for j, i in enumerate(idxs):
            mat_i, q_parent, _, _ = jobs[i]
            if q_parent is None:
                vec = v_leaf[j]
            else:
                with _blk.gemm_engine(_wide):
                    vec = q_parent @ v_leaf[j]
            k = vec.size(-1)
            off = fill[mat_i]
            w_out[mat_i, off : off + k] = w_leaf[j]
            v_out[mat_i, :, off : off + k] = vec
            fill[mat_i] = off + k

    # ascending order across the concatenated leaf spectra
    w_sorted, perm = torch.sort(w_out, dim=-1)
    v_sorted = torch.gather(
        v_out, 2, perm.unsqueeze(1).expand(bsz, n, n),
    )
    return w_sorted, v_sorted
