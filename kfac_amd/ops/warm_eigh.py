"""Warm-started adaptive block-Jacobi eigensolver for K-FAC factors.

K-FAC recomputes the eigendecomposition of each factor every
``inv_update_steps``; between phases the factor is an EMA average whose
eigenbasis drifts SLOWLY.  Measured on real ResNet-50 training
(profiles/jacobi_warm.md): with Q from the previous phase,
T = Q^T F' Q has off-diagonal mass 1e-6..3e-2 of ||F'|| — most factors
need NO rotations at all to meet the 1e-4 reconstruction gate, and the
worst need one adaptive sweep touching a few hundred of the ~2500
dense block pairs.

Algorithm (per same-size factor group, batched over the group):
  1. T = Q_prev^T F' Q_prev           — two batched xf32 GEMMs
  2. diagonal-block pass              — batched in-LDS Jacobi (b x b)
  3. adaptive rounds: pick a maximal matching of block pairs whose
     off-diagonal Frobenius mass matters, solve the 2b x 2b subproblems
     with the in-LDS Jacobi kernel (+ one Newton orthogonality polish),
     apply the rotations as batched GEMMs using T's symmetry
     (T'[E, :] = V^T T[E, :] on full rows; T[:, E] mirrored; T'[E, E]
     = diag from the subproblem), and update Q's columns.
  4. stop when sum of off-block mass <= tol * ||F'||_F per matrix.

Safety: if the warm start is bad (off0 above ``bail_rel`` — e.g. after
a big learning-rate event) the caller falls back to rocSOLVER syevd,
which also serves the cold first phase.  Replaces the reference's
``torch.linalg.eigh`` (kfac/layers/eigen.py:309-344) on the warm path.
"""

from __future__ import annotations

import torch

__all__ = ['warm_eigh_batched']


def _subproblem_eigh(subs: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Batched small symmetric eigensolve (2b <= 64 -> LDS Jacobi)."""
    if subs.is_cuda:
        from kfac_amd import ops

        ext = ops._load_ext()
        if ext is not None and subs.size(-1) <= 64:
            w, v = ext.syevj_small(subs.contiguous(), 30, 1e-7)
            # one Newton polish: v <- v (3I - v^T v) / 2 tightens the
            # kernel's ~1e-5 orthogonality to ~1e-9 so rotation error
            # does not accumulate into Q over many rounds.
            vt_v = v.transpose(-1, -2) @ v
            eye = torch.eye(
                v.size(-1), dtype=v.dtype, device=v.device,
            ).expand_as(vt_v)
            v = v @ (1.5 * eye - 0.5 * vt_v)
            return w, v
    return torch.linalg.eigh(subs)


def _block_off_norms(t: torch.Tensor, b: int) -> torch.Tensor:
    """(B, nb, nb) Frobenius norms of the off-diagonal blocks."""
    bsz, n, _ = t.shape
    nb = n // b
    m = t.reshape(bsz, nb, b, nb, b)
    bn = torch.sqrt((m * m).sum(dim=(2, 4)))
    bn.diagonal(dim1=-2, dim2=-1).zero_()
    return bn


def _apply_diag_pass(
    t: torch.Tensor,
    q: torch.Tensor,
    b: int,
) -> None:
    """Diagonalize every diagonal b x b block (block-diagonal rotation).

    Pair rotations later keep touched diagonal blocks diagonal; this
    pass handles the blocks that are never selected.
    """
    bsz, n, _ = t.shape
    nb = n // b
    diag_blocks = (
        t.reshape(bsz, nb, b, nb, b)
        .diagonal(dim1=1, dim2=3)
        .permute(0, 3, 1, 2)
        .reshape(bsz * nb, b, b)
    )
    _, v = _subproblem_eigh(0.5 * (diag_blocks + diag_blocks.transpose(-1, -2)))
    # rows: T <- V^T T
    rows = t.reshape(bsz * nb, b, n)
    rows.copy_(v.transpose(-1, -2) @ rows)
    # cols: T <- T V
    cols = t.reshape(bsz, n, nb, b).permute(0, 2, 1, 3).reshape(
        bsz * nb, n, b,
    )
    mixed = (cols @ v).reshape(bsz, nb, n, b).permute(0, 2, 1, 3)
    t.copy_(mixed.reshape(bsz, n, n))
    # Q <- Q V
    nq = q.size(1)
    qc = q.reshape(bsz, nq, nb, b).permute(0, 2, 1, 3).reshape(
        bsz * nb, nq, b,
    )
    qm = (qc @ v).reshape(bsz, nb, nq, b).permute(0, 2, 1, 3)
    q.copy_(qm.reshape(bsz, nq, n))


def _greedy_matching(
    bn_row: 'list[list[float]]',
    thresh: float,
    max_pairs: int,
) -> list[tuple[int, int]]:
    """Maximal matching over blocks, heaviest pairs first."""
    cand = []
    nb = len(bn_row)
    for i in range(nb):
        for j in range(i + 1, nb):
            v = bn_row[i][j]
            if v > thresh:
                cand.append((v, i, j))
    cand.sort(reverse=True)
    used: set[int] = set()
    taken: list[tuple[int, int]] = []
    for _, i, j in cand:
        if i not in used and j not in used:
            used.add(i)
            used.add(j)
            taken.append((i, j))
            if len(taken) >= max_pairs:
                break
    return taken


@torch.no_grad()
def warm_eigh_batched(
    stack: torch.Tensor,
    q_prev: torch.Tensor,
    *,
    b: int = 32,
    tol: float = 1e-4,
    bail_rel: float = 0.25,
    max_rounds: int = 150,
) -> tuple[torch.Tensor, torch.Tensor, bool]:
    """Batched warm-started eigendecomposition.

    Args:
        stack: (B, n, n) symmetric fp32 factors.
        q_prev: (B, n, n) previous-phase eigenvector matrices.
        b: block size (2b <= 64 routes subproblems to the LDS kernel).
        tol: stop when per-matrix off-block mass <= tol * ||F||_F.
        bail_rel: give up immediately if the initial off mass exceeds
            this fraction (bad warm start -> caller uses syevd).
        max_rounds: rotation-round budget.

    Returns:
        (d, q, converged): eigenvalue estimates ``d = diag(T)``
        (UNSORTED — aligned with q's columns, which stay maximally
        close to ``q_prev``'s order), eigenvectors ``q``, and whether
        every matrix met ``tol``.  On ``converged=False`` the caller
        should fall back to a dense solve.
    """
    bsz, n_true, _ = stack.shape
    from kfac_amd.ops import blocked

    wide = stack.is_cuda
    with blocked.gemm_engine(wide):
        t = q_prev.transpose(-1, -2) @ stack @ q_prev
    t = 0.5 * (t + t.transpose(-1, -2))

    # pad to a block multiple; pad diagonal entries are decoupled
    # (zero coupling) so they are never selected and never mix.
    n = ((n_true + b - 1) // b) * b
    if n != n_true:
        tp = t.new_zeros(bsz, n, n)
        tp[:, :n_true, :n_true] = t
        scale = (
            t.diagonal(dim1=-2, dim2=-1).abs().amax(dim=-1, keepdim=True)
            + 1.0
        )
        tp.diagonal(dim1=-2, dim2=-1)[:, n_true:] = scale * torch.linspace(
            2.0, 3.0, n - n_true, device=t.device,
        )
        t = tp
        qp = q_prev.new_zeros(bsz, n_true, n)
        qp[:, :, :n_true] = q_prev
        q = qp
    else:
        q = q_prev.clone()

    tn = torch.linalg.norm(stack.reshape(bsz, -1), dim=-1).clamp_min(1e-30)
    nb = n // b

    # quick bail on a bad warm start (one host sync)
    off0 = torch.linalg.norm(
        (t - torch.diag_embed(t.diagonal(dim1=-2, dim2=-1))).reshape(bsz, -1),
        dim=-1,
    )
    if bool((off0 > bail_rel * tn).any()):
        return t.diagonal(dim1=-2, dim2=-1)[:, :n_true], q[:, :, :n_true], False

    _apply_diag_pass(t, q, b)

    converged = False
    for _ in range(max_rounds):
        bn = _block_off_norms(t, b)
        offsq = (bn * bn).sum(dim=(-2, -1))
        # one transfer per round: the block map + the residuals
        bn_host = bn.cpu()
        off_host = torch.sqrt(offsq).cpu()
        pairs: list[tuple[int, int, int]] = []
        all_done = True
        for mi in range(bsz):
            if float(off_host[mi]) <= tol * float(tn[mi]):
                continue
            all_done = False
            thresh = tol * float(tn[mi]) / nb
            for i, j in _greedy_matching(
                bn_host[mi].tolist(), thresh, max_pairs=nb // 2,
            ):
                pairs.append((mi, i, j))
        if all_done:
            converged = True
            break
        if not pairs:
            # residual spread below per-pair threshold but above tol:
            # lower the bar to the heaviest pairs
            for mi in range(bsz):
                if float(off_host[mi]) <= tol * float(tn[mi]):
                    continue
                for i, j in _greedy_matching(
                    bn_host[mi].tolist(), 0.0, max_pairs=nb // 2,
                ):
                    pairs.append((mi, i, j))
            if not pairs:
                converged = True
                break

        p = len(pairs)
        dev = t.device
        idx_local = torch.stack(
            [
                torch.cat(
                    [
                        torch.arange(i * b, (i + 1) * b),
                        torch.arange(j * b, (j + 1) * b),
                    ],
                )
                for _, i, j in pairs
            ],
        ).to(dev)
        mat_idx = torch.tensor([mi for mi, _, _ in pairs], device=dev)
        flat_rows = (mat_idx.unsqueeze(1) * n + idx_local).reshape(-1)

        t_flat = t.reshape(bsz * n, n)
        sub_rows = t_flat.index_select(0, flat_rows).reshape(p, 2 * b, n)
        subs = torch.gather(
            sub_rows, 2, idx_local.unsqueeze(1).expand(p, 2 * b, 2 * b),
        )
        _, v = _subproblem_eigh(
            0.5 * (subs + subs.transpose(-1, -2)),
        )
        # classic parallel block-Jacobi round: T <- V^T T (all pair
        # rows, batched globally), then T <- T V (pair columns, read
        # AFTER the row pass so cross-blocks between two same-matrix
        # pairs get both factors), then Q <- Q V.
        with blocked.gemm_engine(wide):
            new_rows = v.transpose(-1, -2) @ sub_rows
        t_flat.index_copy_(0, flat_rows, new_rows.reshape(p * 2 * b, n))
        by_mat: dict[int, list[int]] = {}
        for pi, (mi, _, _) in enumerate(pairs):
            by_mat.setdefault(mi, []).append(pi)
        for mi, pis in by_mat.items():
            cols = idx_local[pis].reshape(-1)
            tc = (
                t[mi]
                .index_select(1, cols)
                .reshape(n, len(pis), 2 * b)
                .permute(1, 0, 2)
            )
            qs = (
                q[mi]
                .index_select(1, cols)
                .reshape(n_true, len(pis), 2 * b)
                .permute(1, 0, 2)
            )
            with blocked.gemm_engine(wide):
                tr = tc @ v[pis]
                qr = qs @ v[pis]
            t[mi].index_copy_(
                1, cols,
                tr.permute(1, 0, 2).reshape(n, len(pis) * 2 * b),
            )
            q[mi].index_copy_(
                1, cols,
                qr.permute(1, 0, 2).reshape(n_true, len(pis) * 2 * b),
            )

    d = t.diagonal(dim1=-2, dim2=-1)[:, :n_true]
    return d.contiguous(), q[:, :, :n_true].contiguous(), converged
