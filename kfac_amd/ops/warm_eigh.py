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
  3. dense band correction: clustered mixing (a contiguous range of
     high-degree blocks) is solved by ONE batched dense sub-eigh and
     applied with GEMMs — pairwise rotations would need >= width/b
     sequential rounds for a clique.
  4. adaptive rounds: maximal matching of block pairs whose
     off-diagonal Frobenius mass matters; 2b x 2b subproblems on the
     in-LDS Jacobi kernel (+ one Newton orthogonality polish); the
     classic parallel round applies T <- V^T T on pair rows (batched
     globally) then T <- T V on pair columns via payload-only
     gather/scatter kernels, then updates Q's columns.
  5. stop when the off-block mass <= tol * ||F'||_F, per matrix.

Safety is PER MATRIX: bad warm starts (off0 above ``bail_rel``),
stalled progress, and broad scattered drift at small n are reported in
the returned mask and the caller dense-solves just those matrices.
Replaces the reference's ``torch.linalg.eigh``
(kfac/layers/eigen.py:309-344) on the warm path.
"""

from __future__ import annotations

import os

import torch

__all__ = ['warm_eigh_batched']


def _subproblem_eigh(subs: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Batched small symmetric eigensolve (2b <= 64 -> LDS Jacobi)."""
    if subs.is_cuda:
        from kfac_amd import ops

        ext = ops._load_ext()
        if ext is not None and subs.size(-1) <= 64:
            w, v = ext.syevj_small(subs.contiguous(), 20, 1e-6)
            # subproblem tolerance 1e-6 (not tighter): rotation
            # accuracy only needs to clear the 1e-4 gates, and the
            # Newton polish below tightens orthogonality quadratically
            # (~1e-6 -> ~1e-11) so rotation error does not accumulate
            # into Q over many rounds.
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


def _dense_band_pass(
    t: torch.Tensor,
    q: torch.Tensor,
    stack: torch.Tensor,
    b: int,
    tol: float,
    tn: torch.Tensor,
    failed: torch.Tensor,
) -> None:
    """Dense correction for clustered off-mass.

    Real K-FAC factors concentrate inter-phase mixing in a contiguous
    eigenvalue range (the decayed-identity cluster re-mixing with fresh
    covariance directions, profiles/jacobi_warm.md).  Pairwise block
    rotations need >= band-width/b rounds for such a clique; instead,
    eigendecompose the active band ONCE (batched dense solve on the
    w x w submatrix) and apply the rotation with GEMMs.  Scattered
    leftovers are handled by the adaptive rounds afterwards.
    """
    bsz, n, _ = t.shape
    nb = n // b
    n_true = q.size(1)
    bn = _block_off_norms(t, b)
    thresh = (tol * tn).view(-1, 1, 1) / nb
    act = bn > thresh
    pair_counts = act.sum(dim=(-2, -1)) // 2
    # band = the contiguous range of HIGH-DEGREE blocks (clique
    # members); a bounding box over all active pairs is inflated past
    # the width cap by scattered outliers, which the adaptive rounds
    # handle anyway.
    degree = act.sum(dim=-1)
    core = degree >= 3
    idx = torch.arange(nb, device=t.device)
    lo = torch.where(core, idx, nb).min(dim=-1).values
    hi = torch.where(core, idx, -1).max(dim=-1).values
    width = (hi - lo + 1).clamp_min(0)
    use = (
        (pair_counts > 2 * width)
        & (width * b <= int(0.7 * n))
        & (width >= 4)
        & ~failed
    )
    if os.environ.get('KFAC_AMD_WARM_TRACE', '0') == '1':
        print(
            f'[warm] n={n} b={b} pairs={pair_counts.tolist()} '
            f'width={width.tolist()} use={use.tolist()}',
        )
    # fail fast on broad scattered drift at sizes where the dense
    # solver is cheap: grinding ~40 rotation rounds only pays off for
    # large n (measured: 3x4608 full-spread converges at 1.4x syevd,
    # but 12x769 full-spread costs 3x MORE than syevd).
    if t.is_cuda and n < 1536:
        failed |= (~use) & (pair_counts > 3 * nb)
    sel = torch.nonzero(use).flatten().tolist()
    if not sel:
        return
    lo_h = lo.tolist()
    hi_h = hi.tolist()
    w = max((hi_h[i] - lo_h[i] + 1) * b for i in sel)
    from kfac_amd import ops as _ops
    from kfac_amd.ops import blocked

    wide = t.is_cuda
    subs = []
    starts = []
    for i in sel:
        c0 = min(lo_h[i] * b, n - w)
        starts.append(c0)
        subs.append(t[i, c0 : c0 + w, c0 : c0 + w])
    sub = torch.stack(subs).contiguous()
    if t.is_cuda:
        _, v = _ops.eigh_batched(sub)
        v = v.contiguous()
    else:
        _, v = torch.linalg.eigh(sub)
    for k, i in enumerate(sel):
        c0 = starts[k]
        with blocked.gemm_engine(wide):
            rows = v[k].transpose(-1, -2) @ t[i, c0 : c0 + w, :]
            t[i, c0 : c0 + w, :] = rows
            t[i, :, c0 : c0 + w] = t[i, :, c0 : c0 + w] @ v[k]
            q[i, :, c0 : c0 + w] = q[i, :, c0 : c0 + w] @ v[k]


@torch.no_grad()
def warm_eigh_batched(
    stack: torch.Tensor,
    q_prev: torch.Tensor,
    *,
    b: int = 32,
    tol: float = 1e-4,
    bail_rel: float = 0.25,
    max_rounds: int = 40,
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
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
        close to ``q_prev``'s order), eigenvectors ``q``, and a
        per-matrix boolean mask of which matrices met ``tol``.  The
        caller solves the unconverged ones densely; converged entries
        are final.
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
        # contiguous_format is load-bearing: eigh_batched returns the
        # eigenvector matrix as a TRANSPOSED view (rocSOLVER writes
        # V^T row-major), and a strides-preserving clone would make
        # q.reshape(...) below a silent COPY — every index_put_ update
        # of Q would then be dropped.
        q = q_prev.clone(memory_format=torch.contiguous_format)

    tn = torch.linalg.norm(stack.reshape(bsz, -1), dim=-1).clamp_min(1e-30)
    nb = n // b

    # per-matrix bail on a bad warm start (one host sync); the rest of
    # the batch proceeds — one cold matrix must not sink its group.
    off0 = torch.linalg.norm(
        (t - torch.diag_embed(t.diagonal(dim1=-2, dim2=-1))).reshape(bsz, -1),
        dim=-1,
    )
    failed = off0 > bail_rel * tn
    if bool(failed.all()):
        return (
            t.diagonal(dim1=-2, dim2=-1)[:, :n_true],
            q[:, :, :n_true],
            ~failed,
        )

    _apply_diag_pass(t, q, b)
    _dense_band_pass(t, q, stack, b, tol, tn, failed)

    dev = t.device
    tol_sq = (tol * tn) ** 2
    arange_b = torch.arange(b, device=dev)
    entry_offsq: torch.Tensor | None = None
    for rnd in range(max_rounds):
        bn = _block_off_norms(t, b)
        offsq = (bn * bn).sum(dim=(-2, -1))
        if entry_offsq is None:
            entry_offsq = offsq.clamp_min(1e-30)
        elif rnd in (8, 16, 24, 32):
            # progress check: matrices whose off mass is not shrinking
            # go to the dense solver instead of grinding the budget.
            # Thresholds tuned on real ResNet-50 factor groups
            # (profiles/jacobi_warm.md): the legitimate heavy cases
            # (identity-decay clusters re-mixing) converge in ~20-30
            # rounds with steady 10-20%/round reduction, while truly
            # cold starts stall near 1.0.
            limit = {8: 0.25, 16: 0.04, 24: 6.4e-3, 32: 1e-3}[rnd]
            failed = failed | (
                (offsq > tol_sq) & ((offsq / entry_offsq) > limit)
            )
        active = (offsq > tol_sq) & ~failed
        # candidate extraction on device, ONE small host transfer of the
        # (typically short) thresholded candidate list per round
        bnsq = bn * bn
        cand_mask = torch.triu(
            bnsq > (tol_sq / (nb * nb)).view(-1, 1, 1), diagonal=1,
        ) & active.view(-1, 1, 1)
        cand_idx = cand_mask.nonzero()
        if cand_idx.numel() == 0:
            if not bool(active.any()):
                break
            # residual above tol but spread below the per-pair bar:
            # take the heaviest pairs of the active matrices
            topv, topi = bnsq.reshape(bsz, -1).topk(nb, dim=-1)
            rows = []
            for mi in torch.nonzero(active).flatten().tolist():
                for r in range(nb):
                    i, j = divmod(int(topi[mi, r]), nb)
                    if i < j and float(topv[mi, r]) > 0:
                        rows.append((float(topv[mi, r]), mi, i, j))
            cand_host = [(mi, i, j) for _, mi, i, j in sorted(rows, reverse=True)]
        else:
            vals = bnsq[cand_idx[:, 0], cand_idx[:, 1], cand_idx[:, 2]]
            order = torch.argsort(vals, descending=True)
            cand_host = cand_idx[order].tolist()
        if not cand_host:
            break
        # greedy maximal matching per matrix (host, short list); cap the
        # round size to bound the gather scratch (pairs left over are
        # picked up by the next round's fresh norms)
        used: set[tuple[int, int]] = set()
        pairs: list[tuple[int, int, int]] = []
        for mi, i, j in cand_host:
            if (mi, i) in used or (mi, j) in used:
                continue
            used.add((mi, i))
            used.add((mi, j))
            pairs.append((mi, i, j))
            if len(pairs) >= 512:
                break

        p = len(pairs)
        if rnd < 3 and os.environ.get('KFAC_AMD_WARM_TRACE', '0') == '1':
            print(f'[warm]   round {rnd}: {p} pairs')
        pair_t = torch.tensor(pairs, device=dev)
        idx_local = torch.cat(
            [
                pair_t[:, 1:2] * b + arange_b,
                pair_t[:, 2:3] * b + arange_b,
            ],
            dim=1,
        )  # (p, 2b)
        mat_idx = pair_t[:, 0].contiguous()
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
        # pairs get both factors), then Q <- Q V.  Column updates use
        # advanced indexing on the flat (B n, n) views so the whole
        # round is a fixed number of launches regardless of B.
        with blocked.gemm_engine(wide):
            new_rows = v.transpose(-1, -2) @ sub_rows
        t_flat.index_copy_(0, flat_rows, new_rows.reshape(p * 2 * b, n))
        ext = None
        if wide:
            from kfac_amd import ops as _ops

            ext = _ops._load_ext()
        if ext is not None:
            # payload-only column moves (csrc/chol.hip): torch advanced
            # indexing would materialize p x n x 2b int64 index grids
            # (~160 MB/round at n=4608).
            tc = ext.gather_cols(t, mat_idx, idx_local)
            with blocked.gemm_engine(wide):
                tc = (tc @ v).contiguous()
            ext.scatter_cols(t, mat_idx, idx_local, tc)
            qc = ext.gather_cols(q, mat_idx, idx_local)
            with blocked.gemm_engine(wide):
                qc = (qc @ v).contiguous()
            ext.scatter_cols(q, mat_idx, idx_local, qc)
        else:
            rowg = (
                mat_idx.view(p, 1) * n + torch.arange(n, device=dev)
            ).view(p, n, 1)
            colg = idx_local.view(p, 1, 2 * b)
            tc = t_flat[rowg, colg]  # (p, n, 2b)
            with blocked.gemm_engine(wide):
                tc = tc @ v
            t_flat.index_put_((rowg, colg), tc)
            q_flat = q.reshape(bsz * n_true, n)
            rowq = (
                mat_idx.view(p, 1) * n_true
                + torch.arange(n_true, device=dev)
            ).view(p, n_true, 1)
            qc = q_flat[rowq, colg]
            with blocked.gemm_engine(wide):
                qc = qc @ v
            q_flat.index_put_((rowq, colg), qc)

    bnf = _block_off_norms(t, b)
    final_off = (bnf * bnf).sum(dim=(-2, -1))
    mask = ~failed & (final_off <= tol_sq * 1.0001)
    d = t.diagonal(dim1=-2, dim2=-1)[:, :n_true]
    return d.contiguous(), q[:, :, :n_true].contiguous(), mask
