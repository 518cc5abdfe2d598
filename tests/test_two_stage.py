"""Two-stage eigensolver groundwork (kfac_amd/ops/two_stage_eigh.py):
stage-1 band reduction math + full-pipeline validation on CPU.

These gates exist BEFORE the round-2 kernels so the algorithm (band
layout, Q1 accumulation order, back-transform) is already proven; the
GPU kernels will be validated against the same invariants.
"""

from __future__ import annotations

import sys

import pytest
import torch

sys.path.insert(0, '.')

from kfac_amd.ops.two_stage_eigh import (  # noqa: E402
    apply_q1,
    eigh_two_stage_cpu,
    reduce_to_band,
)


def _spd(n: int, seed: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(n, n, generator=g, dtype=torch.float64)
    return r @ r.t() / n + 0.1 * torch.eye(n, dtype=torch.float64)


@pytest.mark.parametrize('n,band', [(65, 16), (128, 32), (200, 32), (257, 64), (96, 96)])
def test_band_reduction_structure_and_similarity(n: int, band: int) -> None:
    a = _spd(n, n)
    b, panels = reduce_to_band(a, band)
    # banded: zero beyond the bandwidth
    for i in range(n):
        for j in range(n):
            if abs(i - j) > band:
                assert abs(b[i, j].item()) < 1e-10
    # exact symmetry
    torch.testing.assert_close(b, b.t())
    # similarity: Q1 B Q1^T == A (reconstruct via apply_q1 on B's columns)
    q1 = apply_q1(panels, torch.eye(n, dtype=torch.float64))
    torch.testing.assert_close(q1 @ q1.t(), torch.eye(n, dtype=torch.float64),
                               rtol=1e-10, atol=1e-10)
    torch.testing.assert_close(q1 @ b @ q1.t(), a, rtol=1e-9, atol=1e-9)
    # spectrum preserved
    torch.testing.assert_close(
        torch.linalg.eigvalsh(b), torch.linalg.eigvalsh(a),
        rtol=1e-9, atol=1e-9,
    )


@pytest.mark.parametrize('n,band', [(150, 32), (257, 64)])
def test_two_stage_pipeline_matches_eigh(n: int, band: int) -> None:
    a = _spd(n, 7 * n)
    w, v = eigh_two_stage_cpu(a, band)
    w_ref = torch.linalg.eigvalsh(a)
    torch.testing.assert_close(w, w_ref, rtol=1e-8, atol=1e-8)
    # eigenpairs: A v = v diag(w), orthonormal v
    torch.testing.assert_close(a @ v, v @ torch.diag(w), rtol=1e-7, atol=1e-7)
    eye = torch.eye(n, dtype=torch.float64)
    torch.testing.assert_close(v.t() @ v, eye, rtol=1e-8, atol=1e-8)


def test_band_reduction_small_edge() -> None:
    # band >= n-1 means nothing to do (already "banded")
    a = _spd(16, 3)
    b, panels = reduce_to_band(a, 16)
    assert panels == []
    torch.testing.assert_close(a, b)


@pytest.mark.parametrize('n,band', [(40, 8), (65, 16), (100, 32), (50, 49)])
def test_bulge_chase_tridiagonalizes(n: int, band: int) -> None:
    from kfac_amd.ops.two_stage_eigh import band_to_tridiag

    a = _spd(n, 11 * n)
    b, _ = reduce_to_band(a, band)
    d, e, q2 = band_to_tridiag(b, band)
    tri = torch.diag(d) + torch.diag(e, -1) + torch.diag(e, 1)
    eye = torch.eye(n, dtype=torch.float64)
    torch.testing.assert_close(q2 @ q2.t(), eye, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(q2 @ tri @ q2.t(), b, rtol=1e-11, atol=1e-11)
    torch.testing.assert_close(
        torch.linalg.eigvalsh(tri), torch.linalg.eigvalsh(a),
        rtol=1e-11, atol=1e-11,
    )


@pytest.mark.parametrize('n,band', [(64, 16), (100, 32)])
def test_self_pipeline_matches_eigh(n: int, band: int) -> None:
    """Fully self-implemented 3-stage pipeline (no LAPACK banded solver)
    — the algorithm the round-2 kernels implement, end to end."""
    from kfac_amd.ops.two_stage_eigh import eigh_two_stage_self

    a = _spd(n, 13 * n)
    w, v = eigh_two_stage_self(a, band)
    torch.testing.assert_close(
        w, torch.linalg.eigvalsh(a), rtol=1e-10, atol=1e-10,
    )
    torch.testing.assert_close(
        a @ v, v @ torch.diag(w), rtol=1e-9, atol=1e-9,
    )


@pytest.mark.parametrize('bsz,n,band', [(3, 100, 32), (5, 65, 16), (2, 150, 64)])
def test_batched_band_reduction_matches_loop(bsz: int, n: int, band: int) -> None:
    from kfac_amd.ops.two_stage_eigh import (
        apply_q1_batched,
        reduce_to_band_batched,
    )

    stack = torch.stack([_spd(n, 100 * i + n) for i in range(bsz)])
    bb, panels = reduce_to_band_batched(stack, band)
    eye = torch.eye(n, dtype=torch.float64).expand(bsz, n, n).contiguous()
    q1 = apply_q1_batched(panels, eye)
    torch.testing.assert_close(
        q1 @ bb @ q1.transpose(1, 2), stack, rtol=1e-9, atol=1e-9,
    )
    for i in range(bsz):
        b1, _ = reduce_to_band(stack[i], band)
        torch.testing.assert_close(bb[i], b1, rtol=1e-9, atol=1e-9)
        torch.testing.assert_close(
            torch.linalg.eigvalsh(bb[i]), torch.linalg.eigvalsh(stack[i]),
            rtol=1e-9, atol=1e-9,
        )
