"""Numerics of the warm-started adaptive block-Jacobi eigensolver
(CPU path; the LDS-subproblem route is covered by GPU tests)."""

from __future__ import annotations

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kfac_amd.ops.warm_eigh import warm_eigh_batched  # noqa: E402


def drifted_pair(n, seed, rot=0.15, val=0.2):
    """(F_prev, F_next) with a small basis rotation + eigenvalue drift."""
    g = torch.Generator().manual_seed(seed)
    q, _ = torch.linalg.qr(torch.randn(n, n, generator=g, dtype=torch.float64))
    vals = torch.logspace(-4, 0, n, dtype=torch.float64)
    f0 = (q * vals) @ q.T
    s = torch.randn(n, n, generator=g, dtype=torch.float64) * rot / n ** 0.5
    s = 0.5 * (s - s.T)
    qd = q @ torch.matrix_exp(s)
    vals2 = vals * (
        1.0 + val * (torch.rand(n, generator=g, dtype=torch.float64) - 0.5)
    )
    f1 = (qd * vals2) @ qd.T
    return f0.to(torch.float32), f1.to(torch.float32)


def gates(f_next, d, q, rec_tol=2e-4, orth_tol=5e-5):
    a64 = f_next.to(torch.float64)
    q64 = q.to(torch.float64)
    rec = (q64 * d.to(torch.float64)) @ q64.transpose(-1, -2)
    rec_err = float(torch.linalg.norm(rec - a64) / torch.linalg.norm(a64))
    n = f_next.size(-1)
    eye = torch.eye(n, dtype=torch.float64)
    orth = float(
        torch.linalg.norm(q64.transpose(-1, -2) @ q64 - eye) / n ** 0.5,
    )
    assert rec_err < rec_tol, rec_err
    assert orth < orth_tol, orth


@pytest.mark.parametrize('n', [96, 200, 257])
def test_warm_converges_small_drift(n: int) -> None:
    f0, f1 = drifted_pair(n, seed=n)
    _, q0 = torch.linalg.eigh(f0)
    d, q, ok = warm_eigh_batched(
        f1.unsqueeze(0), q0.unsqueeze(0), b=32,
    )
    assert bool(ok.all())
    gates(f1, d.squeeze(0), q.squeeze(0))


def test_warm_batch_mixed_drift() -> None:
    n = 160
    pairs = [drifted_pair(n, seed=7, rot=0.02), drifted_pair(n, seed=8, rot=0.2)]
    f1 = torch.stack([p[1] for p in pairs])
    q0 = torch.stack([torch.linalg.eigh(p[0])[1] for p in pairs])
    d, q, ok = warm_eigh_batched(f1, q0, b=32)
    assert bool(ok.all())
    for i in range(2):
        gates(f1[i], d[i], q[i])


def test_warm_zero_drift_is_noop_fast() -> None:
    n = 128
    f0, _ = drifted_pair(n, seed=3)
    _, q0 = torch.linalg.eigh(f0)
    d, q, ok = warm_eigh_batched(f0.unsqueeze(0), q0.unsqueeze(0), b=32)
    assert bool(ok.all())
    gates(f0, d.squeeze(0), q.squeeze(0))


def test_warm_bails_on_cold_start() -> None:
    n = 128
    f0, _ = drifted_pair(n, seed=11)
    _, f1 = drifted_pair(n, seed=12, rot=1.0)  # unrelated basis
    _, q0 = torch.linalg.eigh(f0)
    _, _, ok = warm_eigh_batched(f1.unsqueeze(0), q0.unsqueeze(0), b=32)
    assert not bool(ok.any())


def test_warm_identity_q_on_diagonal_matrix() -> None:
    # T already diagonal: zero rounds, exact result
    n = 96
    d_true = torch.linspace(0.1, 2.0, n)
    f = torch.diag(d_true)
    q0 = torch.eye(n)
    d, q, ok = warm_eigh_batched(f.unsqueeze(0), q0.unsqueeze(0), b=32)
    assert bool(ok.all())
    torch.testing.assert_close(
        d.squeeze(0).sort().values, d_true, atol=1e-5, rtol=1e-5,
    )
    gates(f, d.squeeze(0), q.squeeze(0))


def test_warm_multiple_pairs_same_matrix_cross_blocks() -> None:
    """Regression: two pairs of the SAME matrix in one round must apply
    both rotations to their cross blocks (row pass + column pass read
    order).  Construct coupling that forces >= 2 simultaneous pairs."""
    torch.manual_seed(0)
    n, b = 128, 32
    t = torch.diag(torch.linspace(0.5, 2.0, n).to(torch.float32))
    # strong coupling between blocks (0,1) and (2,3)
    t[0:32, 32:64] = 0.03 * torch.randn(32, 32)
    t[64:96, 96:128] = 0.03 * torch.randn(32, 32)
    t = 0.5 * (t + t.T)
    q0 = torch.eye(n)
    d, q, ok = warm_eigh_batched(t.unsqueeze(0), q0.unsqueeze(0), b=b)
    assert bool(ok.all())
    gates(t, d.squeeze(0), q.squeeze(0))
    w_ref = torch.linalg.eigvalsh(t.to(torch.float64))
    torch.testing.assert_close(
        d.squeeze(0).sort().values.to(torch.float64),
        w_ref,
        atol=1e-4,
        rtol=1e-4,
    )


def test_group_eigh_prev_override_plumbing() -> None:
    """The async worker passes snapshot clones via prev_override instead
    of letting _group_eigh read live layer attributes (race fix); the
    cold CPU path must accept both forms and agree with torch eigh."""
    from kfac_amd.base_preconditioner import BaseKFACPreconditioner

    class _FakeLayer:
        qa = None
        qg = None

    torch.manual_seed(0)
    m = torch.randn(3, 32, 32)
    stack = (m @ m.transpose(-1, -2) + 32 * torch.eye(32)).contiguous()
    layers = [_FakeLayer() for _ in range(3)]
    d_ref, q_ref = torch.linalg.eigh(stack)
    for prev in (None, [None, None, None], list(q_ref.unbind(0))):
        d, q = BaseKFACPreconditioner._group_eigh(
            stack, layers, 'a', prev_override=prev,
        )
        rec = q @ torch.diag_embed(d) @ q.transpose(-1, -2)
        assert torch.allclose(rec, stack, atol=1e-3), (
            (rec - stack).abs().max()
        )


def test_async_worker_snapshots_prev_eigenbases() -> None:
    """_launch_async_inverses must clone qa/qg into the work items (the
    worker may run while the main thread frees the live tensors)."""
    import inspect

    from kfac_amd.base_preconditioner import BaseKFACPreconditioner

    src = inspect.getsource(BaseKFACPreconditioner._launch_async_inverses)
    assert 'prev_override' in src
    assert '.detach().clone()' in src
    sig = inspect.signature(BaseKFACPreconditioner._group_eigh)
    assert 'prev_override' in sig.parameters


def test_async_work_grouping_alignment() -> None:
    """The async worker's grouping must keep each factor aligned with
    ITS layer's snapshot eigenbasis across mixed ownership."""
    from kfac_amd.base_preconditioner import BaseKFACPreconditioner

    la, lb, lc = object(), object(), object()
    a1, a2 = torch.zeros(8, 8), torch.zeros(8, 8)
    g1 = torch.zeros(4, 4)
    qa1, qg1 = torch.ones(8, 8), torch.ones(4, 4)
    work = [
        (la, a1, g1, qa1, qg1),   # owns both
        (lb, a2, None, None, None),  # owns A only, cold (no prev)
        (lc, None, None, None, None),  # owns neither
    ]
    ga = BaseKFACPreconditioner._group_async_work(work, 'a')
    gg = BaseKFACPreconditioner._group_async_work(work, 'g')
    assert set(ga.keys()) == {8} and set(gg.keys()) == {4}
    assert [(lyr, p is qa1) for lyr, _, p in ga[8]] == [
        (la, True),
        (lb, False),
    ]
    assert ga[8][1][2] is None
    assert gg[4] == [(la, g1, qg1)]


def test_warm_randomized_drift_property() -> None:
    """Randomized sweep (CPU torch fallback path): for drifted EMA-like
    factors, every matrix the solver reports converged must satisfy the
    reconstruction and orthogonality gates; bail/failure is only
    permitted, never silent wrongness."""
    from kfac_amd.ops.warm_eigh import warm_eigh_batched

    g = torch.Generator().manual_seed(123)
    for trial in range(6):
        n = int(torch.randint(80, 200, (1,), generator=g))
        b = int(torch.randint(1, 4, (1,), generator=g))
        base = torch.randn(b, n, n, generator=g)
        f0 = base @ base.transpose(-1, -2) / n + 2 * torch.eye(n)
        _, q0 = torch.linalg.eigh(f0)
        scale = [1e-4, 1e-3, 1e-2][trial % 3]
        pert = torch.randn(b, n, n, generator=g) * scale
        f1 = f0 + 0.5 * (pert + pert.transpose(-1, -2))
        d, q, ok = warm_eigh_batched(f1.contiguous(), q0.contiguous())
        for i in range(b):
            if not bool(ok[i]):
                continue  # dense fallback would handle it — allowed
            rec = q[i] @ torch.diag(d[i]) @ q[i].T
            err = (rec - f1[i]).norm() / f1[i].norm()
            assert err < 5e-4, (trial, i, float(err))
            orth = (q[i].T @ q[i] - torch.eye(n)).abs().max()
            assert orth < 1e-3, (trial, i, float(orth))
