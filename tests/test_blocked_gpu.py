"""GPU numerics for the blocked Cholesky engine (csrc/chol.hip +
ops/blocked.py) against plain fp32 torch references."""

from __future__ import annotations

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip('requires a GPU', allow_module_level=True)


def spd(b, n, seed, kappa=1e4):
    g = torch.Generator(device='cuda').manual_seed(seed)
    x = torch.randn(b, n, n, device='cuda', generator=g)
    a = (x @ x.transpose(-1, -2)) / n
    a.diagonal(dim1=-2, dim2=-1).add_(a.diagonal(dim1=-2, dim2=-1).mean() / kappa)
    return 0.5 * (a + a.transpose(-1, -2))


@pytest.mark.parametrize('n', [64, 128, 200, 384, 1024])
def test_potrf_batched_matches_torch(n: int) -> None:
    from kfac_amd.ops import blocked

    a = spd(4, n, seed=n)
    l = blocked.potrf_batched(a, tf32=False)
    ref = torch.linalg.cholesky(a)
    rel = (
        torch.linalg.norm(l - ref, dim=(-2, -1))
        / torch.linalg.norm(ref, dim=(-2, -1))
    ).max()
    assert float(rel) < 5e-5, float(rel)


@pytest.mark.parametrize('n', [96, 256, 1024])
def test_trinv_batched(n: int) -> None:
    from kfac_amd.ops import blocked

    a = spd(3, n, seed=n + 1)
    l, dinvs = blocked.potrf_batched(a, tf32=False, keep_dinv=True)
    t = blocked.trinv_batched(l, dinvs, tf32=False)
    eye = torch.eye(n, device='cuda').expand(3, n, n)
    err = torch.linalg.norm(t @ l - eye, dim=(-2, -1)).max() / n ** 0.5
    assert float(err) < 5e-5, float(err)


@pytest.mark.parametrize('n', [128, 777, 1024])
def test_spd_inverse_batched(n: int) -> None:
    from kfac_amd.ops import blocked

    a = spd(2, n, seed=n + 2)
    inv = blocked.spd_inverse_batched(a, damping=1e-3, tf32=True)
    ad = a.clone()
    ad.diagonal(dim1=-2, dim2=-1).add_(1e-3)
    eye = torch.eye(n, device='cuda').expand(2, n, n)
    err = torch.linalg.norm(inv @ ad - eye, dim=(-2, -1)).max() / n ** 0.5
    # xf32 engine: ~4.5e-6 per-GEMM relative error, kappa ~ 1e4 worst
    assert float(err) < 5e-2, float(err)
    ref = torch.cholesky_inverse(torch.linalg.cholesky(ad))
    rel = (
        torch.linalg.norm(inv - ref, dim=(-2, -1))
        / torch.linalg.norm(ref, dim=(-2, -1))
    ).max()
    assert float(rel) < 5e-3, float(rel)


def test_spd_solve_right() -> None:
    from kfac_amd.ops import blocked

    z = spd(3, 640, seed=9)
    x = torch.randn(3, 640, 640, device='cuda')
    y = blocked.spd_solve_right(x, z, tf32_chol=False, tf32_apply=False)
    ref = torch.cholesky_solve(
        x.transpose(-1, -2), torch.linalg.cholesky(z),
    ).transpose(-1, -2)
    rel = (
        torch.linalg.norm(y - ref, dim=(-2, -1))
        / torch.linalg.norm(ref, dim=(-2, -1))
    ).max()
    assert float(rel) < 1e-3, float(rel)


def test_eigh_qdwh_gpu_gates() -> None:
    """End-to-end QDWH on GPU with the blocked engine: K-FAC acceptance
    gates (reconstruction / orthogonality <= 1e-4)."""
    from kfac_amd import ops
    from kfac_amd.ops.qdwh import eigh_qdwh

    torch.manual_seed(5)
    n, b = 1024, 3
    g = torch.Generator(device='cuda').manual_seed(5)
    w = torch.randn(b, n, 2 * n, device='cuda', generator=g)
    a = (w @ w.transpose(-1, -2)) / (2 * n)
    a = a + torch.diag(torch.logspace(-5, 0.3, n, device='cuda')).unsqueeze(0)
    a = 0.5 * (a + a.transpose(-1, -2))
    hint, _ = ops.eigh_batched(a.clone())
    hint, _ = torch.sort(hint, dim=-1)
    wv, vv = eigh_qdwh(
        a, leaf_size=512, max_levels=1, leaf_fn=ops.eigh_batched,
        shift_hint=hint, generator=g,
    )
    a64 = a.to(torch.float64)
    v64 = vv.to(torch.float64)
    rec = (v64 * wv.to(torch.float64).unsqueeze(1)) @ v64.transpose(-1, -2)
    rec_err = (
        torch.linalg.norm(rec - a64, dim=(-2, -1))
        / torch.linalg.norm(a64, dim=(-2, -1))
    ).max()
    eye = torch.eye(n, dtype=torch.float64, device='cuda')
    orth_err = (
        torch.linalg.norm(
            v64.transpose(-1, -2) @ v64 - eye, dim=(-2, -1),
        )
        / n ** 0.5
    ).max()
    assert float(rec_err) < 1e-4, float(rec_err)
    assert float(orth_err) < 1e-4, float(orth_err)


def test_bucket_unpack_kernel() -> None:
    """Fused bucket scatter (K13) matches per-tensor copies."""
    from kfac_amd import ops

    ext = ops._load_ext()
    assert ext is not None
    torch.manual_seed(0)
    sizes = [(7, 7), (33,), (128, 129), (1,), (255,)]
    srcs = [torch.randn(*s, device='cuda') for s in sizes]
    flat = torch.cat([s.reshape(-1) for s in srcs])
    dsts = [torch.zeros_like(s) for s in srcs]
    ext.bucket_unpack(flat, dsts)
    torch.cuda.synchronize()
    for s, d in zip(srcs, dsts):
        torch.testing.assert_close(s, d)


def test_bucket_roundtrip_through_communicator() -> None:
    """AllreduceTensorBucket pack/communicate/unpack on GPU tensors
    (world 1: the flat buffer is the result) via the fused scatter."""
    from kfac_amd.distributed import AllreduceTensorBucket

    bucket = AllreduceTensorBucket(cap_bytes=1 << 22)
    tensors = [
        torch.full((64, 64), 3.0, device='cuda'),
        torch.full((17,), 5.0, device='cuda'),
    ]
    for t in tensors:
        bucket.append(t)
    bucket.communicate(group=None, scale=0.5)
    bucket.wait_and_unpack()
    torch.cuda.synchronize()
    torch.testing.assert_close(
        tensors[0], torch.full((64, 64), 1.5, device='cuda'),
    )
    torch.testing.assert_close(
        tensors[1], torch.full((17,), 2.5, device='cuda'),
    )
