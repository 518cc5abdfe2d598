"""Measure the warm-started block-Jacobi phase vs rocSOLVER syevd at
the real K-FAC group shapes, across drift levels.

Run: python scripts/bench_warm_eigh.py > gpurun_out/warm_bench.txt 2>&1
"""

from __future__ import annotations

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kfac_amd import ops  # noqa: E402
from kfac_amd.ops.warm_eigh import warm_eigh_batched  # noqa: E402


def timed(fn, warmup=1, iters=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def drifted_batch(bsz, n, seed, rot):
    """Basis rotation by a fixed small angle + eigenvalue jitter —
    modeled on MEASURED real drift (profiles/jacobi_warm.md: off(T)
    between phases is 1e-6..3e-2 of ||F||; mixing toward an independent
    Wishart, as an earlier version did, is orders harsher than
    reality because consecutive EMA updates share the basis)."""
    g = torch.Generator(device='cuda').manual_seed(seed)
    w = torch.randn(bsz, n, 2 * n, device='cuda', generator=g)
    f0 = (w @ w.transpose(-1, -2)) / (2 * n)
    f0 = f0 + torch.diag(torch.logspace(-4, 0, n, device='cuda')).unsqueeze(0)
    f0 = 0.5 * (f0 + f0.transpose(-1, -2))
    s = torch.randn(bsz, n, n, device='cuda', generator=g) * rot / n ** 0.5
    s = 0.5 * (s - s.transpose(-1, -2))
    qd = torch.matrix_exp(s)
    jit = 1.0 + 0.2 * (
        torch.rand(bsz, 1, n, device='cuda', generator=g) - 0.5
    )
    f1 = qd @ (f0 * jit) @ qd.transpose(-1, -2)
    return f0, 0.5 * (f1 + f1.transpose(-1, -2))


def quality(f, d, q):
    a64 = f.to(torch.float64)
    q64 = q.to(torch.float64)
    rec = (q64 * d.to(torch.float64).unsqueeze(1)) @ q64.transpose(-1, -2)
    rec_err = (
        torch.linalg.norm(rec - a64, dim=(-2, -1))
        / torch.linalg.norm(a64, dim=(-2, -1))
    ).max()
    return float(rec_err)


def main() -> None:
    torch.cuda.set_device(0)
    print(f'extension: {ops.extension_available()}')
    for b_, n in [(14, 1024), (6, 2304), (24, 3072), (3, 4608)]:
        f0, _ = drifted_batch(b_, n, seed=n, rot=0.1)
        t_syevd = timed(lambda: ops.eigh_batched(f0.clone()), 1, 2)
        _, q0 = ops.eigh_batched(f0)
        print(f'{b_}x{n}: syevd {t_syevd:7.1f} ms')
        for rot in (0.01, 0.05, 0.2):
            _, f1 = drifted_batch(b_, n, seed=n, rot=rot)

            def run():
                return warm_eigh_batched(f1, q0, b=32)

            d, q, ok = run()
            t_warm = timed(run, 1, 2)
            print(
                f'   warm rot={rot:0.2f}: {t_warm:7.1f} ms ok={bool(ok.all())} '
                f'rec={quality(f1, d, q):.1e} '
                f'speedup {t_syevd / t_warm:5.2f}x',
            )


if __name__ == '__main__':
    main()
