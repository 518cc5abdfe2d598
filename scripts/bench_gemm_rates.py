"""Measure candidate GEMM engines for the QDWH polar iteration.

The polar iteration is ~2.3 n^3 FMA of GEMM-class work per step; the
round-2 QDWH measurement showed torch f32 bmm at ~125-150 TF and the
rocSOLVER potrf/trsm path at 2-4 TF.  This script measures, per engine,
big-shape throughput AND accuracy vs a float64 reference:

  - torch mm fp32 (rocBLAS)
  - torch mm fp32 with allow_tf32 (hipBLASLt xf32, if effective)
  - torch mm bf16 (hipBLASLt; accuracy floor reference)
  - _kfaccore.gemm split-bf16x3 (hand-written MFMA)
  - _kfaccore.gemm exact f32 MFMA

Run: python scripts/bench_gemm_rates.py
"""

from __future__ import annotations

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kfac_amd import _kfaccore as core  # noqa: E402


def timed(fn, warmup=2, iters=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main() -> None:
    device = torch.device('cuda', 0)
    for n in [1024, 2304, 3072, 4608]:
        torch.manual_seed(n)
        a = torch.randn(n, n, device=device) / n ** 0.5
        b = torch.randn(n, n, device=device) / n ** 0.5
        ref = (a.to(torch.float64) @ b.to(torch.float64))
        ref_n = torch.linalg.norm(ref)
        flops = 2.0 * n ** 3

        def report(name, fn):
            try:
                out = fn()
                err = float(
                    torch.linalg.norm(out.to(torch.float64) - ref) / ref_n,
                )
                t = timed(fn)
                print(
                    f'  n={n} {name:<14} {t * 1e3:8.2f} ms '
                    f'{flops / t / 1e12:7.1f} TF  relerr {err:.2e}',
                )
            except Exception as e:  # noqa: BLE001
                print(f'  n={n} {name:<14} FAILED: {e}')

        torch.backends.cuda.matmul.allow_tf32 = False
        report('torch f32', lambda: a @ b)
        torch.backends.cuda.matmul.allow_tf32 = True
        report('torch tf32', lambda: a @ b)
        torch.backends.cuda.matmul.allow_tf32 = False
        a16, b16 = a.to(torch.bfloat16), b.to(torch.bfloat16)
        report('torch bf16', lambda: (a16 @ b16).to(torch.float32))
        report('core split', lambda: core.gemm(a, b, False, False, True))
        report('core f32', lambda: core.gemm(a, b, False, False, False))
        report(
            'core split tN', lambda: core.gemm(a, b, True, False, True),
        )
        report(
            'core split nT', lambda: core.gemm(a, b, False, True, True),
        )
        print()


if __name__ == '__main__':
    main()
