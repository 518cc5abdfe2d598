"""Measure QDWH divide-and-conquer vs rocSOLVER syevd on MI355X.

Covers the real K-FAC factor groups (ResNet-50: 3x4608, 6x2304,
14x1024; GPT-NeoX-125M MLP: 24x3072) plus primitive timings
(batched potrf / trsm / bmm) to target kernel work.

Run on the GPU box:
  python scripts/bench_qdwh.py > gpurun_out/qdwh_bench.txt 2>&1
"""

from __future__ import annotations

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kfac_amd import ops  # noqa: E402
from kfac_amd.ops.qdwh import eigh_qdwh  # noqa: E402


def timed(fn, warmup=1, iters=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def make_batch(b, n, seed, device, with_hint=True):
    """K-FAC-like PSD batch: Wishart bulk + wide-dynamic-range diagonal.

    Built on-GPU (CPU fp64 QR at n=4608 takes minutes). The hint is the
    exact spectrum from one untimed solve — matching production, where
    the hint is the previous phase's eigenvalues.
    """
    g = torch.Generator(device=device).manual_seed(seed)
    w = torch.randn(b, n, 2 * n, device=device, generator=g)
    a = (w @ w.transpose(-1, -2)) / (2 * n)
    a = a + torch.diag(
        torch.logspace(-5, 0.3, n, device=device),
    ).unsqueeze(0)
    a = 0.5 * (a + a.transpose(-1, -2))
    hint = None
    if with_hint:
        hint, _ = ops.eigh_batched(a.clone())
        hint, _ = torch.sort(hint, dim=-1)
    return a, hint


def quality(a, w, v):
    a64 = a.to(torch.float64)
    w64 = w.to(torch.float64)
    v64 = v.to(torch.float64)
    rec = (v64 * w64.unsqueeze(1)) @ v64.transpose(-1, -2)
    rec_err = (
        torch.linalg.norm(rec - a64, dim=(-2, -1))
        / torch.linalg.norm(a64, dim=(-2, -1))
    ).max()
    eye = torch.eye(a.size(-1), dtype=torch.float64, device=a.device)
    orth = v64.transpose(-1, -2) @ v64 - eye
    orth_err = (
        torch.linalg.norm(orth, dim=(-2, -1)) / a.size(-1) ** 0.5
    ).max()
    return float(rec_err), float(orth_err)


def main() -> None:
    device = torch.device('cuda', 0)
    torch.cuda.set_device(device)
    print(f'extension available: {ops.extension_available()}')

    from kfac_amd.ops import blocked

    print('\n== primitive timings (fp32, batched) ==')
    for b, n in [(3, 4608), (24, 3072), (6, 2304), (14, 1024)]:
        x = torch.randn(b, n, n, device=device)
        s = x @ x.transpose(-1, -2) + n * torch.eye(n, device=device)
        t_bmm = timed(lambda: x @ x)
        with blocked.gemm_engine(True):
            t_bmm_w = timed(lambda: x @ x)
        t_chol = timed(lambda: torch.linalg.cholesky(s))
        l = torch.linalg.cholesky(s)
        t_solve = timed(lambda: torch.cholesky_solve(x, l))
        t_bchol = timed(lambda: blocked.potrf_batched(s, tf32=True))
        lb, dinvs = blocked.potrf_batched(s, tf32=True, keep_dinv=True)
        t_btri = timed(
            lambda: blocked.trinv_batched(lb, dinvs, tf32=True),
        )
        t_bsolve = timed(
            lambda: blocked.spd_solve_right(x, s),
        )
        tf = 2 * b * n ** 3 / (t_bmm / 1e3) / 1e12
        tfw = 2 * b * n ** 3 / (t_bmm_w / 1e3) / 1e12
        print(
            f'  {b}x{n}: bmm {t_bmm:7.2f} ms ({tf:5.1f} TF) '
            f'xf32 {t_bmm_w:7.2f} ms ({tfw:5.1f} TF) | '
            f'potrf {t_chol:7.2f} -> {t_bchol:7.2f} ms | '
            f'trinv {t_btri:7.2f} ms | '
            f'potrs {t_solve:7.2f} -> solve {t_bsolve:7.2f} ms',
        )

    print('\n== leaf solver calibration (syevd batched) ==')
    for b, n in [
        (6, 256), (12, 256), (6, 512), (12, 512), (6, 768), (12, 768),
        (6, 1152), (6, 1536), (3, 2304),
    ]:
        a, _ = make_batch(b, n, seed=n + b, device=device, with_hint=False)
        t = timed(lambda: ops.eigh_batched(a.clone()), warmup=2, iters=3)
        print(f'  syevd {b:3d}x{n:5d}: {t:8.1f} ms')

    print('\n== group benchmarks: syevd vs qdwh ==')
    gen = torch.Generator(device=device).manual_seed(1234)
    for b, n in [(14, 1024), (6, 2304), (24, 3072), (3, 4608)]:
        a, spectra = make_batch(b, n, seed=n, device=device)
        t_syevd = timed(lambda: ops.eigh_batched(a.clone()), warmup=1, iters=2)
        w_s, v_s = ops.eigh_batched(a.clone())
        rec_s, orth_s = quality(a, w_s, v_s)
        print(
            f'  {b}x{n}: syevd {t_syevd:8.1f} ms '
            f'(rec {rec_s:.1e}, orth {orth_s:.1e})',
        )
        for leaf, levels in [(512, 3), (768, 2), (768, 3), (1024, 2)]:
            if n <= leaf:
                continue

            def run():
                return eigh_qdwh(
                    a,
                    leaf_size=leaf,
                    max_levels=levels,
                    leaf_fn=ops.eigh_batched,
                    shift_hint=spectra,
                    generator=gen,
                )

            try:
                t_q = timed(run, warmup=1, iters=2)
                w_q, v_q = run()
                rec_q, orth_q = quality(a, w_q, v_q)
                print(
                    f'    qdwh leaf={leaf} lvl={levels}: {t_q:8.1f} ms '
                    f'(rec {rec_q:.1e}, orth {orth_q:.1e}) '
                    f'speedup {t_syevd / t_q:5.2f}x',
                )
            except Exception as e:  # noqa: BLE001
                print(f'    qdwh leaf={leaf} lvl={levels}: FAILED {e}')

    print('\n== whole-phase simulation (ResNet-50 groups together) ==')
    groups = [(14, 1024), (6, 2304), (3, 4608)]
    data = [make_batch(b, n, seed=n + 1, device=device) for b, n in groups]

    def phase_syevd():
        for (a, _h) in data:
            ops.eigh_batched(a.clone())

    def phase_qdwh():
        for (a, h) in data:
            if a.size(-1) > 768:
                eigh_qdwh(
                    a, leaf_size=768, max_levels=3,
                    leaf_fn=ops.eigh_batched, shift_hint=h, generator=gen,
                )
            else:
                ops.eigh_batched(a.clone())

    print(f'  syevd phase: {timed(phase_syevd, 1, 2):8.1f} ms')
    print(f'  qdwh  phase: {timed(phase_qdwh, 1, 2):8.1f} ms')


if __name__ == '__main__':
    main()
