"""Measure stage-1 (full -> band) GPU cost vs full rocSOLVER syevd.

The two-stage plan's premise is that the GEMM-rich band reduction is
cheap next to syevd's latrd chain; this stamps the number."""
from __future__ import annotations

import sys
import time

import torch

sys.path.insert(0, '.')
from kfac_amd.ops.two_stage_eigh import reduce_to_band  # noqa: E402


def t(fn, iters=3):
    fn()
    torch.cuda.synchronize()
    best = 1e9
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return best


def main() -> None:
    for n, band in [(4608, 64), (3072, 64), (2304, 64), (4608, 128)]:
        g = torch.Generator().manual_seed(0)
        r = torch.randn(n, n, generator=g).cuda()
        a = r @ r.t() / n + 0.1 * torch.eye(n, device='cuda')
        tb = t(lambda: reduce_to_band(a, band))
        te = t(lambda: torch.linalg.eigh(a), iters=1)
        b, _ = reduce_to_band(a, band)
        err = (
            torch.linalg.eigvalsh(b.double()) - torch.linalg.eigvalsh(a.double())
        ).abs().max()
        print(
            f'n={n} band={band}: stage1 {tb*1e3:8.1f} ms   '
            f'full eigh {te*1e3:8.1f} ms   |dλ|max {err:.2e}',
            flush=True,
        )


if __name__ == '__main__':
    main()
