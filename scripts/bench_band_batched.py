"""Batched vs per-matrix stage-1 band reduction at the LM group shape."""
from __future__ import annotations

import sys
import time

import torch

sys.path.insert(0, '.')
from kfac_amd.ops.two_stage_eigh import (  # noqa: E402
    reduce_to_band,
    reduce_to_band_batched,
)


def t(fn, iters=2):
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
    for bsz, n, band in [(24, 3072, 64), (24, 3072, 128), (3, 4608, 128)]:
        g = torch.Generator().manual_seed(0)
        r = torch.randn(bsz, n, n, generator=g).cuda()
        stack = r @ r.transpose(1, 2) / n + 0.1 * torch.eye(n, device='cuda')
        tb = t(lambda: reduce_to_band_batched(stack, band))
        tl = t(lambda: [reduce_to_band(stack[i], band) for i in range(bsz)], iters=1)
        print(
            f'B={bsz} n={n} band={band}: batched {tb*1e3:8.1f} ms   '
            f'looped {tl*1e3:8.1f} ms',
            flush=True,
        )


if __name__ == '__main__':
    main()
