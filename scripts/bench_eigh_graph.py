"""Micro: ops.eigh_batched (cold path) vs torch.linalg.eigh on the
ResNet-50 factor groups.  (Round 1 benchmarked the hipGraph-replayed
syevd here; that machinery was removed as unsound in round 2 — this now
measures the direct batched call the warm solver falls back to.)"""
from __future__ import annotations

import sys
import time
from collections import defaultdict

import torch

sys.path.insert(0, '.')
from kfac_amd import ops  # noqa: E402
from scripts.bench_eigh import factor_sizes, make_factors  # noqa: E402


def run(fn, groups, iters=3):
    torch.cuda.synchronize()
    times = []
    for _ in range(iters):
        t0 = time.perf_counter()
        for stack in groups:
            fn(stack)
        torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    return times


def main() -> None:
    sizes = factor_sizes()
    factors = make_factors(sizes)
    by_n = defaultdict(list)
    for f in factors:
        by_n[f.shape[0]].append(f)
    groups = [torch.stack(v) for v in by_n.values()]
    print('groups:', sorted((g.shape[0], g.shape[1]) for g in groups))
    t_t = run(torch.linalg.eigh, groups)
    t_g = run(ops.eigh_batched, groups)
    print('torch.linalg.eigh phases :', ' '.join(f'{t:.3f}' for t in t_t))
    print('ops.eigh_batched phases  :', ' '.join(f'{t:.3f}' for t in t_g))


if __name__ == '__main__':
    main()
