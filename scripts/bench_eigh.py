"""Micro-benchmark: eigendecomposition strategies for ResNet-50 factors.

Strategies:
  loop     - torch.linalg.eigh per factor (reference behavior)
  batched  - group same-size factors, one batched eigh per group
  streams  - per-factor eigh fanned over N side streams
  magma    - loop with preferred_linalg_library('magma') if available
"""

from __future__ import annotations

import sys
import time
from collections import defaultdict

import torch

sys.path.insert(0, '.')

from kfac_amd.models import resnet50  # noqa: E402


def factor_sizes() -> list[int]:
    sizes = []
    for m in resnet50().modules():
        if isinstance(m, torch.nn.Conv2d):
            sizes.append(m.in_channels * m.kernel_size[0] * m.kernel_size[1])
            sizes.append(m.out_channels)
        elif isinstance(m, torch.nn.Linear):
            sizes.append(m.weight.size(1) + 1)
            sizes.append(m.weight.size(0))
    return sizes


def make_factors(sizes: list[int]) -> list[torch.Tensor]:
    out = []
    for n in sizes:
        a = torch.randn(n, n, device='cuda')
        f = 0.95 * torch.eye(n, device='cuda') + 0.05 * (a @ a.t()) / n
        out.append(f)
    return out


def t_loop(factors) -> float:
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    res = [torch.linalg.eigh(f) for f in factors]
    torch.cuda.synchronize()
    return time.perf_counter() - t0


def t_batched(factors) -> float:
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    groups = defaultdict(list)
    for i, f in enumerate(factors):
        groups[f.shape[0]].append(i)
    res: dict[int, tuple] = {}
    for n, idxs in groups.items():
        stack = torch.stack([factors[i] for i in idxs])
        d, q = torch.linalg.eigh(stack)
        for j, i in enumerate(idxs):
            res[i] = (d[j], q[j])
    torch.cuda.synchronize()
    return time.perf_counter() - t0


def t_streams(factors, n_streams=4) -> float:
    streams = [torch.cuda.Stream() for _ in range(n_streams)]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    res = []
    for i, f in enumerate(factors):
        with torch.cuda.stream(streams[i % n_streams]):
            res.append(torch.linalg.eigh(f))
    torch.cuda.synchronize()
    return time.perf_counter() - t0


def main() -> None:
    sizes = factor_sizes()
    print(f'{len(sizes)} factors; largest: {sorted(sizes)[-6:]}')
    factors = make_factors(sizes)
    # warmup
    torch.linalg.eigh(factors[0])
    for name, fn in [('loop', t_loop), ('batched', t_batched), ('streams4', t_streams)]:
        times = [fn(factors) for _ in range(2)]
        print(f'{name:10s}: {min(times):.3f}s')
    try:
        torch.backends.cuda.preferred_linalg_library('magma')
        torch.linalg.eigh(factors[0])
        times = [t_loop(factors) for _ in range(2)]
        print(f'{"magma-loop":10s}: {min(times):.3f}s')
        torch.backends.cuda.preferred_linalg_library('default')
    except Exception as e:
        print('magma unavailable:', e)

    # single large-factor timing detail
    for n in (4608, 2304, 1024, 512, 256):
        f = [x for x in factors if x.shape[0] == n]
        if not f:
            continue
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        torch.linalg.eigh(f[0])
        torch.cuda.synchronize()
        print(f'single n={n}: {time.perf_counter() - t0:.4f}s')


if __name__ == '__main__' and '--jacobi' not in sys.argv:
    main()


def bench_jacobi() -> None:
    """Compare batched syevd vs syevj on the real group distribution."""
    import time as _t

    from kfac_amd import _kfaccore

    sizes = factor_sizes()
    from collections import Counter

    dist = Counter(sizes)
    print('jacobi-vs-syevd per group:')
    total_d = total_j = total_jl = 0.0
    for n, count in sorted(dist.items(), reverse=True):
        a = torch.randn(count, n, n, device='cuda')
        stack = 0.95 * torch.eye(n, device='cuda').expand(count, n, n).clone()
        stack += 0.05 * (a @ a.transpose(1, 2)) / n

        def t(fn):
            fn()
            torch.cuda.synchronize()
            t0 = _t.perf_counter()
            fn()
            torch.cuda.synchronize()
            return _t.perf_counter() - t0

        td = t(lambda: torch.linalg.eigh(stack))
        tj = t(lambda: _kfaccore.eigh_jacobi(stack, 0.0, 100))
        tjl = t(lambda: _kfaccore.eigh_jacobi(stack, 1e-5, 20))
        total_d += td
        total_j += tj
        total_jl += tjl
        # accuracy of the loose variant
        w, q = _kfaccore.eigh_jacobi(stack, 1e-5, 20)
        recon = q @ torch.diag_embed(w) @ q.transpose(1, 2)
        err = (recon - stack).abs().max().item()
        print(
            f'  n={n:5d} x{count:2d}: syevd {td*1000:8.2f} ms | '
            f'syevj {tj*1000:8.2f} | syevj-loose {tjl*1000:8.2f} '
            f'(recon err {err:.2e})',
        )
    print(
        f'TOTAL: syevd {total_d*1000:.1f} ms, syevj {total_j*1000:.1f} ms, '
        f'syevj-loose {total_jl*1000:.1f} ms',
    )


if __name__ == '__main__' and '--jacobi' in sys.argv:
    bench_jacobi()
