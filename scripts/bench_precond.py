"""Micro-benchmark: grouped precond + klclip vs torch-eager equivalents
on ResNet-50's real (m, n) layer shapes."""

from __future__ import annotations

import sys
import time

import torch

sys.path.insert(0, '.')

from kfac_amd.models import resnet50  # noqa: E402


def lm_shapes() -> list[tuple[int, int]]:
    return [(3072, 769), (768, 3073)] * 12


def shapes() -> list[tuple[int, int]]:
    out = []
    for m in resnet50().modules():
        if isinstance(m, torch.nn.Conv2d):
            out.append(
                (m.out_channels, m.in_channels * m.kernel_size[0] * m.kernel_size[1]),
            )
        elif isinstance(m, torch.nn.Linear):
            out.append((m.weight.size(0), m.weight.size(1) + 1))
    return out


def timeit(fn, reps=10) -> float:
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main() -> None:
    from kfac_amd import _kfaccore

    import sys as _sys

    ss = lm_shapes() if '--lm' in _sys.argv else shapes()
    grads, qas, qgs, dgdas = [], [], [], []
    for m, n in ss:
        grads.append(torch.randn(m, n, device='cuda'))
        qas.append(torch.randn(n, n, device='cuda'))
        qgs.append(torch.randn(m, m, device='cuda'))
        dgdas.append(torch.rand(m, n, device='cuda') + 0.5)

    def grouped():
        _kfaccore.precond_eigen_grouped(grads, qas, qgs, dgdas)

    def eager():
        for g, qa, qg, dd in zip(grads, qas, qgs, dgdas):
            v1 = qg.t() @ g @ qa
            (qg @ (v1 * dd) @ qa.t())

    def perlayer_ext():
        for g, qa, qg, dd in zip(grads, qas, qgs, dgdas):
            _kfaccore.precond_eigen_fused(g, qa, qg, dd)

    print(f'{len(ss)} layers')
    print(f'grouped ext : {timeit(grouped)*1000:.2f} ms')
    print(f'eager torch : {timeit(eager)*1000:.2f} ms')
    print(f'perlayer ext: {timeit(perlayer_ext)*1000:.2f} ms')

    # klclip
    accum = torch.zeros((), device='cuda')

    def kl_ext():
        for g in grads:
            _kfaccore.kl_clip_accum(accum, g, g)

    def kl_eager():
        s = 0.0
        for g in grads:
            s += float((g * g).sum())

    def kl_eager_device():
        acc = torch.zeros((), device='cuda')
        for g in grads:
            acc += (g * g).sum()

    print(f'klclip ext        : {timeit(kl_ext)*1000:.2f} ms')
    print(f'klclip eager .item: {timeit(kl_eager)*1000:.2f} ms')
    print(f'klclip eager dev  : {timeit(kl_eager_device)*1000:.2f} ms')

    # update_grad style ops
    scale = torch.tensor(0.5, device='cuda')

    def update():
        for g in grads:
            gg = g * scale
            gg[:, :-1].reshape(-1).contiguous()

    print(f'update-ish        : {timeit(update)*1000:.2f} ms')


if __name__ == '__main__':
    main()
