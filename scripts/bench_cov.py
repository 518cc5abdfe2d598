"""Per-layer-shape covariance kernel micro-benchmark (ResNet-50 bs=64).

Times ext.cov_conv_a / cov_conv_g / cov_linear per distinct layer shape
and reports effective TFLOP/s, plus the torch-eager equivalent.
"""

from __future__ import annotations

import sys
import time

import torch

sys.path.insert(0, '.')

from kfac_amd.models import resnet50  # noqa: E402


def conv_shapes(bs: int = 64):
    """(C,H,W,kh,kw,sh,sw,ph,pw, out_ch) per distinct conv of resnet50."""
    shapes = {}
    model = resnet50()
    x = torch.randn(1, 3, 224, 224)
    hooks = []

    def mk_hook(name, m):
        def hook(mod, inp, out):
            shapes[name] = (
                tuple(inp[0].shape[1:]),
                mod.kernel_size,
                mod.stride,
                mod.padding,
                mod.out_channels,
            )
        return hook

    for name, m in model.named_modules():
        if isinstance(m, torch.nn.Conv2d):
            hooks.append(m.register_forward_hook(mk_hook(name, m)))
    model(x)
    for h in hooks:
        h.remove()
    # dedupe by shape signature, count occurrences
    uniq = {}
    for name, sig in shapes.items():
        uniq.setdefault(sig, []).append(name)
    return uniq


def timeit(fn, reps=5) -> float:
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main() -> None:
    from kfac_amd import _kfaccore

    bs = 64
    total_ext = 0.0
    total_eager = 0.0
    rows = []
    for (in_shape, k, s, p, out_ch), names in sorted(
        conv_shapes().items(), key=lambda kv: -len(kv[1]),
    ):
        C, H, W = in_shape
        x = torch.randn(bs, C, H, W, device='cuda', dtype=torch.bfloat16)
        n = C * k[0] * k[1]
        out = torch.zeros(n, n, device='cuda')
        OH = (H + 2 * p[0] - k[0]) // s[0] + 1
        OW = (W + 2 * p[1] - k[1]) // s[1] + 1
        M = bs * OH * OW
        flops = M * n * n  # MACs in the (symmetric-half-free) product

        def ext_fn():
            _kfaccore.cov_conv_a(
                x, out, k[0], k[1], s[0], s[1], p[0], p[1], False, 0.0, 1.0,
            )

        t = timeit(ext_fn)
        count = len(names)
        total_ext += t * count

        # eager equivalent: unfold + matmul
        def eager_fn():
            pat = torch.nn.functional.unfold(
                x.float(), k, padding=p, stride=s,
            ).transpose(1, 2).reshape(-1, n)
            c = pat.t() @ pat

        te = timeit(eager_fn, reps=3)
        total_eager += te * count
        rows.append(
            f'C{C:4d} k{k[0]} s{s[0]} n={n:5d} M={M:7d} x{count:2d}: '
            f'ext {t*1000:7.2f} ms ({2*flops/t/1e12:6.1f} TF)  '
            f'eager {te*1000:7.2f} ms',
        )
        # G factor for this conv
        g = torch.randn(bs, out_ch, OH, OW, device='cuda', dtype=torch.bfloat16)
        outg = torch.zeros(out_ch, out_ch, device='cuda')

        def g_fn():
            _kfaccore.cov_conv_g(g, outg, 0.0, 1.0)

        tg = timeit(g_fn)
        total_ext += tg * count
        rows.append(f'       G out={out_ch:5d}        x{count:2d}: ext {tg*1000:7.2f} ms')

    print('\n'.join(rows))
    print(f'TOTAL A+G ext ~= {total_ext*1000:.1f} ms  (eager A-only ~= {total_eager*1000:.1f} ms)')


if __name__ == '__main__':
    main()
