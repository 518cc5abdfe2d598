"""Per-group timing of the production inverse phase: warm vs syevd.

Trains ResNet-50 (fresh synthetic batches) through two inverse phases,
then times `_compute_local_inverses` group by group with the warm path
on and off.  Diagnoses where phase time goes.

Usage: python scripts/profile_inverse_phase.py [--model gptneox125m]
"""

from __future__ import annotations

import argparse
import os
import sys
import time
from collections import defaultdict

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument('--model', default='resnet50')
    ap.add_argument('--steps', type=int, default=110)
    args = ap.parse_args()

    from kfac_amd import KFACPreconditioner
    from kfac_amd.layers.eigen import KFACEigenLayer
    from kfac_amd.models import gptneox_125m, resnet50
    from kfac_amd.models.gptneox import KFAC_SKIP_LAYERS

    torch.manual_seed(0)
    is_lm = args.model == 'gptneox125m'
    model = (gptneox_125m() if is_lm else resnet50()).cuda()
    precon = KFACPreconditioner(
        model,
        factor_update_steps=10,
        inv_update_steps=100,
        lr=0.1,
        inv_update_async=False,
        skip_layers=KFAC_SKIP_LAYERS if is_lm else [],
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    crit = torch.nn.CrossEntropyLoss()
    gen = torch.Generator(device='cuda').manual_seed(3)
    vocab = 50304

    def batch():
        if is_lm:
            x = torch.randint(
                0, vocab, (8, 2048), device='cuda', generator=gen,
            )
            y = torch.randint(
                0, vocab, (8 * 2048,), device='cuda', generator=gen,
            )
            return x, y
        x = torch.randn(64, 3, 224, 224, device='cuda', generator=gen)
        y = torch.randint(0, 1000, (64,), device='cuda', generator=gen)
        return x, y

    t0 = time.time()
    for _ in range(args.steps):
        opt.zero_grad(set_to_none=True)
        x, y = batch()
        with torch.autocast('cuda', dtype=torch.bfloat16):
            out = model(x)
            loss = crit(out.view(-1, out.size(-1)) if is_lm else out, y)
        loss.backward()
        precon.step()
        opt.step()
    torch.cuda.synchronize()
    print(f'{args.steps} steps in {time.time() - t0:.1f}s')

    # collect the eigen groups exactly as _compute_local_inverses does
    layers = [
        layer
        for _, (name, layer) in precon._layers.items()
        if isinstance(layer, KFACEigenLayer)
    ]
    for which in ('a', 'g'):
        groups = defaultdict(list)
        for layer in layers:
            f = layer.a_factor if which == 'a' else layer.g_factor
            groups[f.shape[0]].append(layer)
        print(f'\n== {which.upper()} groups ==')
        for n, group in sorted(groups.items()):
            stack = torch.stack(
                [
                    (l.a_factor if which == 'a' else l.g_factor).to(
                        torch.float32,
                    )
                    for l in group
                ],
            )

            def t(fn, iters=3):
                fn()
                torch.cuda.synchronize()
                tt = time.perf_counter()
                for _ in range(iters):
                    fn()
                torch.cuda.synchronize()
                return (time.perf_counter() - tt) / iters * 1000.0

            from kfac_amd import ops as _ops

            t_syevd = t(lambda: _ops.eigh_batched(stack.clone()))
            res = {}

            def run_warm():
                from kfac_amd.base_preconditioner import (
                    BaseKFACPreconditioner,
                )

                for l in group:
                    setattr(l, f'_warm_phases_{which}', 0)
                res['out'] = BaseKFACPreconditioner._group_eigh(
                    stack, group, which,
                )

            t_grp = t(run_warm)
            warm_used = all(
                getattr(l, f'_warm_phases_{which}', 0) > 0 for l in group
            )
            # quality
            d, q = res['out']
            a64 = stack.to(torch.float64)
            q64 = q.to(torch.float64)
            rec = (q64 * d.to(torch.float64).unsqueeze(1)) @ q64.transpose(
                -1, -2,
            )
            rec_err = (
                torch.linalg.norm(rec - a64, dim=(-2, -1))
                / torch.linalg.norm(a64, dim=(-2, -1))
            ).max()
            print(
                f'  {len(group):3d}x{n:5d}: syevd {t_syevd:8.1f} ms | '
                f'group {t_grp:8.1f} ms warm={warm_used} '
                f'rec={float(rec_err):.1e} '
                f'speedup {t_syevd / t_grp:5.2f}x',
            )


if __name__ == '__main__':
    main()
