"""Diagnose (a) syevd garbage on small G-factor groups and (b) the gap
between the forced-phase wall time and the sum of per-group times."""

from __future__ import annotations

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    from kfac_amd import KFACPreconditioner, ops
    from kfac_amd.models import resnet50

    torch.manual_seed(1234)
    model = resnet50().cuda()
    precon = KFACPreconditioner(
        model,
        factor_update_steps=10,
        inv_update_steps=100,
        lr=0.1,
        inv_update_async=True,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    x = torch.randn(64, 3, 224, 224, device='cuda')
    y = torch.randint(0, 1000, (64,), device='cuda')
    crit = torch.nn.CrossEntropyLoss()
    for _ in range(110):
        opt.zero_grad(set_to_none=True)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            loss = crit(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
    torch.cuda.synchronize()

    # (a) the suspicious small G factors
    print('== small G factor stats + syevd info ==')
    from kfac_amd.layers.eigen import KFACEigenLayer

    gs = []
    for _, (name, layer) in precon._layers.items():
        if not isinstance(layer, KFACEigenLayer):
            continue
        g = layer.g_factor
        if isinstance(g, torch.Tensor) and g.shape[0] in (128, 256):
            gs.append((name, g))
    for name, g in gs[:6]:
        g32 = g.to(torch.float32)
        d, q = ops.eigh_batched(g32.unsqueeze(0).contiguous())
        rec = (q[0] * d[0]) @ q[0].T
        rec_err = torch.linalg.norm(rec - g32) / torch.linalg.norm(g32)
        fin = torch.isfinite(g32).all()
        print(
            f'  {name:<24} n={g.shape[0]} norm={float(torch.linalg.norm(g32)):.3e} '
            f'diag[min,max]=({float(g32.diagonal().min()):.2e},'
            f'{float(g32.diagonal().max()):.2e}) finite={bool(fin)} '
            f'rec={float(rec_err):.2e}',
        )
        w_ref = torch.linalg.eigvalsh(g32.to(torch.float64))
        print(
            f'      eig range [{float(w_ref.min()):.2e}, '
            f'{float(w_ref.max()):.2e}] torch_eigh_rec: ',
            end='',
        )
        w2, v2 = torch.linalg.eigh(g32)
        rec2 = (v2 * w2) @ v2.T
        print(
            f'{float(torch.linalg.norm(rec2 - g32) / torch.linalg.norm(g32)):.2e}',
        )

    # (b) forced phase wall time, twice
    for trial in range(3):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        precon._compute_local_inverses()
        precon._broadcast_inverses()
        torch.cuda.synchronize()
        print(
            f'forced phase #{trial}: '
            f'{(time.perf_counter() - t0) * 1000.0:.1f} ms',
        )
    # per-stage breakdown of one more forced phase
    from kfac_amd.layers.eigen import KFACEigenLayer as KEL

    layers = [
        ly for _, (_, ly) in precon._layers.items() if isinstance(ly, KEL)
    ]
    for trial in range(2):
        for ly in layers:
            ly._warm_cooldown_a = 0
            ly._warm_cooldown_g = 0
            ly._warm_phases_a = 0
            ly._warm_phases_g = 0
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        precon._batched_eigh(layers, 'a')
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        precon._batched_eigh(layers, 'g')
        torch.cuda.synchronize()
        t2 = time.perf_counter()
        print(
            f'batched_eigh (cooldowns cleared) A: {(t1 - t0) * 1e3:.1f} ms, '
            f'G: {(t2 - t1) * 1e3:.1f} ms',
        )


if __name__ == '__main__':
    main()
