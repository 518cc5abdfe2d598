"""Phase-level timing of the K-FAC pipeline on ResNet-50 (bs 64, bf16).

Times each pipeline phase with cuda-sync brackets, for both the HIP
kernel path and the torch-eager (reference-algorithm) path:
  - cov: factor accumulation (fwd+bwd hooks) cost = (fwd+bwd with
    factor step) - (fwd+bwd without)
  - eigh: compute_a_inv/compute_g_inv for all layers
  - precond: preconditioned_grad for all layers
  - klclip+update: grad scale + write-back

Usage: python scripts/profile_phases.py [--eager]
"""

from __future__ import annotations

import argparse
import sys
import time

import torch

sys.path.insert(0, '.')


def sync_time() -> float:
    torch.cuda.synchronize()
    return time.perf_counter()


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument('--eager', action='store_true')
    ap.add_argument('--batch-size', type=int, default=64)
    args = ap.parse_args()
    if args.eager:
        import os

        os.environ['KFAC_AMD_FORCE_EAGER'] = '1'

    from kfac_amd import KFACPreconditioner
    from kfac_amd.models import resnet50

    torch.manual_seed(0)
    model = resnet50().cuda()
    precon = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=1000000,  # never inside loop; we call phases manually
        lr=0.1,
        update_factors_in_hook=False,
    )
    x = torch.randn(args.batch_size, 3, 224, 224, device='cuda')
    y = torch.randint(0, 1000, (args.batch_size,), device='cuda')
    crit = torch.nn.CrossEntropyLoss()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)

    def fwd_bwd():
        opt.zero_grad(set_to_none=True)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            loss = crit(model(x), y)
        loss.backward()

    # warmup
    for _ in range(3):
        fwd_bwd()
        precon.reset_batch()
    # fwd+bwd with hooks saving factors (but updates deferred)
    t0 = sync_time()
    for _ in range(5):
        fwd_bwd()
        for _, layer in precon._layers.values():
            layer.reset_batch()
    t_fwdbwd_hooks = (sync_time() - t0) / 5

    # fwd+bwd without factor work (eval-mode hooks skip)
    precon._factor_update_steps = 1000000
    precon._steps = 1
    t0 = sync_time()
    for _ in range(5):
        fwd_bwd()
    t_fwdbwd_plain = (sync_time() - t0) / 5
    precon._factor_update_steps = 1
    precon._steps = 0

    # cov only (explicit)
    fwd_bwd()
    layers = list(precon._layers.values())

    # factor EMA update
    t0 = sync_time()
    for name, layer in layers:
        layer.update_a_factor(0.95)
        layer.update_g_factor(0.95)
    t_ema = sync_time() - t0

    # eigh per layer
    t0 = sync_time()
    for name, layer in layers:
        layer.compute_a_inv(damping=0.001)
    t_eigh_a = sync_time() - t0
    t0 = sync_time()
    for name, layer in layers:
        layer.compute_g_inv(damping=0.001)
    t_eigh_g = sync_time() - t0

    # precond (twice: first has allocation effects)
    for rep in range(2):
        t0 = sync_time()
        for name, layer in layers:
            layer.preconditioned_grad(damping=0.001)
        t_precond = sync_time() - t0
        if rep == 0:
            for name, layer in layers:
                layer.grad  # keep
    # kl clip + update
    t0 = sync_time()
    scale = precon._compute_grad_scale()
    for name, layer in layers:
        layer.update_grad(scale)
    t_update = sync_time() - t0

    # per-layer eigh breakdown for the 6 largest
    sizes = sorted(
        (
            (layer.module.a_factor_shape[0], name)
            for _, (name, layer) in precon._layers.items()
        ),
        reverse=True,
    )[:6]
    by_name = {name: layer for _, (name, layer) in precon._layers.items()}
    eigh_big = []
    for n, name in sizes:
        layer = by_name[name]
        t0 = sync_time()
        layer.compute_a_inv(damping=0.001)
        eigh_big.append((name, n, sync_time() - t0))

    mode = 'eager' if args.eager else 'hip'
    print(f'=== {mode} phase times (s) ===')
    print(f'fwd+bwd with factor hooks : {t_fwdbwd_hooks:.4f}')
    print(f'fwd+bwd plain             : {t_fwdbwd_plain:.4f}')
    print(f'  -> cov cost             : {t_fwdbwd_hooks - t_fwdbwd_plain:.4f}')
    print(f'factor EMA update         : {t_ema:.4f}')
    print(f'eigh A (54 layers)        : {t_eigh_a:.4f}')
    print(f'eigh G (54 layers)        : {t_eigh_g:.4f}')
    print(f'precond chain             : {t_precond:.4f}')
    print(f'klclip + update_grad      : {t_update:.4f}')
    print('largest-factor eigh:')
    for name, n, t in eigh_big:
        print(f'  {name} (n={n}): {t:.4f}')


if __name__ == '__main__':
    main()
