"""Stage-by-stage CPU-vs-GPU comparison of the K-FAC pipeline on LeNet."""

from __future__ import annotations

import sys

import torch

sys.path.insert(0, '.')

from kfac_amd import KFACPreconditioner
from testing.models import LeNet


def run(device: str):
    torch.manual_seed(123)
    model = LeNet().to(device)
    x = torch.randn(32, 1, 28, 28, device=device)
    y = torch.randint(0, 10, (32,), device=device)
    precon = KFACPreconditioner(
        model, factor_update_steps=1, inv_update_steps=1, lr=0.01,
    )
    loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()
    # grads BEFORE precondition
    grads_pre = {
        n: p.grad.detach().cpu().clone() for n, p in model.named_parameters()
    }
    precon.step()
    out = {
        'loss': loss.item(),
        'grads_pre': grads_pre,
        'grads_post': {
            n: p.grad.detach().cpu().clone()
            for n, p in model.named_parameters()
        },
        'factors': {},
        'eig': {},
    }
    for _, (name, layer) in precon._layers.items():
        out['factors'][name] = {
            'A': layer.a_factor.detach().cpu().clone(),
            'G': layer.g_factor.detach().cpu().clone(),
        }
        out['eig'][name] = {
            'qa': layer.qa.detach().cpu().clone(),
            'qg': layer.qg.detach().cpu().clone(),
            'dgda': layer.dgda.detach().cpu().clone(),
        }
    return out


cpu = run('cpu')
gpu = run('cuda')

print('loss:', cpu['loss'], gpu['loss'])
for name in cpu['factors']:
    fa = (cpu['factors'][name]['A'] - gpu['factors'][name]['A']).abs().max()
    fg = (cpu['factors'][name]['G'] - gpu['factors'][name]['G']).abs().max()
    qa = (cpu['eig'][name]['qa'].abs() - gpu['eig'][name]['qa'].abs()).abs().max()
    dd = (cpu['eig'][name]['dgda'] - gpu['eig'][name]['dgda']).abs().max()
    print(f'{name}: dA={fa:.2e} dG={fg:.2e} |qa| diff={qa:.2e} dgda={dd:.2e}')
for name in cpu['grads_pre']:
    d_pre = (cpu['grads_pre'][name] - gpu['grads_pre'][name]).abs().max()
    d_post = (cpu['grads_post'][name] - gpu['grads_post'][name]).abs().max()
    scale = cpu['grads_post'][name].abs().max()
    print(f'{name}: pre={d_pre:.2e} post={d_post:.2e} post_scale={scale:.2e}')
