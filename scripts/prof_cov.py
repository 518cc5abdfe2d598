"""Run one cov shape in a loop (for rocprofv3 PMC counter capture)."""

from __future__ import annotations

import sys

import torch

sys.path.insert(0, '.')

from kfac_amd import _kfaccore  # noqa: E402

shape = sys.argv[1] if len(sys.argv) > 1 else '2304'

if shape == '2304':
    C, H, W, k, s, p = 256, 14, 14, 3, 1, 1
elif shape == '4608':
    C, H, W, k, s, p = 512, 7, 7, 3, 1, 1
elif shape == '576':
    C, H, W, k, s, p = 64, 56, 56, 3, 1, 1
else:
    raise SystemExit(f'unknown shape {shape}')

x = torch.randn(64, C, H, W, device='cuda', dtype=torch.bfloat16)
n = C * k * k
out = torch.zeros(n, n, device='cuda')
for _ in range(10):
    _kfaccore.cov_conv_a(x, out, k, k, s, s, p, p, False, 0.0, 1.0)
torch.cuda.synchronize()
