"""Multi-rank GPU coverage for the driver's round-end gate.

Launches the 2-ranks-on-one-GPU torchrun smoke (gloo backend with CUDA
tensors) as a subprocess: exercises the KAISA HYBRID/MEM-OPT collectives,
grad/inverse broadcasts and the async-inverse swap on a real GPU — the
same code paths the 8-GPU RCCL scaling run takes (reference parity:
tests/distributed_test.py + gpt_neox integration, run CPU-only there).
"""

from __future__ import annotations

import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.skipif(not torch.cuda.is_available(), reason='no GPU')
def test_two_rank_torchrun_smoke() -> None:
    env = dict(os.environ)
    env.setdefault('MASTER_ADDR', '127.0.0.1')
    proc = subprocess.run(
        [
            sys.executable,
            '-m',
            'torch.distributed.run',
            '--standalone',
            '--local-addr',
            '127.0.0.1',
            '--nproc-per-node',
            '2',
            os.path.join(REPO, 'scripts', 'dist_async_smoke.py'),
        ],
        cwd=REPO,
        env=env,
        capture_output=True,
        text=True,
        timeout=240,
    )
    assert proc.returncode == 0, (
        f'torchrun smoke failed\nstdout:\n{proc.stdout[-2000:]}\n'
        f'stderr:\n{proc.stderr[-2000:]}'
    )
    assert 'dist async smoke ok' in proc.stdout + proc.stderr
