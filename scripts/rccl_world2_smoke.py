"""Attempt 2-rank RCCL comm construction on ONE GPU (diagnostic).

NCCL/RCCL historically refuses multiple ranks on one device
("Duplicate GPU detected"); this smoke records what RCCL 7.2 actually
does so the first true multi-GPU contact (the driver's 8-GPU run) is
not the first time the code path executes. Outcome is informational:
rc 0 with either __W2_OK__ or __W2_REFUSED__ printed.

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 scripts/rccl_world2_smoke.py
"""

from __future__ import annotations

import datetime
import os
import sys

import torch
import torch.distributed as dist


def main() -> None:
    rank = int(os.environ['RANK'])
    torch.cuda.set_device(0)  # both ranks: the one GPU
    try:
        dist.init_process_group(
            'nccl', timeout=datetime.timedelta(seconds=60),
        )
        x = torch.ones(1024, device='cuda') * (rank + 1)
        dist.all_reduce(x)
        torch.cuda.synchronize()
        ok = float(x[0]) == 3.0
        print(f'[rank {rank}] __W2_OK__ allreduce sum={float(x[0])} ok={ok}')
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001 - diagnostic by design
        print(f'[rank {rank}] __W2_REFUSED__ {type(e).__name__}: {e}')
        sys.exit(0)


if __name__ == '__main__':
    main()
