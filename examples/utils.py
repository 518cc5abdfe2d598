"""Shared example utilities (parity: reference examples/utils.py)."""

from __future__ import annotations

import os
from typing import Callable

import torch
import torch.distributed as dist


class Metric:
    """Distributed-averaged running metric (reference utils.py:66-89)."""

    def __init__(self, name: str):
        self.name = name
        self.total = torch.tensor(0.0)
        self.n = torch.tensor(0.0)

    def update(self, val: torch.Tensor, n: int = 1) -> None:
        val = val.detach().cpu() / n
        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(val, op=dist.ReduceOp.SUM)
            val /= dist.get_world_size()
        self.total += val
        self.n += 1

    @property
    def avg(self) -> torch.Tensor:
        return self.total / self.n


class LabelSmoothLoss(torch.nn.Module):
    """Cross entropy with label smoothing (reference utils.py)."""

    def __init__(self, smoothing: float = 0.0):
        super().__init__()
        self.smoothing = smoothing

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        log_prob = torch.nn.functional.log_softmax(input, dim=-1)
        weight = (
            input.new_ones(input.size())
            * self.smoothing
            / (input.size(-1) - 1.0)
        )
        weight.scatter_(-1, target.unsqueeze(-1), 1.0 - self.smoothing)
        return (-weight * log_prob).sum(dim=-1).mean()


def save_checkpoint(
    model: torch.nn.Module,
    optimizer: torch.optim.Optimizer,
    preconditioner: object | None,
    schedulers: list[object],
    filepath: str,
    **extra: object,
) -> None:
    """Bundle model/optimizer/preconditioner/scheduler state to one file.

    Reference examples/utils.py:20-38.
    """
    state = {
        'model': model.state_dict(),
        'optimizer': optimizer.state_dict(),
        'preconditioner': (
            preconditioner.state_dict()  # type: ignore[attr-defined]
            if preconditioner is not None
            else None
        ),
        'schedulers': [
            s.state_dict() if hasattr(s, 'state_dict') else None
            for s in schedulers
        ],
        **extra,
    }
    tmp = filepath + '.tmp'
    torch.save(state, tmp)
    os.replace(tmp, filepath)


def create_lr_schedule(
    workers: int,
    warmup_epochs: int,
    decay_schedule: list[int],
    alpha: float = 0.1,
) -> Callable[[int], float]:
    """Warmup to linear-scaled LR, then staircase decay.

    Reference examples/utils.py:92-114.
    """

    def lr_schedule(epoch: int) -> float:
        lr_adj = 1.0
        if epoch < warmup_epochs:
            lr_adj = (
                1.0 / workers
                * (epoch * (workers - 1) / warmup_epochs + 1)
            )
        else:
            # one factor of alpha per decay epoch already passed
            # (reference examples/utils.py:108-111 multiplies alpha for
            # every e <= epoch)
            lr_adj = alpha ** sum(1 for e in decay_schedule if epoch >= e)
        return lr_adj

    return lr_schedule


def accuracy(output: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Top-1 accuracy."""
    pred = output.argmax(dim=1)
    return (pred == target).float().mean()
