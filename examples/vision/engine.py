"""Train/validate loops for the vision examples.

Parity with reference examples/vision/engine.py:152: grad accumulation
with ``model.no_sync()``, AMP GradScaler unscale BEFORE preconditioner
step, allreduce-averaged metrics.
"""

from __future__ import annotations

import contextlib
from typing import Any

import torch

from examples.utils import accuracy
from examples.utils import Metric


def train(
    epoch: int,
    model: torch.nn.Module,
    optimizer: torch.optim.Optimizer,
    preconditioner: Any | None,
    loss_func: torch.nn.Module,
    train_loader: Any,
    device: torch.device,
    scaler: torch.cuda.amp.GradScaler | None = None,
    accumulation_steps: int = 1,
    log_interval: int = 50,
    max_steps: int | None = None,
) -> Metric:
    """One training epoch; returns the averaged loss metric."""
    model.train()
    sampler = getattr(train_loader, 'sampler', None)
    if hasattr(sampler, 'set_epoch'):
        sampler.set_epoch(epoch)
    train_loss = Metric('train_loss')
    train_acc = Metric('train_acc')

    optimizer.zero_grad()
    for batch_idx, (data, target) in enumerate(train_loader):
        if max_steps is not None and batch_idx >= max_steps:
            break
        data = data.to(device, non_blocking=True)
        target = target.to(device, non_blocking=True)

        boundary = (batch_idx + 1) % accumulation_steps == 0
        # Skip DDP grad sync on non-boundary accumulation steps.
        ctx = (
            model.no_sync()
            if not boundary and hasattr(model, 'no_sync')
            else contextlib.nullcontext()
        )
        with ctx:
            if scaler is not None:
                with torch.autocast(device.type, dtype=torch.float16):
                    output = model(data)
                    loss = loss_func(output, target)
                scaler.scale(loss / accumulation_steps).backward()
            else:
                output = model(data)
                loss = loss_func(output, target)
                (loss / accumulation_steps).backward()

        train_loss.update(loss.detach(), 1)
        train_acc.update(accuracy(output.detach(), target), 1)

        if boundary:
            if scaler is not None:
                # Unscale so the preconditioner sees true gradients
                # (reference engine.py:80-88).
                scaler.unscale_(optimizer)
            if preconditioner is not None:
                preconditioner.step()
            if scaler is not None:
                scaler.step(optimizer)
                scaler.update()
            else:
                optimizer.step()
            optimizer.zero_grad()

        if batch_idx % log_interval == 0 and _is_rank_zero():
            print(
                f'Epoch {epoch} [{batch_idx}/{len(train_loader)}] '
                f'loss={float(train_loss.avg):.4f} '
                f'acc={float(train_acc.avg):.4f}',
            )
    return train_loss


@torch.no_grad()
def validate(
    epoch: int,
    model: torch.nn.Module,
    loss_func: torch.nn.Module,
    val_loader: Any,
    device: torch.device,
    max_steps: int | None = None,
) -> tuple[Metric, Metric]:
    """Validation pass; returns (loss, accuracy) metrics."""
    model.eval()
    val_loss = Metric('val_loss')
    val_acc = Metric('val_acc')
    for batch_idx, (data, target) in enumerate(val_loader):
        if max_steps is not None and batch_idx >= max_steps:
            break
        data = data.to(device, non_blocking=True)
        target = target.to(device, non_blocking=True)
        output = model(data)
        val_loss.update(loss_func(output, target), 1)
        val_acc.update(accuracy(output, target), 1)
    if _is_rank_zero():
        print(
            f'Epoch {epoch} validation: loss={float(val_loss.avg):.4f} '
            f'acc={float(val_acc.avg):.4f}',
        )
    return val_loss, val_acc


def _is_rank_zero() -> bool:
    import torch.distributed as dist

    return not (dist.is_available() and dist.is_initialized()) or dist.get_rank() == 0
