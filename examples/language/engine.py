"""Train/eval loops for the LM example (reference examples/language/engine.py)."""

from __future__ import annotations

import math
import time
from typing import Any
from typing import Callable

import torch


def train_epoch(
    model: torch.nn.Module,
    optimizer: torch.optim.Optimizer,
    preconditioner: Any | None,
    criterion: torch.nn.Module,
    batch_fn: Callable[[int], tuple[torch.Tensor, torch.Tensor]],
    steps: int,
    vocab: int,
    grad_clip: float = 0.5,  # reference examples/language/engine.py:53
) -> float:
    """One epoch over ``steps`` batches; returns train perplexity."""
    model.train()
    total_loss = 0.0
    for i in range(steps):
        data, target = batch_fn(i)
        optimizer.zero_grad()
        output = model(data)
        loss = criterion(output.view(-1, vocab), target)
        loss.backward()
        if grad_clip:
            torch.nn.utils.clip_grad_norm_(model.parameters(), grad_clip)
        if preconditioner is not None:
            preconditioner.step()
        optimizer.step()
        total_loss += loss.item()
    return math.exp(total_loss / steps)


@torch.no_grad()
def evaluate(
    model: torch.nn.Module,
    criterion: torch.nn.Module,
    batch_fn: Callable[[int], tuple[torch.Tensor, torch.Tensor]],
    steps: int,
    vocab: int,
) -> float:
    """Evaluation perplexity over ``steps`` batches."""
    model.eval()
    total_loss = 0.0
    for i in range(steps):
        data, target = batch_fn(i)
        output = model(data)
        total_loss += criterion(output.view(-1, vocab), target).item()
    return math.exp(total_loss / steps)


def run_training(
    model: torch.nn.Module,
    optimizer: torch.optim.Optimizer,
    preconditioner: Any | None,
    batch_fn: Callable[[int], tuple[torch.Tensor, torch.Tensor]],
    epochs: int,
    steps_per_epoch: int,
    vocab: int,
    rank: int = 0,
) -> None:
    criterion = torch.nn.CrossEntropyLoss()
    for epoch in range(epochs):
        t0 = time.time()
        ppl = train_epoch(
            model, optimizer, preconditioner, criterion, batch_fn,
            steps_per_epoch, vocab,
        )
        if rank == 0:
            print(f'epoch {epoch}: ppl={ppl:.2f} ({time.time() - t0:.1f}s)')
