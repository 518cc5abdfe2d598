"""Optimizer + K-FAC preconditioner wiring for the vision examples.

Parity with reference examples/vision/optimizers.py:114: SGD + optional
KFACPreconditioner + LambdaParamScheduler, with the preconditioner's lr
tracking the optimizer's (reference pattern
``lr=lambda x: optimizer.param_groups[0]['lr']``).
"""

from __future__ import annotations

import argparse
from typing import Any

import torch

import kfac_amd
from examples.utils import create_lr_schedule
from kfac_amd.scheduler import LambdaParamScheduler


def get_optimizer(
    model: torch.nn.Module,
    args: argparse.Namespace,
) -> tuple[
    torch.optim.Optimizer,
    kfac_amd.KFACPreconditioner | None,
    list[Any],
]:
    """Build SGD, optional K-FAC preconditioner, and LR/param schedulers."""
    use_kfac = args.kfac_inv_update_steps > 0

    optimizer = torch.optim.SGD(
        model.parameters(),
        lr=args.base_lr,
        momentum=args.momentum,
        weight_decay=args.weight_decay,
    )

    preconditioner = None
    if use_kfac:
        preconditioner = kfac_amd.KFACPreconditioner(
            model,
            factor_update_steps=args.kfac_factor_update_steps,
            inv_update_steps=args.kfac_inv_update_steps,
            damping=args.kfac_damping,
            factor_decay=args.kfac_factor_decay,
            kl_clip=args.kfac_kl_clip,
            lr=lambda x: optimizer.param_groups[0]['lr'],
            accumulation_steps=args.batches_per_allreduce,
            colocate_factors=args.kfac_colocate_factors,
            compute_method=kfac_amd.enums.ComputeMethod.INVERSE
            if args.kfac_inv_method
            else kfac_amd.enums.ComputeMethod.EIGEN,
            grad_worker_fraction=args.kfac_grad_worker_fraction,
            grad_scaler=args.grad_scaler if hasattr(args, 'grad_scaler') else None,
            skip_layers=args.kfac_skip_layers,
        )

    schedulers: list[Any] = []
    if preconditioner is not None and (
        args.kfac_damping_alpha != 1 or args.kfac_update_steps_alpha != 1
    ):

        def damping_lambda(step: int) -> float:
            if args.kfac_damping_decay and step in args.kfac_damping_decay:
                return args.kfac_damping_alpha
            return 1.0

        def steps_lambda(step: int) -> float:
            if (
                args.kfac_update_steps_decay
                and step in args.kfac_update_steps_decay
            ):
                return args.kfac_update_steps_alpha
            return 1.0

        schedulers.append(
            LambdaParamScheduler(
                preconditioner,
                damping_lambda=damping_lambda,
                factor_update_steps_lambda=steps_lambda,
                inv_update_steps_lambda=steps_lambda,
            ),
        )

    lrs = create_lr_schedule(
        getattr(args, 'backend_size', 1),
        args.warmup_epochs,
        args.lr_decay,
    )
    schedulers.append(torch.optim.lr_scheduler.LambdaLR(optimizer, lrs))
    return optimizer, preconditioner, schedulers
