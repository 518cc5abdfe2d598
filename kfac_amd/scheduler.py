"""Multiplicative hyperparameter scheduler.

Parity with reference kfac/scheduler.py:10-167: per-step multiplicative
updates to the preconditioner's scalar hyperparameters; refuses to manage
a parameter that is already a callable.
"""

from __future__ import annotations

from typing import Callable

from kfac_amd.base_preconditioner import BaseKFACPreconditioner

_PARAMS = (
    'factor_update_steps',
    'inv_update_steps',
    'damping',
    'factor_decay',
    'kl_clip',
    'lr',
)
_INT_PARAMS = {'factor_update_steps', 'inv_update_steps'}


class LambdaParamScheduler:
    """Multiplicative per-step scheduler for K-FAC hyperparameters.

    Each lambda maps the current K-FAC step to a multiplicative factor
    applied to the corresponding scalar hyperparameter. Call
    ``scheduler.step()`` after ``preconditioner.step()``.
    """

    def __init__(
        self,
        preconditioner: BaseKFACPreconditioner,
        *,
        factor_update_steps_lambda: Callable[[int], float] | None = None,
        inv_update_steps_lambda: Callable[[int], float] | None = None,
        damping_lambda: Callable[[int], float] | None = None,
        factor_decay_lambda: Callable[[int], float] | None = None,
        kl_clip_lambda: Callable[[int], float] | None = None,
        lr_lambda: Callable[[int], float] | None = None,
    ) -> None:
        """Init LambdaParamScheduler.

        Raises:
            ValueError: if a lambda is given for a parameter that is
                already a callable on the preconditioner.
        """
        self._preconditioner = preconditioner
        self._lambdas: dict[str, Callable[[int], float] | None] = {
            'factor_update_steps': factor_update_steps_lambda,
            'inv_update_steps': inv_update_steps_lambda,
            'damping': damping_lambda,
            'factor_decay': factor_decay_lambda,
            'kl_clip': kl_clip_lambda,
            'lr': lr_lambda,
        }
        for name, fn in self._lambdas.items():
            if fn is None:
                continue
            current = getattr(preconditioner, f'_{name}')
            if callable(current):
                raise ValueError(
                    f'preconditioner.{name} is already a callable and '
                    'cannot be updated by the LambdaParamScheduler.',
                )
            if current is None:
                # e.g. kl_clip=None disables clipping; multiplying None
                # at step time would be a bare TypeError much later
                raise ValueError(
                    f'preconditioner.{name} is None (disabled) and '
                    'cannot be scheduled.',
                )

    def step(self, step: int | None = None) -> None:
        """Apply one multiplicative update (call after preconditioner.step())."""
        for name in _PARAMS:
            fn = self._lambdas[name]
            if fn is None:
                continue
            at = step if step is not None else self._preconditioner.steps
            factor = fn(at)
            attr = f'_{name}'
            current = getattr(self._preconditioner, attr)
            assert not callable(current)
            new = current * factor
            if name in _INT_PARAMS:
                new = int(new)
                if new < 1:
                    # A decayed step-count hitting 0 would crash the
                    # ``steps % update_steps`` modulo in the hooks; the
                    # reference silently truncates (kfac/scheduler.py) —
                    # fail loudly instead.
                    raise ValueError(
                        f'{name} schedule produced {current * factor:.4g} '
                        f'at step {at}, which truncates to {new}; '
                        'step-count hyperparameters must stay >= 1.',
                    )
            setattr(self._preconditioner, attr, new)
