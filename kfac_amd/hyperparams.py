"""Common hyperparameter schedules (parity: reference kfac/hyperparams.py:8-47)."""

from __future__ import annotations

from typing import Callable


def exp_decay_factor_averaging(
    min_value: float = 0.95,
) -> Callable[[int], float]:
    """Exponentially decaying factor-averaging schedule (Martens 2015).

    Running-average weight at K-FAC step k is ``min(1 - 1/k, min_value)``
    (k=0 is treated as k=1). Pass as ``factor_decay`` to a preconditioner.

    Raises:
        ValueError: if ``min_value <= 0``.
    """
    if min_value <= 0:
        raise ValueError(f'min_value must be positive, got {min_value}')

    def _factor_weight(step: int) -> float:
        if step < 0:
            raise ValueError(f'negative step ({step}) passed to schedule')
        if step == 0:
            step = 1
        return min(1 - (1 / step), min_value)

    return _factor_weight
