"""Function tracing / timing utilities.

Parity with reference kfac/tracing.py:19-108 (@trace decorator,
get_trace/log_trace/clear_trace), extended for HIP: wall clock over async
HIP streams under-reports, so ``trace(cuda_sync=True)`` brackets the call
with ``torch.cuda.synchronize()`` to charge queued GPU work to the
function that launched it, and roctx-style named ranges are emitted via
``torch.cuda.nvtx`` (maps to roctx on ROCm) so traced sections line up in
rocprofv3 timelines.
"""

from __future__ import annotations

import logging
import time
from typing import Any
from typing import Callable

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

_func_traces: dict[str, list[float]] = {}


def clear_trace() -> None:
    """Drop all recorded timings."""
    _func_traces.clear()


def get_trace(
    average: bool = True,
    max_history: int | None = None,
) -> dict[str, float]:
    """Per-function timing summary.

    Args:
        average: report the mean over calls instead of the sum.
        max_history: only consider the last ``max_history`` calls.
    """
    out: dict[str, float] = {}
    for name, times in _func_traces.items():
        if max_history is not None:
            times = times[-max_history:]
        if len(times) == 0:
            continue
        out[name] = sum(times) / len(times) if average else sum(times)
    return out


def log_trace(
    average: bool = True,
    max_history: int | None = None,
    loglevel: int = logging.INFO,
) -> None:
    """Log the timing summary on rank 0."""
    if dist.is_available() and dist.is_initialized() and dist.get_rank() != 0:
        return
    for name, value in get_trace(average, max_history).items():
        logger.log(loglevel, f'{name}: {value:.6f}s')


def trace(
    sync: bool = False,
    cuda_sync: bool = False,
) -> Callable[[Callable[..., Any]], Callable[..., Any]]:
    """Decorator recording wall time per call.

    Args:
        sync: bracket the call with dist.barrier() for honest
            distributed timings (reference tracing.py:93-97).
        cuda_sync: bracket with torch.cuda.synchronize() so queued HIP
            work is charged to this call.
    """

    def decorator(func: Callable[..., Any]) -> Callable[..., Any]:
        name = func.__qualname__

        def wrapper(*args: Any, **kwargs: Any) -> Any:
            use_dist = sync and dist.is_available() and dist.is_initialized()
            use_cuda = cuda_sync and torch.cuda.is_available()
            if use_dist:
                dist.barrier()
            if use_cuda:
                torch.cuda.synchronize()
                torch.cuda.nvtx.range_push(name)
            start = time.perf_counter()
            result = func(*args, **kwargs)
            if use_dist:
                dist.barrier()
            if use_cuda:
                torch.cuda.synchronize()
                torch.cuda.nvtx.range_pop()
            _func_traces.setdefault(name, []).append(
                time.perf_counter() - start,
            )
            return result

        wrapper.__name__ = func.__name__
        wrapper.__qualname__ = name
        wrapper.__doc__ = func.__doc__
        return wrapper

    return decorator
