"""Multi-node-without-a-cluster test harness.

Forks N OS processes per test, each initializing torch.distributed with
the gloo backend on 127.0.0.1 (CPU gloo stands in for RCCL — same
semantics, SURVEY.md §4 / reference testing/distributed.py:24-141).
Failures in any rank surface as pytest failures via exit codes; hangs via
join timeout.
"""

from __future__ import annotations

import multiprocessing
import os
import socket
import sys
import traceback
from typing import Any
from typing import Callable

import torch.distributed as dist


def find_free_port() -> int:
    """Ask the OS for a free TCP port on 127.0.0.1."""
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(('127.0.0.1', 0))
        return s.getsockname()[1]


def _worker(
    rank: int,
    world_size: int,
    port: int,
    func: Callable[..., Any],
    args: tuple,
    kwargs: dict,
) -> None:
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world_size)
    try:
        import torch

        torch.set_num_threads(1)
        dist.init_process_group('gloo', rank=rank, world_size=world_size)
        func(*args, **kwargs)
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        traceback.print_exc()
        sys.stderr.flush()
        os._exit(1)
    os._exit(0)


def run_distributed(
    world_size: int,
    func: Callable[..., Any],
    *args: Any,
    timeout: float = 120.0,
    **kwargs: Any,
) -> None:
    """Run ``func`` in ``world_size`` forked processes under gloo.

    Raises AssertionError if any rank fails or hangs.
    """
    ctx = multiprocessing.get_context('fork')
    port = find_free_port()
    procs = [
        ctx.Process(
            target=_worker,
            args=(rank, world_size, port, func, args, kwargs),
        )
        for rank in range(world_size)
    ]
    for p in procs:
        p.start()
    failed = []
    for rank, p in enumerate(procs):
        p.join(timeout)
        if p.is_alive():
            p.terminate()
            p.join(5)
            failed.append((rank, 'timeout'))
        elif p.exitcode != 0:
            failed.append((rank, f'exit={p.exitcode}'))
    if failed:
        raise AssertionError(f'distributed test failed on ranks: {failed}')


def distributed_test(
    world_size: int | list[int] = 1,
    timeout: float = 120.0,
) -> Callable[[Callable[..., Any]], Callable[..., Any]]:
    """Decorator: run the test body in forked gloo process groups.

    ``world_size`` may be a list to sweep sizes
    (reference testing/distributed.py:128-133).
    """
    sizes = [world_size] if isinstance(world_size, int) else list(world_size)

    def decorator(func: Callable[..., Any]) -> Callable[..., Any]:
        def wrapper(*args: Any, **kwargs: Any) -> None:
            for ws in sizes:
                run_distributed(ws, func, *args, timeout=timeout, **kwargs)

        wrapper.__name__ = func.__name__
        wrapper.__doc__ = func.__doc__
        return wrapper

    return decorator
