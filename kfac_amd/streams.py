"""Side-stream management for overlap of K-FAC work with backward.

On MI355X the covariance kernels are memory/MFMA work that can run
concurrently with the (largely compute-bound) backward kernels: the
hooks enqueue factor accumulation on a dedicated HIP side stream, and
``preconditioner.step()`` joins it once. Factor allreduces issued from
the side stream context land on RCCL's comm stream ordered after the
covariance kernels, so comm still overlaps backward like the reference's
hook-launched allreduce.
"""

from __future__ import annotations

import torch

_streams: dict[int, torch.cuda.Stream] = {}


def cov_stream(device: torch.device) -> torch.cuda.Stream:
    """The per-device covariance side stream."""
    idx = device.index if device.index is not None else torch.cuda.current_device()
    if idx not in _streams:
        _streams[idx] = torch.cuda.Stream(device=idx)
    return _streams[idx]


def join_cov_stream(device: torch.device) -> None:
    """Make the current stream wait for all queued covariance work."""
    idx = device.index if device.index is not None else torch.cuda.current_device()
    s = _streams.get(idx)
    if s is not None:
        torch.cuda.current_stream(device).wait_stream(s)
