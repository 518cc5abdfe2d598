"""Communicator tests: allreduce/broadcast/bucketing/triu at world 1-4.

Mirrors the coverage of reference tests/distributed_test.py:25-313.
"""

from __future__ import annotations

import sys

import pytest
import torch
import torch.distributed as dist

sys.path.insert(0, '.')

from kfac_amd.distributed import get_rank  # noqa: E402
from kfac_amd.distributed import get_world_size  # noqa: E402
from kfac_amd.distributed import NonSquareTensorError  # noqa: E402
from kfac_amd.distributed import TorchDistributedCommunicator  # noqa: E402
from testing.distributed import distributed_test  # noqa: E402


def test_rank_world_uninitialized() -> None:
    assert get_rank() == 0
    assert get_world_size() == 1


def _maybe_wait(x):
    return x.wait() if hasattr(x, 'wait') else x


@distributed_test(world_size=4)
def _allreduce_average() -> None:
    comm = TorchDistributedCommunicator()
    rank = dist.get_rank()
    t = torch.ones(4, 4) * (rank + 1)
    result = _maybe_wait(comm.allreduce(t))
    # mean of 1..4 = 2.5
    torch.testing.assert_close(result, torch.full((4, 4), 2.5))


@distributed_test(world_size=4)
def _allreduce_symmetric() -> None:
    comm = TorchDistributedCommunicator()
    rank = dist.get_rank()
    base = torch.arange(16, dtype=torch.float32).reshape(4, 4)
    t = (base + base.t()) * (rank + 1)
    expected = (base + base.t()) * 2.5
    result = _maybe_wait(comm.allreduce(t, symmetric=True))
    torch.testing.assert_close(result, expected)


@distributed_test(world_size=2)
def _allreduce_symmetric_nonsquare_raises() -> None:
    comm = TorchDistributedCommunicator()
    try:
        comm.allreduce(torch.ones(2, 3), symmetric=True)
    except NonSquareTensorError:
        return
    raise AssertionError('expected NonSquareTensorError')


@distributed_test(world_size=4)
def _broadcast() -> None:
    comm = TorchDistributedCommunicator()
    rank = dist.get_rank()
    t = torch.ones(3, 3) * rank
    result = _maybe_wait(comm.broadcast(t, src=2))
    torch.testing.assert_close(result, torch.full((3, 3), 2.0))


@distributed_test(world_size=4)
def _broadcast_symmetric() -> None:
    comm = TorchDistributedCommunicator()
    rank = dist.get_rank()
    base = torch.arange(9, dtype=torch.float32).reshape(3, 3)
    t = (base + base.t()) * (rank + 1)
    result = _maybe_wait(comm.broadcast(t, src=1, symmetric=True))
    torch.testing.assert_close(result, (base + base.t()) * 2)


@distributed_test(world_size=4)
def _bucketed_allreduce() -> None:
    comm = TorchDistributedCommunicator(bucket_cap_mb=25)
    rank = dist.get_rank()
    tensors = [torch.ones(5, 5) * (rank + 1) * (i + 1) for i in range(6)]
    futures = [comm.allreduce_bucketed(t) for t in tensors]
    comm.flush_allreduce_buckets()
    for i, fut in enumerate(futures):
        result = _maybe_wait(fut)
        torch.testing.assert_close(result, torch.full((5, 5), 2.5 * (i + 1)))


@distributed_test(world_size=2)
def _bucketed_allreduce_overflow() -> None:
    # bucket cap 1 KB: each 20x20 fp32 tensor = 1600 B > cap, so every
    # tensor gets its own bucket and correctness must be unaffected.
    comm = TorchDistributedCommunicator(bucket_cap_mb=1.0 / 1024)
    rank = dist.get_rank()
    tensors = [torch.ones(20, 20) * (rank + 1) * (i + 1) for i in range(3)]
    futures = [comm.allreduce_bucketed(t) for t in tensors]
    comm.flush_allreduce_buckets()
    for i, fut in enumerate(futures):
        result = _maybe_wait(fut)
        torch.testing.assert_close(result, torch.full((20, 20), 1.5 * (i + 1)))


@distributed_test(world_size=4)
def _bucketed_allreduce_symmetric() -> None:
    comm = TorchDistributedCommunicator()
    rank = dist.get_rank()
    base = torch.arange(36, dtype=torch.float32).reshape(6, 6)
    sym = base + base.t()
    futures = [
        comm.allreduce_bucketed(sym * (rank + 1), symmetric=True)
        for _ in range(3)
    ]
    comm.flush_allreduce_buckets()
    for fut in futures:
        torch.testing.assert_close(_maybe_wait(fut), sym * 2.5)


@distributed_test(world_size=4)
def _bucketed_wait_without_flush() -> None:
    # Waiting a future before flush must trigger the flush itself.
    comm = TorchDistributedCommunicator()
    rank = dist.get_rank()
    fut = comm.allreduce_bucketed(torch.ones(4) * (rank + 1))
    result = _maybe_wait(fut)
    torch.testing.assert_close(result, torch.full((4,), 2.5))


@distributed_test(world_size=[1, 4])
def _allreduce_world1_passthrough() -> None:
    comm = TorchDistributedCommunicator()
    t = torch.ones(2, 2)
    if dist.get_world_size() == 1:
        assert comm.allreduce(t) is t


def test_allreduce_average() -> None:
    _allreduce_average()


def test_allreduce_symmetric() -> None:
    _allreduce_symmetric()


def test_allreduce_symmetric_nonsquare() -> None:
    _allreduce_symmetric_nonsquare_raises()


def test_broadcast() -> None:
    _broadcast()


def test_broadcast_symmetric() -> None:
    _broadcast_symmetric()


def test_bucketed_allreduce() -> None:
    _bucketed_allreduce()


def test_bucketed_allreduce_overflow() -> None:
    _bucketed_allreduce_overflow()


def test_bucketed_allreduce_symmetric() -> None:
    _bucketed_allreduce_symmetric()


def test_bucketed_wait_without_flush() -> None:
    _bucketed_wait_without_flush()


def test_allreduce_world1() -> None:
    _allreduce_world1_passthrough()


def _bucket_invariants_body() -> None:
    from kfac_amd.distributed import AllreduceTensorBucket as _Bucket

    b = _Bucket(1024)
    t = torch.ones(4)
    b.append(t)
    b.communicate(None, 1.0)
    with pytest.raises(RuntimeError):
        b.communicate(None, 1.0)
    with pytest.raises(RuntimeError):
        b.append(torch.ones(2))
    b.wait_and_unpack()

    b2 = _Bucket(1024)
    assert b2.size == 0
    b2.add_tensor(torch.ones(4))
    assert b2.size == 16 and not b2.communicated()
    with pytest.raises(RuntimeError):
        b2.wait_and_unpack()


def test_bucket_invariants() -> None:
    """Misuse of a bucket raises (single-flight discipline,
    reference distributed.py:89-95,180-188)."""
    from testing.distributed import run_distributed

    run_distributed(1, _bucket_invariants_body)


def _inflight_pruned_worker() -> None:
    import torch

    from kfac_amd.distributed import TorchDistributedCommunicator

    comm = TorchDistributedCommunicator(bucket_cap_mb=0.001)
    futures = [
        comm.allreduce_bucketed(torch.full((64, 64), float(i)))
        for i in range(8)
    ]
    comm.flush_allreduce_buckets()
    for i, f in enumerate(futures):
        t = f.wait() if hasattr(f, 'wait') else f
        assert torch.allclose(t, torch.full((64, 64), float(i)))
    # buckets whose tensors were unpacked must be dropped on the next
    # flush (regression: in-flight list grew without bound)
    comm.flush_allreduce_buckets()
    assert comm._inflight == []


def test_inflight_buckets_pruned() -> None:
    from testing.distributed import run_distributed

    run_distributed(2, _inflight_pruned_worker)
