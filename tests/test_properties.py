"""Property-based tests (hypothesis, derandomized for a deterministic CI
gate): wire-format round trips, grid partition algebra, bucket-capacity
boundaries — the invariants the distributed paths rely on."""

from __future__ import annotations

import sys

import torch
from hypothesis import given, settings, strategies as st

sys.path.insert(0, '.')

from kfac_amd.assignment import KAISAAssignment  # noqa: E402
from kfac_amd.distributed import get_triu, fill_triu  # noqa: E402

SETTINGS = settings(derandomize=True, max_examples=50, deadline=None)


@SETTINGS
@given(n=st.integers(min_value=1, max_value=64), seed=st.integers(0, 2**16))
def test_triu_roundtrip_property(n: int, seed: int) -> None:
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, n, generator=g)
    x = x + x.t()
    v = get_triu(x)
    assert v.numel() == n * (n + 1) // 2
    y = fill_triu(x.shape, v)
    torch.testing.assert_close(x, y)


@SETTINGS
@given(
    world_pow=st.integers(min_value=0, max_value=7),
    gw_pick=st.integers(min_value=0, max_value=10),
)
def test_grid_partition_property(world_pow: int, gw_pick: int) -> None:
    world = 2**world_pow
    divisors = [w for w in range(1, world + 1) if world % w == 0]
    gw = divisors[gw_pick % len(divisors)]
    cols = KAISAAssignment.partition_grad_workers(world, gw)
    rows = KAISAAssignment.partition_grad_receivers(world, gw)
    assert sorted(r for s in cols for r in s) == list(range(world))
    assert sorted(r for s in rows for r in s) == list(range(world))
    for c in cols:
        for r in rows:
            assert len(c & r) == 1


@SETTINGS
@given(
    sizes=st.lists(
        st.integers(min_value=1, max_value=2000), min_size=1, max_size=20,
    ),
    cap_kb=st.integers(min_value=1, max_value=64),
)
def test_bucket_capacity_property(sizes: list[int], cap_kb: int) -> None:
    """Buckets never exceed cap unless a single tensor alone does, and
    every element survives the flatten/unflatten round trip."""
    from kfac_amd.distributed import AllreduceTensorBucket

    cap = cap_kb * 1024
    bucket = AllreduceTensorBucket(cap)
    buckets = [bucket]
    tensors = []
    for i, n in enumerate(sizes):
        t = torch.full((n,), float(i))
        tensors.append(t)
        if not bucket.fits(t) and bucket.size > 0:
            # cap respected unless a single tensor alone exceeds it
            assert bucket.size <= cap or len(bucket._tensors) == 1
            bucket.communicate(None, 2.0)
            bucket = AllreduceTensorBucket(cap)
            buckets.append(bucket)
        bucket.append(t)
    if bucket.size > 0:
        bucket.communicate(None, 2.0)
    for b in buckets:
        if b.communicated:
            b.wait_and_unpack()
    # unpack writes the (scale-applied) result back into the originals
    for i, t in enumerate(tensors):
        assert torch.all(t == 2.0 * float(i))
