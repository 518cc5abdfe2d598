"""Property tests for the KAISA grid assignment."""

from __future__ import annotations

import pytest

from kfac_amd.assignment import KAISAAssignment


def _mock_group(ranks):
    return tuple(ranks)


@pytest.mark.parametrize('world', [1, 2, 4, 8, 16])
def test_grid_partition_properties(world: int) -> None:
    for gw in [w for w in range(1, world + 1) if world % w == 0]:
        cols = KAISAAssignment.partition_grad_workers(world, gw)
        rows = KAISAAssignment.partition_grad_receivers(world, gw)
        # partitions cover the world disjointly
        assert sorted(r for s in cols for r in s) == list(range(world))
        assert sorted(r for s in rows for r in s) == list(range(world))
        assert all(len(c) == gw for c in cols)
        assert all(len(r) == world // gw for r in rows)
        # every column x row intersect in exactly one rank
        for c in cols:
            for r in rows:
                assert len(c & r) == 1


def test_partition_invalid() -> None:
    with pytest.raises(ValueError):
        KAISAAssignment.partition_grad_workers(8, 3)
    with pytest.raises(ValueError):
        KAISAAssignment.partition_grad_workers(0, 1)


@pytest.mark.parametrize('world,frac', [(1, 1.0), (4, 1.0), (4, 0.5), (8, 0.25), (8, 1.0 / 8)])
def test_assignment_invariants(world: int, frac: float) -> None:
    work = {
        f'layer{i}': {'A': float((i + 1) ** 3), 'G': float((i + 2) ** 3)}
        for i in range(10)
    }
    for rank in range(world):
        asn = KAISAAssignment(
            work,
            local_rank=rank,
            world_size=world,
            grad_worker_fraction=frac,
            group_func=_mock_group,
            colocate_factors=True,
        )
        assert set(asn.get_layers()) == set(work.keys())
        for layer in asn.get_layers():
            a_w = asn.inv_worker(layer, 'A')
            g_w = asn.inv_worker(layer, 'G')
            assert a_w == g_w  # colocated
            # inverse worker within the layer's grad worker group
            assert a_w in asn._grad_worker_groups[layer].ranks
            # this rank's receiver group contains it
            assert rank in asn._grad_receiver_groups[layer].ranks
            # src grad worker is in both groups
            src = asn.src_grad_worker(layer)
            assert src in asn._grad_worker_groups[layer].ranks
            assert src in asn._grad_receiver_groups[layer].ranks
            if asn.is_grad_worker(layer):
                assert src == rank
        gw = max(1, int(world * frac))
        assert asn.broadcast_gradients() == (gw < world)
        assert asn.broadcast_inverses() == (gw > 1)


def test_assignment_deterministic_across_ranks() -> None:
    work = {f'l{i}': {'A': float(i + 1), 'G': float(i + 1)} for i in range(7)}
    assignments = [
        KAISAAssignment(
            work,
            local_rank=r,
            world_size=4,
            grad_worker_fraction=0.5,
            group_func=_mock_group,
            colocate_factors=True,
        )._inv_assignments
        for r in range(4)
    ]
    assert all(a == assignments[0] for a in assignments)


def test_greedy_balance() -> None:
    # Equal work, world=4, one group: loads should be perfectly balanced.
    work = {f'l{i}': {'A': 1.0, 'G': 1.0} for i in range(8)}
    assignments = KAISAAssignment.greedy_assignment(
        work, [[0, 1, 2, 3]], 4, True,
    )
    loads = [0.0] * 4
    for layer, factors in assignments.items():
        for factor, rank in factors.items():
            loads[rank] += 1.0
    assert loads == [4.0, 4.0, 4.0, 4.0]


def test_greedy_no_colocate_splits_factors() -> None:
    work = {'l0': {'A': 5.0, 'G': 1.0}}
    assignments = KAISAAssignment.greedy_assignment(work, [[0, 1]], 2, False)
    assert assignments['l0']['A'] != assignments['l0']['G']


def test_validation() -> None:
    work = {'l0': {'A': 1.0, 'G': 1.0}}
    with pytest.raises(ValueError):
        KAISAAssignment(
            work, local_rank=0, world_size=4, grad_worker_fraction=2.0,
            group_func=_mock_group,
        )
    with pytest.raises(ValueError):
        KAISAAssignment(
            work, local_rank=5, world_size=4, grad_worker_fraction=1.0,
            group_func=_mock_group,
        )
    with pytest.raises(ValueError):
        # 4 * 0.3 = 1.2 not integer
        KAISAAssignment(
            work, local_rank=0, world_size=4, grad_worker_fraction=0.3,
            group_func=_mock_group,
        )


@pytest.mark.parametrize('world', [2, 4, 6, 8, 12, 16, 32, 64])
def test_assignment_sweep_all_fractions(world: int) -> None:
    """Broad invariant sweep: every divisible grad-worker count, mixed
    factor sizes, every rank's view must agree (reference
    assignment_test.py property style)."""
    work = {
        f'l{i}': {'A': float(((7 * i) % 13 + 1) ** 3), 'G': float((i % 5 + 1) ** 3)}
        for i in range(17)
    }
    for gw in [w for w in range(1, world + 1) if world % w == 0]:
        frac = gw / world
        views = [
            KAISAAssignment(
                work,
                local_rank=rank,
                world_size=world,
                grad_worker_fraction=frac,
                group_func=_mock_group,
                colocate_factors=False,
            )
            for rank in range(world)
        ]
        base = views[0]
        for layer in base.get_layers():
            for fac in ('A', 'G'):
                owners = {v.inv_worker(layer, fac) for v in views}
                # every rank agrees on the single inverse worker
                assert len(owners) == 1
            # exactly one source grad worker per receiver group, and it
            # is a grad worker for the layer
            for rank, v in enumerate(views):
                src = v.src_grad_worker(layer)
                assert views[src].is_grad_worker(layer)
        # greedy balance sanity: no rank has more than total work
        # (degenerate) and every rank's view of broadcast flags agrees
        assert len({v.broadcast_gradients() for v in views}) == 1
        assert len({v.broadcast_inverses() for v in views}) == 1
