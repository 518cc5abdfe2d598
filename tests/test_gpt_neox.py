"""Tests for the tensor/pipeline-parallel (GPT-NeoX) subsystem."""

from __future__ import annotations

import sys
import tempfile

import pytest
import torch
import torch.distributed as dist

sys.path.insert(0, '.')

from kfac_amd.gpt_neox.topology import PipeModelDataTopology  # noqa: E402
from kfac_amd.gpt_neox.assignment import GPTNeoXAssignment  # noqa: E402
from kfac_amd.gpt_neox.mpu import get_group_with_rank  # noqa: E402
from kfac_amd.gpt_neox.mpu import split_tensor_along_dim  # noqa: E402
from testing.distributed import run_distributed  # noqa: E402


@pytest.mark.parametrize('pp,mp,dp', [(1, 1, 4), (2, 1, 2), (1, 2, 2), (2, 2, 2)])
def test_topology_properties(pp: int, mp: int, dp: int) -> None:
    topo = PipeModelDataTopology(num_pp=pp, num_mp=mp, num_dp=dp)
    world = pp * mp * dp
    assert topo.world_size() == world
    # coords roundtrip and are unique
    coords = {topo.get_coord(r) for r in range(world)}
    assert len(coords) == world
    for r in range(world):
        c = topo.get_coord(r)
        assert topo.get_rank(c.pipe, c.data, c.model) == r
    # axis groups partition the world
    for axis, size in (('data', dp), ('model', mp), ('pipe', pp)):
        groups = topo.get_axis_comm_lists(axis)
        assert all(len(g) == size for g in groups)
        flat = sorted(r for g in groups for r in g)
        assert flat == list(range(world))


def test_split_tensor() -> None:
    t = torch.arange(12.0).reshape(2, 6)
    parts = split_tensor_along_dim(t, 3, dim=-1, contiguous_split_chunks=True)
    assert len(parts) == 3
    assert all(p.shape == (2, 2) for p in parts)
    torch.testing.assert_close(torch.cat(parts, dim=-1), t)
    with pytest.raises(ValueError):
        split_tensor_along_dim(t, 5, dim=-1)


def test_get_group_with_rank() -> None:
    groups = [[0, 1], [2, 3]]
    assert get_group_with_rank(2, groups) == [2, 3]
    with pytest.raises(ValueError):
        get_group_with_rank(9, groups)


def test_assignment_dp_only() -> None:
    """pp=1, mp=1: everything reduces to MEM-OPT data parallelism."""
    topo = PipeModelDataTopology(num_pp=1, num_mp=1, num_dp=4)
    work = {f'l{i}': {'A': float(i + 1), 'G': float(i + 1)} for i in range(6)}
    for rank in range(4):
        asn = GPTNeoXAssignment(
            work,
            local_rank=rank,
            topology=topo,
            data_parallel_group=None,
            model_parallel_group=None,
        )
        assert asn.broadcast_gradients()
        assert not asn.broadcast_inverses()
        for layer in asn.get_layers():
            inv = asn.inv_worker(layer, 'A')
            assert inv == asn.inv_worker(layer, 'G')
            assert 0 <= inv < 4
            # mp=1: the primary (factor gatherer) for this rank is itself;
            # src grad worker is the inv worker (it is a dp peer).
            assert asn.factor_worker(layer, 'A') == rank
            assert asn.src_grad_worker(layer) == inv
            assert asn.is_grad_worker(layer) == (rank == inv)


def test_assignment_balances_across_pipe_peers() -> None:
    topo = PipeModelDataTopology(num_pp=2, num_mp=1, num_dp=2)
    # stage-0 ranks: 0,1 ; stage-1 ranks: 2,3
    work = {f'l{i}': {'A': 1.0, 'G': 1.0} for i in range(4)}
    asn = GPTNeoXAssignment(
        work,
        local_rank=0,
        topology=topo,
        data_parallel_group=None,
        model_parallel_group=None,
    )
    workers = [asn.inv_worker(layer, 'A') for layer in asn.get_layers()]
    # only stage-0 peers get stage-0 layers, evenly
    assert sorted(workers) == [0, 0, 1, 1]


def _mlp_training(tmpdir: str) -> None:
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import ParallelMLP

    rank = dist.get_rank()
    world = dist.get_world_size()
    topo = PipeModelDataTopology(num_pp=1, num_mp=1, num_dp=world)
    dp_group = dist.new_group(list(range(world)))
    torch.manual_seed(0)
    model = ParallelMLP()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    precon = GPTNeoXKFACPreconditioner(
        model,
        topology=topo,
        data_parallel_group=dp_group,
        model_parallel_group=None,
        factor_update_steps=1,
        inv_update_steps=1,
        lr=0.05,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    x = torch.randn(16, 10)
    y = torch.randint(0, 4, (16,))
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        for p in model.parameters():
            dist.all_reduce(p.grad)
            p.grad /= world
        precon.step()
        opt.step()
        losses.append(loss.item())
    assert losses[0] > losses[-1], losses

    # sharded checkpoint: gather-based
    sd = precon.state_dict()
    assert set(sd['layers'].keys()) == {'dense_h_to_4h', 'dense_4h_to_h'}
    precon.load_state_dict(sd, compute_inverses=True)

    # factor-dir based
    precon.factor_checkpoint_dir = f'{tmpdir}/factors'
    sd2 = precon.state_dict()
    assert 'layers' not in sd2
    precon.load_factors_from_dir(compute_inverses=False)


def test_gpt_neox_dp_training() -> None:
    with tempfile.TemporaryDirectory() as td:
        run_distributed(2, _mlp_training, td)


def _tp_layer_protocol() -> None:
    """mp=2: gather -> precondition on primary -> scatter equals the
    single-process preconditioning of the full gradient."""
    from kfac_amd.distributed import TorchDistributedCommunicator
    from kfac_amd.gpt_neox.layer import GPTNeoXKFACEigenLayer
    from kfac_amd.gpt_neox.modules import GPTNeoXLinearModuleHelper
    from kfac_amd.ops import reference as ref
    from testing.gpt_neox import RowParallelLinear

    rank = dist.get_rank()
    mp_group = dist.new_group([0, 1])
    # dp size 1: every rank must call new_group for every rank set
    dp_groups = [dist.new_group([r]) for r in range(2)]
    dp_group = dp_groups[rank]
    torch.manual_seed(5)

    in_dim, out_dim, batch = 8, 6, 32
    # full data, same on both ranks
    x = torch.randn(batch, in_dim)
    full_grad = torch.randn(out_dim, in_dim)
    bias_grad = torch.randn(out_dim)

    shard = in_dim // 2
    module = RowParallelLinear(shard, out_dim, bias=True)
    module.weight.grad = full_grad[:, rank * shard : (rank + 1) * shard].clone()
    module.bias.grad = bias_grad.clone()

    tdc = TorchDistributedCommunicator()
    layer = GPTNeoXKFACEigenLayer(
        GPTNeoXLinearModuleHelper(module, mp_group, parallelism='input'),
        parallelism='input',
        model_parallel_group=mp_group,
        data_parallel_group=dp_group,
        pipe_parallel_peer_group=mp_group,
        primary_rank=0,
        tdc=tdc,
        prediv_eigenvalues=False,
    )

    # factor shapes account for the gathered input dim
    assert layer.module.a_factor_shape == (in_dim + 1, in_dim + 1)
    assert layer.module.g_factor_shape == (out_dim, out_dim)

    # save input: each rank contributes its shard
    layer.save_layer_input([x[:, rank * shard : (rank + 1) * shard]])
    g_out = torch.randn(batch, out_dim)
    layer.save_layer_grad_output((g_out,))
    layer.update_a_factor(0.95)
    layer.update_g_factor(0.95)
    layer.reduce_a_factor()
    layer.reduce_g_factor()
    tdc.flush_allreduce_buckets()  # all ranks: launch pending buckets
    if rank == 0:
        layer.compute_a_inv(damping=1e-3)
        layer.compute_g_inv(damping=1e-3)
    layer.preconditioned_grad(damping=1e-3)

    # single-process reference on the full gradient
    if rank == 0:
        grad = torch.cat([full_grad, bias_grad.view(-1, 1)], 1)
        expected = ref.precond_eigen(
            grad, layer.qa, layer.qg, da=layer.da, dg=layer.dg, damping=1e-3,
        )
    # each rank's shard of the result must match
    result = layer.grad
    assert result.shape == (out_dim, shard + 1)
    if rank == 0:
        shard_expected = torch.cat(
            [expected[:, :shard], expected[:, -1:]], 1,
        )
        torch.testing.assert_close(result, shard_expected, rtol=1e-4, atol=1e-5)
        # send expected to rank 1 for its shard check
        dist.broadcast(expected, src=0, group=mp_group)
    else:
        expected = torch.empty(out_dim, in_dim + 1)
        dist.broadcast(expected, src=0, group=mp_group)
        shard_expected = torch.cat(
            [expected[:, shard : 2 * shard], expected[:, -1:]], 1,
        )
        torch.testing.assert_close(result, shard_expected, rtol=1e-4, atol=1e-5)


def test_tp_layer_gather_precondition_scatter() -> None:
    run_distributed(2, _tp_layer_protocol)


def _tp_batches(steps: int = 8):
    g = torch.Generator().manual_seed(11)
    xs = [torch.randn(16, 10, generator=g) for _ in range(steps)]
    ys = [
        torch.randint(0, 4, (16,), generator=g) for _ in range(steps)
    ]
    return xs, ys


_TP_HP = dict(
    factor_update_steps=1,
    inv_update_steps=2,
    damping=1e-3,
    factor_decay=0.95,
    lr=0.05,
)


def _tp_mlp_training(tmpdir: str, kl_clip: float | None) -> None:
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import FullMLP
    from testing.gpt_neox import ShardedParallelMLP

    rank = dist.get_rank()
    topo = PipeModelDataTopology(num_pp=1, num_mp=2, num_dp=1)
    mp_group = dist.new_group([0, 1])
    dp_groups = [dist.new_group([r]) for r in range(2)]
    torch.manual_seed(21)
    full = FullMLP()
    model = ShardedParallelMLP(full, rank, 2, mp_group)
    precon = GPTNeoXKFACPreconditioner(
        model,
        topology=topo,
        data_parallel_group=dp_groups[rank],
        model_parallel_group=mp_group,
        kl_clip=kl_clip,
        **_TP_HP,
    )
    opt = torch.optim.SGD(model.parameters(), lr=_TP_HP['lr'])
    xs, ys = _tp_batches()
    losses = []
    for x, y in zip(xs, ys):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        losses.append(loss.item())
    # assemble the full weights from the shards for comparison
    w1s = [torch.empty_like(model.dense_h_to_4h.weight) for _ in range(2)]
    b1s = [torch.empty_like(model.dense_h_to_4h.bias) for _ in range(2)]
    w2s = [torch.empty_like(model.dense_4h_to_h.weight) for _ in range(2)]
    dist.all_gather(w1s, model.dense_h_to_4h.weight.data, group=mp_group)
    dist.all_gather(b1s, model.dense_h_to_4h.bias.data, group=mp_group)
    dist.all_gather(w2s, model.dense_4h_to_h.weight.data, group=mp_group)
    if rank == 0:
        torch.save(
            {
                'losses': losses,
                'w1': torch.cat(w1s, 0),
                'b1': torch.cat(b1s, 0),
                'w2': torch.cat(w2s, 1),
            },
            f'{tmpdir}/tp.pt',
        )


def _full_mlp_training(tmpdir: str, kl_clip: float | None) -> None:
    from kfac_amd import KFACPreconditioner
    from testing.gpt_neox import FullMLP

    torch.manual_seed(21)
    model = FullMLP()
    precon = KFACPreconditioner(model, kl_clip=kl_clip, **_TP_HP)
    opt = torch.optim.SGD(model.parameters(), lr=_TP_HP['lr'])
    xs, ys = _tp_batches()
    losses = []
    for x, y in zip(xs, ys):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        losses.append(loss.item())
    torch.save(
        {
            'losses': losses,
            'w1': model.dense_h_to_4h.weight.data,
            'b1': model.dense_h_to_4h.bias.data,
            'w2': model.dense_4h_to_h.weight.data,
        },
        f'{tmpdir}/full.pt',
    )


@pytest.mark.parametrize('kl_clip', [None, 0.001])
def test_tp_training_matches_single_process(kl_clip: float | None) -> None:
    """mp=2 K-FAC training == single-process K-FAC on the full model.

    End-to-end through GPTNeoXKFACPreconditioner: class-name
    registration, mp-aware factor shapes, gather->precondition->scatter
    every inverse phase, over 8 optimizer steps. The sharded run's loss
    trajectory and assembled weights must match a plain
    KFACPreconditioner run on the unsharded twin. With kl_clip set, the
    TP-consistent grad scale (allreduced over the mp group with the
    replicated bias column down-weighted) must reproduce the full-model
    scale exactly — the shard-local scale the reference computes does
    not."""
    with tempfile.TemporaryDirectory() as td:
        run_distributed(2, _tp_mlp_training, td, kl_clip)
        run_distributed(1, _full_mlp_training, td, kl_clip)
        tp = torch.load(f'{td}/tp.pt')
        full = torch.load(f'{td}/full.pt')
        torch.testing.assert_close(
            torch.tensor(tp['losses']),
            torch.tensor(full['losses']),
            rtol=1e-4,
            atol=1e-5,
        )
        for k in ('w1', 'b1', 'w2'):
            torch.testing.assert_close(
                tp[k], full[k], rtol=1e-3, atol=1e-5,
            )


def _tp_dp_mlp_training(tmpdir: str, kl_clip: float | None) -> None:
    """world=4 as a 2x2 grid: mp groups [0,1],[2,3]; dp groups [0,2],[1,3]."""
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import FullMLP
    from testing.gpt_neox import ShardedParallelMLP

    rank = dist.get_rank()
    mp_rank, dp_rank = rank % 2, rank // 2
    topo = PipeModelDataTopology(num_pp=1, num_mp=2, num_dp=2)
    mp_pg = [dist.new_group([0, 1]), dist.new_group([2, 3])]
    dp_pg = [dist.new_group([0, 2]), dist.new_group([1, 3])]
    mp_group = mp_pg[dp_rank]
    dp_group = dp_pg[mp_rank]
    torch.manual_seed(21)
    full = FullMLP()
    model = ShardedParallelMLP(full, mp_rank, 2, mp_group)
    precon = GPTNeoXKFACPreconditioner(
        model,
        topology=topo,
        data_parallel_group=dp_group,
        model_parallel_group=mp_group,
        kl_clip=kl_clip,
        **_TP_HP,
    )
    opt = torch.optim.SGD(model.parameters(), lr=_TP_HP['lr'])
    xs, ys = _tp_batches()
    losses = []
    half = 8
    for x, y in zip(xs, ys):
        xh = x[dp_rank * half : (dp_rank + 1) * half]
        yh = y[dp_rank * half : (dp_rank + 1) * half]
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(xh), yh)
        loss.backward()
        # DDP-equivalent: average gradients over the dp replicas (K-FAC
        # assumes pre-averaged grads)
        for p in model.parameters():
            dist.all_reduce(p.grad, group=dp_group)
            p.grad /= 2
        precon.step()
        opt.step()
        # full-batch loss = mean of the two half-batch losses
        lt = loss.detach().clone()
        dist.all_reduce(lt, group=dp_group)
        losses.append(lt.item() / 2)
    # assemble full weights from the mp shards (collective within mp group)
    w1s = [torch.empty_like(model.dense_h_to_4h.weight) for _ in range(2)]
    b1s = [torch.empty_like(model.dense_h_to_4h.bias) for _ in range(2)]
    w2s = [torch.empty_like(model.dense_4h_to_h.weight) for _ in range(2)]
    dist.all_gather(w1s, model.dense_h_to_4h.weight.data, group=mp_group)
    dist.all_gather(b1s, model.dense_h_to_4h.bias.data, group=mp_group)
    dist.all_gather(w2s, model.dense_4h_to_h.weight.data, group=mp_group)
    w1, b1, w2 = torch.cat(w1s, 0), torch.cat(b1s, 0), torch.cat(w2s, 1)
    # dp replicas must agree exactly on the assembled model
    for t in (w1, b1, w2):
        gathered = [torch.empty_like(t) for _ in range(4)]
        dist.all_gather(gathered, t)
        if rank == 0:
            for g in gathered[1:]:
                torch.testing.assert_close(g, gathered[0], rtol=1e-5, atol=1e-6)
    if rank == 0:
        torch.save(
            {'losses': losses, 'w1': w1, 'b1': b1, 'w2': w2},
            f'{tmpdir}/tp.pt',
        )


@pytest.mark.parametrize('kl_clip', [None, 0.001])
def test_tp_dp_training_matches_single_process(kl_clip: float | None) -> None:
    """2x2 (mp=2, dp=2) K-FAC == single-process K-FAC on the full model.

    Half-batches per dp replica with DDP-style grad averaging; factors
    dp-averaged by the layer's reduce routing; preconditioned grads
    broadcast over dp (MEM-OPT); gather->precondition->scatter within
    each mp group. Losses and assembled weights must match the world-1
    run, and the two dp replicas must stay bitwise-consistent."""
    with tempfile.TemporaryDirectory() as td:
        run_distributed(4, _tp_dp_mlp_training, td, kl_clip)
        run_distributed(1, _full_mlp_training, td, kl_clip)
        tp = torch.load(f'{td}/tp.pt')
        full = torch.load(f'{td}/full.pt')
        torch.testing.assert_close(
            torch.tensor(tp['losses']),
            torch.tensor(full['losses']),
            rtol=1e-4,
            atol=1e-5,
        )
        for k in ('w1', 'b1', 'w2'):
            torch.testing.assert_close(
                tp[k], full[k], rtol=1e-3, atol=1e-5,
            )


def _pp_stage_training(tmpdir: str, kl_clip: float | None) -> None:
    """pp=2, mp=1, dp=1: each rank owns ONE layer of a replicated model.

    Pipeline mechanics are simulated: both ranks run the full
    forward/backward locally (identical data and params), each stage's
    preconditioner is registered over only its own layer, and after the
    optimizer step each layer's parameters are broadcast from the stage
    that owns them — exactly the state a real pipeline would hold."""
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import ParallelMLP

    rank = dist.get_rank()
    topo = PipeModelDataTopology(num_pp=2, num_mp=1, num_dp=1)
    dp_groups = [dist.new_group([r]) for r in range(2)]
    pp_group = dist.new_group([0, 1])
    torch.manual_seed(21)
    model = ParallelMLP()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    stage_name = 'dense_h_to_4h' if rank == 0 else 'dense_4h_to_h'
    stage = torch.nn.ModuleDict({stage_name: getattr(model, stage_name)})
    precon = GPTNeoXKFACPreconditioner(
        stage,
        topology=topo,
        data_parallel_group=dp_groups[rank],
        model_parallel_group=None,
        pipeline_parallel_group=pp_group,
        kl_clip=kl_clip,  # with the pipe group provided, the clip is
        # the TRUE full-model scale (stage sums allreduced over it)
        **_TP_HP,
    )
    # work balanced over this stage's single peer: itself
    assert precon._assignment.pipe_parallel_peers == [rank]
    opt = torch.optim.SGD(
        getattr(model, stage_name).parameters(), lr=_TP_HP['lr'],
    )
    xs, ys = _tp_batches()
    losses = []
    for x, y in zip(xs, ys):
        for p in model.parameters():
            p.grad = None
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        # pipeline state exchange: each layer's params live on its stage
        for p in model.dense_h_to_4h.parameters():
            dist.broadcast(p.data, src=0)
        for p in model.dense_4h_to_h.parameters():
            dist.broadcast(p.data, src=1)
        losses.append(loss.item())
    if rank == 0:
        torch.save(
            {
                'losses': losses,
                'w1': model.dense_h_to_4h.weight.data,
                'b1': model.dense_h_to_4h.bias.data,
                'w2': model.dense_4h_to_h.weight.data,
                'b2': model.dense_4h_to_h.bias.data,
            },
            f'{tmpdir}/pp.pt',
        )


def _serial_parallel_mlp_training(tmpdir: str, kl_clip: float | None) -> None:
    from kfac_amd import KFACPreconditioner
    from testing.gpt_neox import ParallelMLP

    torch.manual_seed(21)
    model = ParallelMLP()
    precon = KFACPreconditioner(model, kl_clip=kl_clip, **_TP_HP)
    opt = torch.optim.SGD(model.parameters(), lr=_TP_HP['lr'])
    xs, ys = _tp_batches()
    losses = []
    for x, y in zip(xs, ys):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        losses.append(loss.item())
    torch.save(
        {
            'losses': losses,
            'w1': model.dense_h_to_4h.weight.data,
            'b1': model.dense_h_to_4h.bias.data,
            'w2': model.dense_4h_to_h.weight.data,
            'b2': model.dense_4h_to_h.bias.data,
        },
        f'{tmpdir}/serial.pt',
    )


@pytest.mark.parametrize('kl_clip', [None, 0.001])
def test_pp_training_matches_single_process(kl_clip: float | None) -> None:
    """pp=2 per-stage K-FAC == single-process K-FAC on the same model.

    Regression for the pipe-peer group selection: with mp=1 (group None)
    and pp=2, the peer group must be the rank's own singleton — reusing
    group=None (the GLOBAL group) would cross stages owning different
    layers and hang or corrupt the factor allreduce. With kl_clip set,
    the stage sums must combine over the provided pipeline group into
    the exact full-model clip scale."""
    with tempfile.TemporaryDirectory() as td:
        run_distributed(2, _pp_stage_training, td, kl_clip)
        run_distributed(1, _serial_parallel_mlp_training, td, kl_clip)
        pp = torch.load(f'{td}/pp.pt')
        serial = torch.load(f'{td}/serial.pt')
        torch.testing.assert_close(
            torch.tensor(pp['losses']),
            torch.tensor(serial['losses']),
            rtol=1e-4,
            atol=1e-5,
        )
        for k in ('w1', 'b1', 'w2', 'b2'):
            torch.testing.assert_close(
                pp[k], serial[k], rtol=1e-3, atol=1e-5,
            )


def _pp_tp_dp_training(tmpdir: str, kl_clip: float | None) -> None:
    """world=8 as pp=2 x mp=2 x dp=2 — the full 3D grid.

    rank = (pipe*2 + data)*2 + model. Each rank holds shard `model` of
    BOTH layers (forward needs the whole model) but registers K-FAC only
    over its own stage's layer; pipeline state exchange is simulated by
    broadcasting each layer's shards from the stage that owns them.
    This is the only configuration where pipe peers != mp peers != dp
    peers, so the assignment must build per-stage peer groups itself."""
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import FullMLP
    from testing.gpt_neox import ShardedParallelMLP

    rank = dist.get_rank()
    topo = PipeModelDataTopology(num_pp=2, num_mp=2, num_dp=2)
    c = topo.get_coord(rank)
    # every rank creates every group, same order (collective contract)
    mp_group = dp_group = ex_group = None
    for p in range(2):
        for d in range(2):
            g = dist.new_group([topo.get_rank(p, d, 0), topo.get_rank(p, d, 1)])
            if (p, d) == (c.pipe, c.data):
                mp_group = g
    for p in range(2):
        for m in range(2):
            g = dist.new_group([topo.get_rank(p, 0, m), topo.get_rank(p, 1, m)])
            if (p, m) == (c.pipe, c.model):
                dp_group = g
    for d in range(2):
        for m in range(2):
            g = dist.new_group([topo.get_rank(0, d, m), topo.get_rank(1, d, m)])
            if (d, m) == (c.data, c.model):
                ex_group = g
    assert None not in (mp_group, dp_group, ex_group)

    torch.manual_seed(21)
    full = FullMLP()
    model = ShardedParallelMLP(full, c.model, 2, mp_group)
    stage_name = 'dense_h_to_4h' if c.pipe == 0 else 'dense_4h_to_h'
    stage = torch.nn.ModuleDict({stage_name: getattr(model, stage_name)})
    precon = GPTNeoXKFACPreconditioner(
        stage,
        topology=topo,
        data_parallel_group=dp_group,
        model_parallel_group=mp_group,
        pipeline_parallel_group=ex_group,
        kl_clip=kl_clip,
        **_TP_HP,
    )
    assert sorted(precon._assignment.pipe_parallel_peers) == sorted(
        topo.get_rank(c.pipe, d, m) for d in range(2) for m in range(2)
    )
    opt = torch.optim.SGD(
        getattr(model, stage_name).parameters(), lr=_TP_HP['lr'],
    )
    xs, ys = _tp_batches()
    losses = []
    half = 8
    stage_srcs = {
        'dense_h_to_4h': topo.get_rank(0, c.data, c.model),
        'dense_4h_to_h': topo.get_rank(1, c.data, c.model),
    }
    for x, y in zip(xs, ys):
        xh = x[c.data * half : (c.data + 1) * half]
        yh = y[c.data * half : (c.data + 1) * half]
        for p in model.parameters():
            p.grad = None
        loss = torch.nn.functional.cross_entropy(model(xh), yh)
        loss.backward()
        for p in model.parameters():
            dist.all_reduce(p.grad, group=dp_group)
            p.grad /= 2
        precon.step()
        opt.step()
        # pipeline state exchange within each (data, model) column
        for lname, src in stage_srcs.items():
            for p in getattr(model, lname).parameters():
                dist.broadcast(p.data, src=src, group=ex_group)
        lt = loss.detach().clone()
        dist.all_reduce(lt, group=dp_group)
        losses.append(lt.item() / 2)
    # assemble full weights from this rank's mp shards, then require all
    # 8 ranks to agree bitwise
    w1s = [torch.empty_like(model.dense_h_to_4h.weight) for _ in range(2)]
    b1s = [torch.empty_like(model.dense_h_to_4h.bias) for _ in range(2)]
    w2s = [torch.empty_like(model.dense_4h_to_h.weight) for _ in range(2)]
    dist.all_gather(w1s, model.dense_h_to_4h.weight.data, group=mp_group)
    dist.all_gather(b1s, model.dense_h_to_4h.bias.data, group=mp_group)
    dist.all_gather(w2s, model.dense_4h_to_h.weight.data, group=mp_group)
    w1, b1, w2 = torch.cat(w1s, 0), torch.cat(b1s, 0), torch.cat(w2s, 1)
    for t in (w1, b1, w2):
        gathered = [torch.empty_like(t) for _ in range(8)]
        dist.all_gather(gathered, t)
        if rank == 0:
            for g in gathered[1:]:
                torch.testing.assert_close(g, gathered[0], rtol=1e-5, atol=1e-6)
    if rank == 0:
        torch.save(
            {'losses': losses, 'w1': w1, 'b1': b1, 'w2': w2},
            f'{tmpdir}/grid.pt',
        )


@pytest.mark.parametrize('kl_clip', [None, 0.001])
def test_3d_grid_training_matches_single_process(kl_clip: float | None) -> None:
    """pp=2 x mp=2 x dp=2 (world 8) == single-process K-FAC.

    The full 3D protocol at once: per-stage work assignment over 4 pipe
    peers (built via the fallback group path), TP gather/precondition/
    scatter within mp pairs, dp-averaged factors and MEM-OPT grad
    broadcasts, pipeline state exchange across stages — with kl-clip,
    the mp-allreduced stage sums combine over the pipe axis into the
    exact single-process clip scale."""
    with tempfile.TemporaryDirectory() as td:
        run_distributed(8, _pp_tp_dp_training, td, kl_clip)
        run_distributed(1, _full_mlp_training, td, kl_clip)
        grid = torch.load(f'{td}/grid.pt')
        full = torch.load(f'{td}/full.pt')
        torch.testing.assert_close(
            torch.tensor(grid['losses']),
            torch.tensor(full['losses']),
            rtol=1e-4,
            atol=1e-5,
        )
        for k in ('w1', 'b1', 'w2'):
            torch.testing.assert_close(
                grid[k], full[k], rtol=1e-3, atol=1e-5,
            )


def _tp_checkpoint_resume(factor_dir: str | None = None) -> None:
    """mp=2: checkpoint at step 6, rebuild, resume — trajectory must
    match an uninterrupted 10-step run exactly. With ``factor_dir``,
    factors round-trip through per-layer files written by the inverse
    workers instead of the gathered state dict."""
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import FullMLP
    from testing.gpt_neox import ShardedParallelMLP

    rank = dist.get_rank()
    topo = PipeModelDataTopology(num_pp=1, num_mp=2, num_dp=1)
    mp_group = dist.new_group([0, 1])
    dp_groups = [dist.new_group([r]) for r in range(2)]
    xs, ys = _tp_batches(10)

    def build():
        torch.manual_seed(21)
        model = ShardedParallelMLP(FullMLP(), rank, 2, mp_group)
        precon = GPTNeoXKFACPreconditioner(
            model,
            topology=topo,
            data_parallel_group=dp_groups[rank],
            model_parallel_group=mp_group,
            kl_clip=0.001,
            **_TP_HP,
        )
        return model, precon

    def steps(model, precon, opt, lo, hi):
        out = []
        for x, y in zip(xs[lo:hi], ys[lo:hi]):
            opt.zero_grad()
            loss = torch.nn.functional.cross_entropy(model(x), y)
            loss.backward()
            precon.step()
            opt.step()
            out.append(loss.item())
        return out

    # uninterrupted run
    model, precon = build()
    opt = torch.optim.SGD(model.parameters(), lr=_TP_HP['lr'])
    full_losses = steps(model, precon, opt, 0, 10)

    # interrupted run: 6 steps, checkpoint, rebuild, resume 4 more
    model, precon = build()
    opt = torch.optim.SGD(model.parameters(), lr=_TP_HP['lr'])
    losses = steps(model, precon, opt, 0, 6)
    if factor_dir is not None:
        precon.factor_checkpoint_dir = factor_dir
    sd = precon.state_dict()
    msd = {k: v.clone() for k, v in model.state_dict().items()}

    model2, precon2 = build()
    model2.load_state_dict(msd)
    if factor_dir is not None:
        precon2.factor_checkpoint_dir = factor_dir
    precon2.load_state_dict(sd, compute_inverses=True)
    assert precon2.steps == precon.steps
    opt2 = torch.optim.SGD(model2.parameters(), lr=_TP_HP['lr'])
    losses += steps(model2, precon2, opt2, 6, 10)

    torch.testing.assert_close(
        torch.tensor(losses), torch.tensor(full_losses), rtol=1e-4, atol=1e-6,
    )


@pytest.mark.parametrize('mode', ['gather', 'dir'])
def test_tp_checkpoint_resume_matches_uninterrupted(mode: str) -> None:
    """Sharded checkpointing round-trips the mp=2 state exactly, via
    the gathered state dict or per-layer factor files. Non-worker pipe
    peers restore their unsharded-dim factor copies too — without that
    the first post-resume factor reduce averages in a fresh identity
    factor and the trajectory drifts (reference-inherited wart)."""
    if mode == 'gather':
        run_distributed(2, _tp_checkpoint_resume)
    else:
        with tempfile.TemporaryDirectory() as td:
            run_distributed(2, _tp_checkpoint_resume, f'{td}/factors')


def test_reshape_data_semantics() -> None:
    from kfac_amd.layers.utils import reshape_data

    a = torch.arange(12.0).reshape(2, 3, 2)
    b = torch.arange(6.0).reshape(1, 3, 2)
    out = reshape_data([a, b])
    assert out.shape == (3, 3, 2)
    torch.testing.assert_close(out[:2], a)
    flat = reshape_data([a, b], collapse_dims=True)
    assert flat.shape == (9, 2)
    sf = reshape_data(
        [a.transpose(0, 1), b.transpose(0, 1)], batch_first=False,
    )
    assert sf.shape == (3, 3, 2)


def _gather_mp_region() -> None:
    from kfac_amd.gpt_neox.mpu import gather_from_model_parallel_region

    rank = dist.get_rank()
    group = dist.new_group([0, 1])
    shard = torch.full((2, 3), float(rank))
    out = gather_from_model_parallel_region(shard, dst=0, model_parallel_group=group)
    if rank == 0:
        assert out.shape == (2, 6)
        torch.testing.assert_close(out[:, :3], torch.zeros(2, 3))
        torch.testing.assert_close(out[:, 3:], torch.ones(2, 3))
    else:
        assert out is None
    # bf16 + fp32_allreduce roundtrip keeps the bf16 dtype
    shard16 = torch.full((2, 2), float(rank + 1), dtype=torch.bfloat16)
    out16 = gather_from_model_parallel_region(
        shard16, dst=0, model_parallel_group=group, fp32_allreduce=True,
    )
    if rank == 0:
        assert out16.dtype == torch.bfloat16
        assert out16.shape == (2, 4)
    # size-1 group short-circuits to the input
    same = gather_from_model_parallel_region(
        shard, dst=rank, model_parallel_group=None,
    )
    assert same is shard


def test_gather_from_model_parallel_region() -> None:
    run_distributed(2, _gather_mp_region)
