"""RCCL contact tests — first actual NCCL-backend execution on MI355X.

Round-1 verdict: every multi-rank run to date used gloo; the nccl(=RCCL)
backend had never been initialized anywhere. These tests initialize RCCL
for real on the GPU box (world 1 — a valid RCCL communicator with real
ncclAllReduce/ncclBroadcast launches) and drive the exact communicator
code paths the 8-GPU driver run will exercise: plain/bucketed allreduce,
broadcast on a cached sub-group, Work.wait() stream semantics, and the
preconditioner's hook-launched side-stream reduces.

A world-2-on-one-GPU RCCL attempt is exercised separately in a
subprocess (scripts/rccl_world2_smoke.py): NCCL historically refuses two
ranks on one device, so that smoke records the outcome without gating.
"""

from __future__ import annotations

import os
import sys

import pytest
import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def rccl_world1():
    """Initialize a real single-rank RCCL process group."""
    if not torch.cuda.is_available():
        pytest.skip('requires a GPU')
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', '29611')
    os.environ['RANK'] = '0'
    os.environ['WORLD_SIZE'] = '1'
    torch.cuda.set_device(0)
    dist.init_process_group('nccl', rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def test_rccl_init_allreduce_broadcast(rccl_world1) -> None:
    """ncclAllReduce / ncclBroadcast launch and complete over RCCL."""
    x = torch.randn(1024, device='cuda')
    ref = x.clone()
    work = dist.all_reduce(x, async_op=True)
    work.wait()
    torch.cuda.synchronize()
    torch.testing.assert_close(x, ref)

    y = torch.randn(257, device='cuda')
    ref = y.clone()
    work = dist.broadcast(y, src=0, async_op=True)
    work.wait()
    torch.cuda.synchronize()
    torch.testing.assert_close(y, ref)


def test_rccl_subgroup_collective(rccl_world1) -> None:
    """Collectives on a cached sub-communicator (assignment group path)."""
    group = dist.new_group([0])
    x = torch.randn(64, 64, device='cuda')
    ref = x.clone()
    work = dist.all_reduce(x, group=group, async_op=True)
    work.wait()
    torch.cuda.synchronize()
    torch.testing.assert_close(x, ref)


def test_rccl_communicator_paths(rccl_world1) -> None:
    """TorchDistributedCommunicator over an initialized RCCL backend.

    World size is 1 so the communicator's early-return path applies for
    its own world checks; drive the bucket machinery directly so the
    flat-buffer allreduce + unpack runs through real NCCL work objects.
    """
    from kfac_amd.distributed import AllreduceTensorBucket

    bucket = AllreduceTensorBucket(cap_bytes=1 << 20)
    a = torch.full((128,), 2.0, device='cuda')
    b = torch.full((64,), 3.0, device='cuda')
    bucket.append(a)
    bucket.append(b)
    bucket.communicate(group=None, scale=0.5)
    bucket.wait_and_unpack()
    torch.cuda.synchronize()
    torch.testing.assert_close(a, torch.full((128,), 1.0, device='cuda'))
    torch.testing.assert_close(b, torch.full((64,), 1.5, device='cuda'))


def test_rccl_sidestream_collective_ordering(rccl_world1) -> None:
    """A collective launched from the cov side stream must order after
    kernels queued there (the hook-launched factor reduce pattern)."""
    from kfac_amd.streams import cov_stream, join_cov_stream

    device = torch.device('cuda', 0)
    x = torch.zeros(1 << 22, device=device)
    s = cov_stream(device)
    s.wait_stream(torch.cuda.current_stream(device))
    with torch.cuda.stream(s):
        for _ in range(8):
            x.add_(0.125)  # queued work the collective must wait for
        work = dist.all_reduce(x, async_op=True)
    join_cov_stream(device)
    work.wait()
    torch.cuda.synchronize()
    torch.testing.assert_close(
        x, torch.ones(1 << 22, device=device),
    )


def test_rccl_full_kfac_step(rccl_world1) -> None:
    """One full K-FAC train step with torch.distributed initialized on
    the nccl backend (world 1): hooks, side-stream covariance, bucketed
    reduce short-circuit, precondition, grad write."""
    from kfac_amd import KFACPreconditioner

    torch.manual_seed(7)
    model = torch.nn.Sequential(
        torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 10),
    ).cuda()
    precon = KFACPreconditioner(
        model, factor_update_steps=1, inv_update_steps=1, lr=0.1,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    # fixed batch: the model must overfit it, so the loss decrease is
    # deterministic (fresh batches each step made this assertion flaky)
    x = torch.randn(16, 32, device='cuda')
    y = torch.randint(0, 10, (16,), device='cuda')
    losses = []
    for _ in range(20):
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        losses.append(loss.item())
    # kl-clip bounds the per-step movement, so expect steady (not
    # dramatic) descent on the fixed batch: >= 10% in 20 steps.
    assert losses[-1] < 0.9 * losses[0], losses


def test_gpt_neox_training_over_rccl(rccl_world1) -> None:
    """GPT-NeoX preconditioner end-to-end on cuda over a real RCCL
    group (world 1): gather/precondition/scatter branches, dp-group
    factor reduce, and the gloo-subgroup checkpoint gather."""
    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology
    from testing.gpt_neox import ParallelMLP

    torch.manual_seed(0)
    model = ParallelMLP(in_dim=32, hidden=64, out_dim=8).cuda()
    precon = GPTNeoXKFACPreconditioner(
        model,
        topology=PipeModelDataTopology(num_pp=1, num_mp=1, num_dp=1),
        data_parallel_group=dist.new_group([0]),
        model_parallel_group=None,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.05,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    x = torch.randn(64, 32, device='cuda')
    y = torch.randint(0, 8, (64,), device='cuda')
    losses = []
    for _ in range(8):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        losses.append(loss.item())
    assert losses[0] > losses[-1], losses
    sd = precon.state_dict()
    assert set(sd['layers'].keys()) == {'dense_h_to_4h', 'dense_4h_to_h'}
    precon.load_state_dict(sd, compute_inverses=True)
    torch.cuda.synchronize()
