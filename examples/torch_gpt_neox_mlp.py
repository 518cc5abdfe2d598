"""Tensor-parallel K-FAC example (GPT-NeoX-style sharded linears).

Minimal runnable wiring of ``GPTNeoXKFACPreconditioner`` without
DeepSpeed: a Megatron-style column+row parallel MLP block sharded over
all ranks (mp = world size), trained on synthetic data. Shows the three
things a real integration needs — the topology object, the process
groups, and modules named ``ColumnParallelLinear`` /
``RowParallelLinear`` so registration matches them by class name.

Launch (CPU/gloo, 2-way tensor parallel):

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 --standalone \
        examples/torch_gpt_neox_mlp.py --steps 20 --backend gloo

On MI355X boxes use ``--backend nccl`` (RCCL) with one rank per GPU.

The full 3D wiring (adding dp/pp groups) is the same pattern; see
tests/test_gpt_neox.py::test_3d_grid_training_matches_single_process
for a complete 2x2x2 grid, and docs/MIGRATION.md for the API map.
"""

from __future__ import annotations

import argparse
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


class ColumnParallelLinear(torch.nn.Linear):
    """Output-sharded linear: this rank holds rows [r*h/mp, (r+1)*h/mp)."""


class RowParallelLinear(torch.nn.Linear):
    """Input-sharded linear: this rank holds the matching columns."""


class _CopyToModelParallel(torch.autograd.Function):
    """Identity forward; allreduce backward (megatron f operator)."""

    @staticmethod
    def forward(ctx, x, group):  # type: ignore[override]
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        grad = grad.clone()
        dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromModelParallel(torch.autograd.Function):
    """Allreduce forward; identity backward (megatron g operator)."""

    @staticmethod
    def forward(ctx, x, group):  # type: ignore[override]
        x = x.clone()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        return grad, None


class ParallelMLPBlock(torch.nn.Module):
    """hidden -> 4*hidden/mp (per rank) -> hidden, megatron-sharded."""

    def __init__(self, hidden: int, mp_world: int, mp_group) -> None:
        super().__init__()
        assert (4 * hidden) % mp_world == 0
        shard = 4 * hidden // mp_world
        self.dense_h_to_4h = ColumnParallelLinear(hidden, shard, bias=True)
        self.dense_4h_to_h = RowParallelLinear(shard, hidden, bias=False)
        self.mp_group = mp_group

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = _CopyToModelParallel.apply(x, self.mp_group)
        h = torch.nn.functional.gelu(self.dense_h_to_4h(x))
        return _ReduceFromModelParallel.apply(
            self.dense_4h_to_h(h), self.mp_group,
        )


def main() -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument('--hidden', type=int, default=64)
    p.add_argument('--classes', type=int, default=8)
    p.add_argument('--batch-size', type=int, default=32)
    p.add_argument('--steps', type=int, default=50)
    p.add_argument('--lr', type=float, default=0.05)
    p.add_argument('--kfac-inv-update-steps', type=int, default=10)
    p.add_argument('--backend', default='nccl', choices=['nccl', 'gloo'])
    args = p.parse_args()

    from kfac_amd.gpt_neox import GPTNeoXKFACPreconditioner
    from kfac_amd.gpt_neox.topology import PipeModelDataTopology

    dist.init_process_group(args.backend)
    rank, world = dist.get_rank(), dist.get_world_size()
    device = torch.device('cpu')
    if args.backend == 'nccl':
        device = torch.device('cuda', int(os.environ.get('LOCAL_RANK', 0)))
        torch.cuda.set_device(device)

    # mp = world: one tensor-parallel group spanning all ranks; dp = 1:
    # every rank needs a (its own) dp group object — create all of them
    # on all ranks (new_group is collective over the world).
    topology = PipeModelDataTopology(num_pp=1, num_mp=world, num_dp=1)
    mp_group = dist.new_group(list(range(world)))
    dp_group = [dist.new_group([r]) for r in range(world)][rank]

    torch.manual_seed(0)  # same init on every rank, then shard
    full_in = torch.nn.Linear(args.hidden, 4 * args.hidden)
    model = ParallelMLPBlock(args.hidden, world, mp_group).to(device)
    shard = 4 * args.hidden // world
    with torch.no_grad():
        sl = slice(rank * shard, (rank + 1) * shard)
        model.dense_h_to_4h.weight.copy_(full_in.weight[sl].to(device))
        model.dense_h_to_4h.bias.copy_(full_in.bias[sl].to(device))
    head = torch.nn.Linear(args.hidden, args.classes).to(device)

    preconditioner = GPTNeoXKFACPreconditioner(
        model,
        topology=topology,
        data_parallel_group=dp_group,
        model_parallel_group=mp_group,
        factor_update_steps=1,
        inv_update_steps=args.kfac_inv_update_steps,
        lr=args.lr,
    )
    params = list(model.parameters()) + list(head.parameters())
    optimizer = torch.optim.SGD(params, lr=args.lr)

    gen = torch.Generator().manual_seed(7)
    for step in range(args.steps):
        x = torch.randn(
            args.batch_size, args.hidden, generator=gen,
        ).to(device)
        y = torch.randint(
            0, args.classes, (args.batch_size,), generator=gen,
        ).to(device)
        optimizer.zero_grad()
        loss = torch.nn.functional.cross_entropy(head(model(x)), y)
        loss.backward()
        preconditioner.step()
        optimizer.step()
        if rank == 0 and (step + 1) % 10 == 0:
            print(f'step {step + 1}: loss={loss.item():.4f}', flush=True)
    if rank == 0:
        print('done', flush=True)
    dist.destroy_process_group()


if __name__ == '__main__':
    main()
