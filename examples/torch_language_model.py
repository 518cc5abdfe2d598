"""Transformer language model training with distributed K-FAC.

Feature parity with reference examples/torch_language_model.py:297:
encoder-only causal LM; K-FAC is applied to the MLP linears only
(embedding/decoder/attention skip-listed by default, reference :162-167).
Offline image -> synthetic token streams by default.
"""

from __future__ import annotations

import argparse
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import kfac_amd  # noqa: E402
from examples.language.dataset import synthetic_batch  # noqa: E402
from examples.language.engine import run_training  # noqa: E402
from kfac_amd.models import TransformerModel  # noqa: E402


def parse_args() -> argparse.Namespace:
    p = argparse.ArgumentParser(
        description='Transformer LM + K-FAC',
        formatter_class=argparse.ArgumentDefaultsHelpFormatter,
    )
    # flag names and defaults match the reference CLI
    # (torch_language_model.py:35-160); --vocab replaces its dataset
    # download flags (offline synthetic token streams)
    p.add_argument('--seq-len', type=int, default=64)
    p.add_argument('--batch-size', type=int, default=20)
    p.add_argument('--vocab', type=int, default=8192)
    p.add_argument('--embedding-dim', '--emsize', dest='emsize',
                   type=int, default=256)
    p.add_argument('--attention-heads', '--nhead', dest='nhead',
                   type=int, default=4)
    p.add_argument('--hidden-dim', '--nhid', dest='nhid',
                   type=int, default=256)
    p.add_argument('--layers', '--nlayers', dest='nlayers',
                   type=int, default=2)
    p.add_argument('--dropout', type=float, default=0.2)
    p.add_argument('--epochs', type=int, default=20)
    p.add_argument('--steps-per-epoch', type=int, default=200)
    p.add_argument('--lr', type=float, default=1.0)
    p.add_argument('--backend', type=str, default=None, choices=['nccl', 'gloo'])
    p.add_argument('--seed', type=int, default=42)
    p.add_argument('--no-cuda', action='store_true', default=False)
    p.add_argument('--kfac-inv-update-steps', type=int, default=10)
    p.add_argument('--kfac-factor-update-steps', type=int, default=1)
    p.add_argument('--kfac-factor-decay', type=float, default=0.95)
    p.add_argument('--kfac-damping', type=float, default=0.003)
    p.add_argument('--kfac-kl-clip', type=float, default=0.001)
    p.add_argument(
        '--kfac-skip-layers',
        nargs='+',
        type=str,
        default=['embedding', 'decoder', '.*self_attn.*'],
        help='K-FAC on MLP linears only (reference default; note '
        'self_attn.out_proj must be skipped because '
        'F.multi_head_attention_forward bypasses its module forward)',
    )
    return p.parse_args()


def main() -> None:
    args = parse_args()
    world = int(os.environ.get('WORLD_SIZE', '1'))
    local_rank = int(os.environ.get('LOCAL_RANK', '0'))
    use_cuda = torch.cuda.is_available() and not args.no_cuda
    if world > 1:
        dist.init_process_group(args.backend or ('nccl' if use_cuda else 'gloo'))
    if use_cuda:
        # modulo so N-rank smokes run on fewer GPUs (e.g. 2 ranks, 1 GPU)
        local_rank = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(local_rank)
    device = torch.device('cuda', local_rank) if use_cuda else torch.device('cpu')
    torch.manual_seed(args.seed)

    model = TransformerModel(
        args.vocab, args.emsize, args.nhead, args.nhid, args.nlayers,
        args.dropout,
    ).to(device)
    if world > 1:
        model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None,
        )

    optimizer = torch.optim.SGD(model.parameters(), lr=args.lr)
    preconditioner = None
    if args.kfac_inv_update_steps > 0:
        preconditioner = kfac_amd.KFACPreconditioner(
            model,
            factor_update_steps=args.kfac_factor_update_steps,
            inv_update_steps=args.kfac_inv_update_steps,
            factor_decay=args.kfac_factor_decay,
            damping=args.kfac_damping,
            kl_clip=args.kfac_kl_clip,
            lr=lambda x: optimizer.param_groups[0]['lr'],
            skip_layers=args.kfac_skip_layers,
        )
    rank = dist.get_rank() if world > 1 else 0

    def batch_fn(i: int) -> tuple[torch.Tensor, torch.Tensor]:
        return synthetic_batch(
            args.vocab, args.batch_size, args.seq_len, device,
            seed=i * world + rank,
        )

    run_training(
        model,
        optimizer,
        preconditioner,
        batch_fn,
        args.epochs,
        args.steps_per_epoch,
        args.vocab,
        rank=rank,
    )

    if world > 1:
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
