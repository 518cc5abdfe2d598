#!/bin/bash
# Single-node multi-GPU launcher for the example CLIs / bench
# (parity role of reference scripts/run_imagenet.sh, adapted to one
# MI355X node: one rank per GPU over RCCL via torchrun).
#
# Usage: scripts/launch_node.sh NPROC SCRIPT [ARGS...]
# e.g.:  scripts/launch_node.sh 8 examples/torch_imagenet_resnet.py --epochs 1

set -euo pipefail
NPROC=${1:?usage: launch_node.sh NPROC SCRIPT [ARGS...]}
shift
SCRIPT=${1:?usage: launch_node.sh NPROC SCRIPT [ARGS...]}
shift

export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}

exec python -m torch.distributed.run \
    --nnodes=1 \
    --nproc-per-node "$NPROC" \
    --master-addr 127.0.0.1 \
    --master-port "${MASTER_PORT:-29517}" \
    "$SCRIPT" "$@"
