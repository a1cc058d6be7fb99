#!/bin/bash
# Direct (no scheduler) launch on one MI355X node: one rank per GPU.
# Usage: scripts/run_node.sh <ngpus> [train.py args...]
set -e
N=${1:-8}; shift || true
export HSA_ENABLE_IPC_MODE_LEGACY=0
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29511 \
    train.py "$@"
