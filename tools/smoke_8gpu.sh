#!/usr/bin/env bash
# 8-GPU RCCL smoke: first-contact script for a multi-GPU node (VERDICT r1
# item 5). Runs a short all-to-all-heavy query subset at SF=10 with one
# rank per GPU over RCCL/xGMI. Safe to run standalone on any 2/4/8-GPU box.
set -euo pipefail
N=${1:-8}
SF=${2:-10}
export HSA_ENABLE_IPC_MODE_LEGACY=0   # host driver supports dmabuf IPC only
export NCCL_IB_DISABLE=1              # single node: xGMI only, no fabric
export TORCH_NCCL_BLOCKING_WAIT=1     # surface hangs as errors, not stalls
cd "$(dirname "$0")/.."
exec python -m torch.distributed.run --standalone --nnodes=1 \
  --nproc-per-node "$N" --local-addr 127.0.0.1 \
  bench.py --gpus "$N" --steps 2 --warmup 1 --sf "$SF" \
  --queries q3,q23,q72,q59,q95,q14
