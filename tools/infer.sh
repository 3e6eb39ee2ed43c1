#!/usr/bin/env bash
# Inference launcher (reference: tools/infer.sh): one rank per GPU over
# RCCL for TP/PP-sharded generation scripts.
#   bash tools/infer.sh <script.py> <GPUS> [script args...]
# Multi-node: set NNODES / NODE_RANK / MASTER_ADDR / MASTER_PORT everywhere.
FILE=$1
GPUS=${2:-1}
shift 2 || shift 1
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
export TORCH_NCCL_ASYNC_ERROR_HANDLING=${TORCH_NCCL_ASYNC_ERROR_HANDLING:-1}
NNODES=${NNODES:-1}
NODE_RANK=${NODE_RANK:-0}
MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
MASTER_PORT=${MASTER_PORT:-29500}
if [ "$GPUS" -gt 1 ] || [ "$NNODES" -gt 1 ]; then
  if [ "$NNODES" -gt 1 ]; then
    python -m torch.distributed.run --nnodes "$NNODES" --node-rank "$NODE_RANK" \
      --master-addr "$MASTER_ADDR" --master-port "$MASTER_PORT" \
      --nproc-per-node "$GPUS" "$FILE" "$@"
  else
    python -m torch.distributed.run --standalone --nnodes=1 \
      --nproc-per-node "$GPUS" --local-addr 127.0.0.1 "$FILE" "$@"
  fi
else
  python "$FILE" "$@"
fi
