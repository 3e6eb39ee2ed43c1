#!/usr/bin/env bash
# Launcher wrapper (reference: tools/train.sh): one rank per GPU over RCCL.
#   bash tools/train.sh tools/train_net.py configs/gpt2_pretrain.py 8 [opts...]
# Multi-node (RCCL over IB): set NNODES / NODE_RANK / MASTER_ADDR / MASTER_PORT
# on every node and run the same command.
FILE=$1
CONFIG=$2
GPUS=${3:-1}
shift 3 || shift 2
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
# defensive RCCL env: fail fast on wedged collectives, log comm setup issues
export TORCH_NCCL_ASYNC_ERROR_HANDLING=${TORCH_NCCL_ASYNC_ERROR_HANDLING:-1}
export NCCL_DEBUG=${NCCL_DEBUG:-WARN}
NNODES=${NNODES:-1}
NODE_RANK=${NODE_RANK:-0}
MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
MASTER_PORT=${MASTER_PORT:-29500}
if [ "$GPUS" -gt 1 ] || [ "$NNODES" -gt 1 ]; then
  if [ "$NNODES" -gt 1 ]; then
    python -m torch.distributed.run --nnodes "$NNODES" --node-rank "$NODE_RANK" \
      --master-addr "$MASTER_ADDR" --master-port "$MASTER_PORT" \
      --nproc-per-node "$GPUS" "$FILE" --config-file "$CONFIG" "$@"
  else
    python -m torch.distributed.run --standalone --nnodes=1 \
      --nproc-per-node "$GPUS" --local-addr 127.0.0.1 \
      "$FILE" --config-file "$CONFIG" "$@"
  fi
else
  python "$FILE" --config-file "$CONFIG" "$@"
fi
