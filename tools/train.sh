#!/usr/bin/env bash
# Launcher wrapper (reference: tools/train.sh): one rank per GPU over RCCL.
#   bash tools/train.sh tools/train_net.py configs/gpt2_pretrain.py 8 [opts...]
FILE=$1
CONFIG=$2
GPUS=${3:-1}
shift 3 || shift 2
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
if [ "$GPUS" -gt 1 ]; then
  python -m torch.distributed.run --standalone --nnodes=1 --nproc-per-node "$GPUS" \
    --local-addr 127.0.0.1 "$FILE" --config-file "$CONFIG" "$@"
else
  python "$FILE" --config-file "$CONFIG" "$@"
fi
