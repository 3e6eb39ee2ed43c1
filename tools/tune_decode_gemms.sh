#!/usr/bin/env bash
# Tune the M=32 decode GEMM shapes into a fresh TunableOp CSV (GPU box):
#   bash tools/tune_decode_gemms.sh
# Writes gpurun_out/tunableop_decode.csv — merge new rows into
# libai_amd/data/tunableop_gfx950.csv.
set -e
ROOT=${GRAFT_REPO_ROOT:-/root/repo}
mkdir -p "$ROOT/gpurun_out" /tmp/tune && cd /tmp/tune
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=/tmp/tune/tunableop.csv
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=80
# seed with the existing table so training shapes aren't re-tuned
cp "$ROOT/libai_amd/data/tunableop_gfx950.csv" /tmp/tune/tunableop0.csv
timeout 400 python - <<'PY'
import torch
# GPT-2 345M + Llama-1B decode GEMM shapes at serving batch 32 (x [32,K] @ w[N,K]^T)
shapes = [
    (32, 3072, 1024), (32, 1024, 1024), (32, 4096, 1024), (32, 1024, 4096),
    (32, 50304, 1024),                     # gpt2 sites + lm head
    (32, 2048, 2048), (32, 1024, 2048),    # llama1b qkv-ish / kv
    (32, 11008, 2048), (32, 2048, 5504), (32, 32000, 2048),
]
for m, n, k in shapes:
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        torch.nn.functional.linear(x, w)
    torch.cuda.synchronize()
    print("tuned", m, n, k)
PY
cp /tmp/tune/tunableop.csv "$ROOT/gpurun_out/tunableop_decode.csv" 2>/dev/null || \
  cp /tmp/tune/tunableop0.csv "$ROOT/gpurun_out/tunableop_decode.csv"
wc -l "$ROOT/gpurun_out/tunableop_decode.csv"
