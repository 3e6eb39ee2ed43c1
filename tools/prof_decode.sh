#!/usr/bin/env bash
# Kernel-level profile of the captured serving loop (run on a GPU box):
#   bash tools/prof_decode.sh
# Writes gpurun_out/decode_kernel_stats.txt (top kernels of the replayed
# decode graph, rocprofv3 --kernel-trace --stats only: PMC-safe flags).
set -e
cd /tmp && export TMPDIR=/tmp
ROOT=${GRAFT_REPO_ROOT:-/root/repo}
mkdir -p "$ROOT/gpurun_out" /tmp/dprof
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/dprof -o dec -- \
  python "$ROOT/tools/serving_bench.py" --slots 32 --rounds 4 > /tmp/dprof/run.log 2>&1 || true
tail -2 /tmp/dprof/run.log > "$ROOT/gpurun_out/decode_kernel_stats.txt"
python "$ROOT/tools/summarize_prof.py" "/tmp/dprof/**/*.db" \
  "$ROOT/gpurun_out/decode_kernel_stats.md" 1 \
  "captured continuous-batching serving loop, GPT-2 345M, 32 slots (tools/serving_bench.py --rounds 4; totals over the whole run incl. prefills)" \
  >> "$ROOT/gpurun_out/decode_kernel_stats.txt" 2>&1 || \
  find /tmp/dprof -type f >> "$ROOT/gpurun_out/decode_kernel_stats.txt"
