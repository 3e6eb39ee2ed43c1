#!/bin/bash
# A/B: baseline vs setprio vs stagger flash kernels, bench ms/step
run_bench() {
  timeout 300 python bench.py --steps 8 --warmup 3 2>/dev/null | python -c "import json,sys; d=json.loads(sys.stdin.readlines()[-1]); print(f'{d[\"ms_per_step\"]:.2f} ms  {d[\"value\"]:.0f} tok/s')"
}
rebuild() {
  touch libai_amd/csrc/kernels/flash_attn.hip
  EXTRA_HIP_FLAGS="$1" python libai_amd/csrc/build.py >/dev/null 2>&1
}
echo "=== baseline ==="; run_bench
echo "=== FLASH_SETPRIO ==="; rebuild "-DFLASH_SETPRIO=1"; run_bench
echo "=== FLASH_SETPRIO+STAGGER ==="; rebuild "-DFLASH_SETPRIO=1 -DFLASH_STAGGER=1"; run_bench
echo "=== FLASH_STAGGER ==="; rebuild "-DFLASH_STAGGER=1"; run_bench
echo "=== numerics (last variant) ==="
timeout 300 python -m pytest tests/gpu/test_flash_attn_gpu.py -x -q 2>&1 | tail -1
