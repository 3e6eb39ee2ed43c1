#!/usr/bin/env python3
"""Run bench.py's training step with fp8 forward GEMMs enabled (evidence
run — the HEADLINE bench stays bf16; this prints dtype fp8-fwd)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from libai_amd.ops import fp8
fp8.set_fp8_gemms(True)
sys.argv = ["bench.py", "--gpus", "1", "--steps", "8", "--warmup", "3"]
root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
exec(compile(open(os.path.join(root, "bench.py")).read(), "bench.py", "exec"))
