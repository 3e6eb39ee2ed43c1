"""bench.py driver contract: torchrun multi-rank launch + JSON output schema.

The round driver runs `python -m torch.distributed.run --nproc-per-node N
bench.py --gpus N --steps K --warmup W` and parses one JSON line from rank 0.
This exercises that path end-to-end on CPU (gloo) with tiny shapes.
"""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_two_rank_json_contract():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29781", "bench.py", "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--micro-batch", "2", "--layers", "2",
         "--seq-len", "128", "--hidden", "128", "--heads", "4",
         "--vocab", "1024"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    # the driver-facing schema
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 2
    assert d["steps"] == 1
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4  # micro 2 x dp 2
    assert d["value"] > 0


def test_bench_eight_rank_json_contract():
    """8-rank torchrun launch (the driver's SCALE run shape) on gloo CPU."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29783", "bench.py", "--gpus", "8", "--steps", "1",
         "--warmup", "0", "--micro-batch", "1", "--layers", "2",
         "--seq-len", "64", "--hidden", "64", "--heads", "4",
         "--vocab", "512"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=900,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "dp8"
    assert d["config"]["global_batch"] == 8
    assert d["value"] > 0
