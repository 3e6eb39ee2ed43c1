"""Multi-process test helper: run a function under world_size gloo workers."""

import os
import pickle
import tempfile
import traceback

import torch.multiprocessing as mp


def _worker(rank, world_size, fn, args, port, result_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    try:
        import torch.distributed as dist

        result = fn(rank, world_size, *args)
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("ok", result), f)
        if dist.is_initialized():
            dist.barrier()
            dist.destroy_process_group()
    except Exception:
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("err", traceback.format_exc()), f)
        raise


def run_dist(fn, world_size=2, args=()):
    """Run fn(rank, world_size, *args) in world_size processes (gloo).

    Returns [result_rank0, ...]. Raises on any rank failure.
    """
    import random

    port = random.randint(20000, 50000)
    with tempfile.TemporaryDirectory() as result_dir:
        mp.start_processes(
            _worker,
            args=(world_size, fn, args, port, result_dir),
            nprocs=world_size,
            join=True,
            start_method="spawn",
        )
        results = []
        for r in range(world_size):
            with open(os.path.join(result_dir, f"rank{r}.pkl"), "rb") as f:
                status, payload = pickle.load(f)
            if status == "err":
                raise RuntimeError(f"rank {r} failed:\n{payload}")
            results.append(payload)
        return results
