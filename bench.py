#!/usr/bin/env python3
"""Flagship training benchmark: GPT-2 345M (hidden 1024, 24 layers, 16 heads,
seq 1024) on synthetic data, bf16, whole-node tokens/s.

Metric/config match BASELINE.json ("tokens/sec (whole node), GPT-2 345M
3D-parallel at 1/2/4/8 MI355X"); the reference numbers are LiBai v0.2.0 fp16
(17.52 samples/s 1n1g ... 125.64 samples/s 1n8g DP8).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched via torch.distributed.run with one rank per GPU over RCCL.  W untimed
warmup steps, then exactly K steps bracketed by barrier+synchronize on both
sides, MAX elapsed over ranks, rank 0 prints ONE JSON line.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

BASELINE_TOKENS_PER_S = {1: 17.52 * 1024, 2: None, 4: 63.45 * 1024, 8: 125.64 * 1024}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--micro-batch", type=int, default=None, help="per-rank micro batch")
    p.add_argument("--acc", type=int, default=1, help="grad accumulation steps")
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--seq-len", type=int, default=1024)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--layers", type=int, default=24)
    p.add_argument("--heads", type=int, default=16)
    p.add_argument("--vocab", type=int, default=50304)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--act-ckpt", action="store_true")
    return p.parse_args()


def _enable_tuned_gemms():
    """Load the committed hipBLASLt TunableOp algo table (gfx950) if present:
    ~3% on the 345M step, no tuning cost at runtime."""
    table = os.path.join(os.path.dirname(os.path.abspath(__file__)), "libai_amd",
                         "data", "tunableop_gfx950.csv")
    if os.path.exists(table) and "PYTORCH_TUNABLEOP_ENABLED" not in os.environ:
        import shutil
        import tempfile

        # torch inserts the device ordinal before .csv; provide all 8
        d = tempfile.mkdtemp(prefix="tunableop_")
        for i in range(8):
            shutil.copy(table, os.path.join(d, f"tunableop{i}.csv"))
        os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
        os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
        os.environ["PYTORCH_TUNABLEOP_FILENAME"] = os.path.join(d, "tunableop.csv")


def main():
    args = parse_args()
    _enable_tuned_gemms()
    world = int(os.environ.get("WORLD_SIZE", 1))
    assert world == args.gpus or world == 1, (
        f"WORLD_SIZE {world} != --gpus {args.gpus}; launch N>1 with torchrun"
    )
    n = max(world, 1)

    from libai_amd.utils import distributed as du

    dutil = du.setup_dist_util(
        dict(
            tensor_parallel_size=args.tp,
            pipeline_parallel_size=args.pp,
            pipeline_num_layers=args.layers,
        )
    )
    rank = dutil.rank
    device = du.get_device()
    if device.type == "cuda":
        torch.cuda.set_device(device)
    torch.manual_seed(1234)  # identical init across DP ranks

    dp = dutil.data_parallel_size
    micro = args.micro_batch
    if micro is None:
        micro = 8 if device.type == "cuda" else 2
    global_batch = micro * dp * args.acc

    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    if device.type == "cpu" and args.dtype == "bf16":
        dtype = torch.float32  # CPU smoke runs in fp32

    # model: GPT-2 345M-class random init
    from libai_amd.models import GPTForPreTraining
    from libai_amd.optim import FusedAdamW, get_default_optimizer_params

    model = GPTForPreTraining(
        hidden_layers=args.layers,
        vocab_size=args.vocab,
        hidden_size=args.hidden,
        ffn_hidden_size=4 * args.hidden,
        num_attention_heads=args.heads,
        max_seq_length=args.seq_len,
        embedding_dropout_prob=0.1,
        attention_dropout_prob=0.1,
        output_dropout_prob=0.1,
    )
    if args.act_ckpt:
        model.set_activation_checkpoint(True)
    model = model.to(dtype)

    pipeline = None
    if args.pp > 1:
        from libai_amd.parallel.pipeline import PipelineScheduler

        model.hidden_size = args.hidden
        pipeline = PipelineScheduler(model, dtype=dtype)
    model = model.to(device)
    model.train()

    optimizer = FusedAdamW(
        get_default_optimizer_params(model, base_lr=1.5e-4, weight_decay=0.01),
        lr=1.5e-4,
        weight_decay=0.01,
        clip_grad=1.0,
    )

    torch.manual_seed(du.same_seed_for_tp_group(1234))  # diverge dropout per dp rank

    # synthetic batches, pre-generated on host, moved in-step
    gen = torch.Generator().manual_seed(4321 + dutil.data_parallel_rank)
    pool = []
    for _ in range(4):
        toks = torch.randint(0, args.vocab, (micro, args.seq_len + 1), generator=gen)
        pool.append(
            {
                "input_ids": toks[:, :-1].contiguous().pin_memory()
                if device.type == "cuda"
                else toks[:, :-1].contiguous(),
                "labels": toks[:, 1:].contiguous().pin_memory()
                if device.type == "cuda"
                else toks[:, 1:].contiguous(),
            }
        )

    import torch.distributed as dist

    def to_dev(b):
        return {k: v.to(device, non_blocking=True) for k, v in b.items()}

    def sync_dp_grads():
        optimizer.grad_sync()

    step_i = 0

    def one_step():
        nonlocal step_i
        if pipeline is not None:
            batches = [to_dev(pool[(step_i * args.acc + j) % len(pool)])
                       for j in range(max(args.acc, 1))]
            pipeline.run_1f1b(batches)
        else:
            for j in range(args.acc):
                data = to_dev(pool[(step_i * args.acc + j) % len(pool)])
                losses = model(**data)
                total = sum(v for v in losses.values()) / args.acc
                total.backward()
        sync_dp_grads()
        optimizer.step()
        optimizer.zero_grad()
        step_i += 1

    def barrier_sync():
        if dist.is_initialized():
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens = args.steps * global_batch * args.seq_len
    tokens_per_s = tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000
    base = BASELINE_TOKENS_PER_S.get(n)
    par = f"dp{dp}"
    if args.tp > 1:
        par += f"_tp{args.tp}"
    if args.pp > 1:
        par += f"_pp{args.pp}"

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "tokens/sec (whole node), GPT-2 345M 3D-parallel at 1/2/4/8 MI355X",
                    "value": round(tokens_per_s, 1),
                    "unit": "tokens/s",
                    "n_gpus": n,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": round(ms_per_step, 2),
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": round(tokens_per_s / base, 2) if base else None,
                    "dtype": args.dtype if device.type == "cuda" else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": "gpt2-345m (nl24 h1024 nah16 seq1024)",
                        "global_batch": global_batch,
                        "seq_len": args.seq_len,
                        "parallelism": par,
                    },
                }
            ),
            flush=True,
        )


if __name__ == "__main__":
    main()
