#!/usr/bin/env python3
"""Padded-batch BERT attention A/B: per-sequence kv_len flash kernels vs the
materialized-scores fallback the padding mask used to force.

Usage (GPU box): python tools/padding_bench.py [--mb 96]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--mb", type=int, default=96)
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--steps", type=int, default=6)
    args = p.parse_args()

    import bench as bench_mod

    bench_mod._enable_tuned_gemms()
    import libai_amd.ops.attention as A
    from libai_amd.models import BertForPreTraining
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    m = BertForPreTraining(
        vocab_size=30592, hidden_size=1024, hidden_layers=24,
        num_attention_heads=16, intermediate_size=4096,
        max_position_embeddings=args.seq,
    ).to(torch.bfloat16).cuda()
    from libai_amd.optim import FusedAdamW, get_default_optimizer_params

    opt = FusedAdamW(get_default_optimizer_params(m, base_lr=1e-4), lr=1e-4)

    # right-padded batch: lengths uniform in [seq/4, seq]
    g = torch.Generator().manual_seed(7)
    lens = torch.randint(args.seq // 4, args.seq + 1, (args.mb,), generator=g)
    ids = torch.randint(5, 30592, (args.mb, args.seq), generator=g).cuda()
    mask = (torch.arange(args.seq)[None, :] < lens[:, None]).to(torch.uint8).cuda()
    batch = dict(
        input_ids=ids, attention_mask=mask,
        ns_labels=torch.randint(0, 2, (args.mb,), generator=g).cuda(),
        lm_labels=ids.clone(),
        loss_mask=(torch.rand(args.mb, args.seq, generator=g) < 0.15).long().cuda(),
    )

    def run(fused):
        orig = A.flash_attention_available
        if not fused:
            A.flash_attention_available = lambda *a, **k: False
        try:
            for _ in range(2):
                opt.zero_grad()
                out = m(**batch)
                sum(v for v in out.values() if v.requires_grad).backward()
                opt.step()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                opt.zero_grad()
                out = m(**batch)
                sum(v for v in out.values() if v.requires_grad).backward()
                opt.step()
            torch.cuda.synchronize()
            return (time.perf_counter() - t0) / args.steps
        finally:
            A.flash_attention_available = orig

    t_unfused = run(False)
    t_fused = run(True)
    tok = args.mb * args.seq
    print(f"# padded BERT-large (mb {args.mb}, seq {args.seq}, random "
          f"[{args.seq // 4}, {args.seq}] lengths)")
    print(f"materialized-scores fallback: {t_unfused * 1e3:.1f} ms/step "
          f"({tok / t_unfused:.0f} tok/s)")
    print(f"kv_len flash kernels:         {t_fused * 1e3:.1f} ms/step "
          f"({tok / t_fused:.0f} tok/s)  "
          f"[{100 * (t_unfused - t_fused) / t_unfused:+.1f}%]")


if __name__ == "__main__":
    main()
