#!/usr/bin/env python3
"""Serving decode micro-bench: fused flash_decode vs the unfused
bmm+softmax chain, GPT-2 345M shape, batch x 1-token steps over a warm
KV cache.

Usage (GPU box): python tools/decode_bench.py [--batch 32] [--ctx 1024]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--ctx", type=int, default=1024)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--model", default="gpt2",
                   choices=["gpt2", "llama1b", "llama7b"])
    args = p.parse_args()

    import bench as bench_mod

    bench_mod._enable_tuned_gemms()
    from libai_amd.models import GPTForPreTraining
    from libai_amd.utils import distributed as du
    import libai_amd.ops.attention as A

    du.setup_dist_util({})
    torch.manual_seed(0)
    if args.model in ("llama1b", "llama7b"):
        from libai_amd.models import LlamaForCausalLM

        shape = (dict(hidden_layers=16, hidden_size=2048,
                      intermediate_size=5504, num_attention_heads=16,
                      num_key_value_heads=4)
                 if args.model == "llama1b" else
                 dict(hidden_layers=32, hidden_size=4096,
                      intermediate_size=11008, num_attention_heads=32,
                      num_key_value_heads=32))
        m = LlamaForCausalLM(
            vocab_size=32000, max_position_embeddings=4096, **shape,
        ).to(torch.bfloat16).cuda().eval()
        vocab = 32000
    else:
        m = GPTForPreTraining(
            hidden_layers=24, vocab_size=50304, hidden_size=1024,
            ffn_hidden_size=4096, num_attention_heads=16, max_seq_length=2048,
            embedding_dropout_prob=0.0, attention_dropout_prob=0.0,
            output_dropout_prob=0.0,
        ).to(torch.bfloat16).cuda().eval()
        vocab = 50304

    ids = torch.randint(0, vocab, (args.batch, args.ctx), device="cuda")

    def run(use_fused):
        orig = A.decode_attention_available
        if not use_fused:
            A.decode_attention_available = lambda q, hd: False
        try:
            with torch.no_grad():
                out = m(input_ids=ids, use_cache=True)
                past = out["past_key_values"]
                tok = ids[:, -1:]
                for _ in range(4):  # warmup
                    s = m(input_ids=tok, past_key_values=past, use_cache=True)
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for _ in range(args.steps):
                    s = m(input_ids=tok, past_key_values=past, use_cache=True)
                    tok = s["prediction_scores"][:, -1:].argmax(-1)
                torch.cuda.synchronize()
                dt = (time.perf_counter() - t0) / args.steps
        finally:
            A.decode_attention_available = orig
        return dt

    t_unfused = run(False)
    t_fused = run(True)
    print(f"# decode bench: {args.model}, batch {args.batch}, ctx {args.ctx}")
    print(f"unfused bmm+softmax decode: {t_unfused * 1e3:.3f} ms/step "
          f"({args.batch / t_unfused:.0f} tok/s)")
    print(f"fused flash_decode:         {t_fused * 1e3:.3f} ms/step "
          f"({args.batch / t_fused:.0f} tok/s)  "
          f"[{100 * (t_unfused - t_fused) / t_unfused:+.1f}% step time]")

    # hipGraph-captured step: one replay per token
    from libai_amd.inference.captured_decode import (
        CapturedGPTDecoder,
        CapturedLlamaDecoder,
    )

    cls = CapturedGPTDecoder if args.model == "gpt2" else CapturedLlamaDecoder
    dec = cls(m, max_batch=args.batch, max_seq_len=args.ctx + args.steps + 8)
    dec.generate(ids, args.steps)  # capture + warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    dec.generate(ids, args.steps)
    torch.cuda.synchronize()
    t_cap = (time.perf_counter() - t0) / args.steps
    # subtract the prefill (measured separately) to isolate the step
    with torch.no_grad():
        torch.cuda.synchronize(); t0 = time.perf_counter()
        m(input_ids=ids, use_cache=True)
        torch.cuda.synchronize()
    t_prefill = time.perf_counter() - t0
    t_step = (t_cap * args.steps - t_prefill) / args.steps
    print(f"hipGraph-captured decode:   {t_step * 1e3:.3f} ms/step "
          f"({args.batch / t_step:.0f} tok/s)  "
          f"[{100 * (t_fused - t_step) / t_fused:+.1f}% vs fused eager]")


if __name__ == "__main__":
    main()
