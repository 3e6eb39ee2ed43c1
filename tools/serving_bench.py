#!/usr/bin/env python3
"""Continuous-batching serving bench: mixed-length request stream through
the captured decode loop (GPT-2 345M), aggregate generated tokens/s.

Usage (GPU box): python tools/serving_bench.py [--slots 32] [--rounds 20]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--slots", type=int, default=32)
    p.add_argument("--rounds", type=int, default=20)
    p.add_argument("--chunk", type=int, default=32)  # replays per harvest
    p.add_argument("--max-seq", type=int, default=1024)
    p.add_argument("--model", default="gpt2", choices=["gpt2", "llama1b"])
    args = p.parse_args()

    import bench as bench_mod

    bench_mod._enable_tuned_gemms()
    from libai_amd.inference.captured_decode import (
        ContinuousGPTDecoder,
        ContinuousLlamaDecoder,
    )
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    if args.model == "llama1b":
        from libai_amd.models import LlamaForCausalLM

        m = LlamaForCausalLM(
            hidden_layers=16, vocab_size=32000, hidden_size=2048,
            intermediate_size=5504, num_attention_heads=16,
            num_key_value_heads=4, max_position_embeddings=4096,
        ).to(torch.bfloat16).cuda().eval()
        vocab = 32000
        cls = ContinuousLlamaDecoder
    else:
        from libai_amd.models import GPTForPreTraining

        m = GPTForPreTraining(
            hidden_layers=24, vocab_size=50304, hidden_size=1024,
            ffn_hidden_size=4096, num_attention_heads=16, max_seq_length=2048,
            embedding_dropout_prob=0.0, attention_dropout_prob=0.0,
            output_dropout_prob=0.0,
        ).to(torch.bfloat16).cuda().eval()
        vocab = 50304
        cls = ContinuousGPTDecoder

    dec = cls(m, max_batch=args.slots,
              max_seq_len=args.max_seq, ring_cap=256)
    g = torch.Generator().manual_seed(1)

    def rand_prompt():
        L = int(torch.randint(64, 512, (1,), generator=g))
        return torch.randint(0, vocab, (L,), generator=g).cuda()

    # fill all slots, warm + capture
    for s in range(args.slots):
        dec.add_request(s, rand_prompt())
    dec.step(args.chunk)
    torch.cuda.synchronize()

    done_tokens = 0
    t0 = time.perf_counter()
    for _ in range(args.rounds):
        dec.step(args.chunk)
        torch.cuda.synchronize()
        # harvest/recycle: treat every chunk as a finished request tail
        # (EOS-free synthetic stream: recycle the slot with a new prompt)
        for s in range(args.slots):
            n = int(dec.slot_step[s])
            if n + args.chunk + 2 >= dec.ring_cap or \
               int(dec.pos[s]) + args.chunk + 2 >= args.max_seq:
                done_tokens += n
                dec.release(s)
                dec.add_request(s, rand_prompt())
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    total = args.rounds * args.chunk * args.slots
    print(f"# continuous-batching serving bench: {args.model}, "
          f"{args.slots} slots, chunk {args.chunk}, {args.rounds} rounds")
    print(f"decode throughput: {total / dt:.0f} tok/s aggregate "
          f"({dt / (args.rounds * args.chunk) * 1e3:.3f} ms/step incl. "
          f"admission prefills)")


if __name__ == "__main__":
    main()
