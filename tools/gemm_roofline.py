#!/usr/bin/env python3
"""Per-shape GEMM roofline for the GPT-2 345M Linear1D sites vs hipBLASLt.

Measures fwd + both bwd GEMMs for each of the five hot linear shapes
(qkv, attn-out, h->4h, 4h->h, lm-logits) at the bench micro-batch, plus the
epilogue-fused MLP pair, and reports TF/s and % of the 2.5 PFLOP/s bf16
dense peak.  Writes markdown to stdout (redirect into profiles/).

Usage (GPU box): python tools/gemm_roofline.py [--mb 48] [--seq 1024]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

PEAK_TFLOPS = 2500.0  # MI355X dense bf16 (no sparsity)


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def gemm_tf(M, N, K, secs):
    return 2.0 * M * N * K / secs / 1e12


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--mb", type=int, default=48)
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--vocab", type=int, default=50304)
    p.add_argument("--fp8", action="store_true",
                   help="also measure fp8 e4m3 via torch._scaled_mm "
                        "(gfx950 fp8 dense peak ~5 PF/s)")
    args = p.parse_args()

    import bench as bench_mod  # noqa: F401  (loads the tunableop table)

    bench_mod._enable_tuned_gemms()
    torch.cuda.init()

    M = args.mb * args.seq
    H = args.hidden
    shapes = [
        ("qkv (h->3h)", M, 3 * H, H),
        ("attn-out (h->h)", M, H, H),
        ("mlp h->4h", M, 4 * H, H),
        ("mlp 4h->h", M, H, 4 * H),
        ("lm-logits (h->v)", M, args.vocab, H),
    ]
    print(f"# GEMM roofline — GPT-2 345M shapes, mb{args.mb} seq{args.seq} bf16\n")
    print(f"M = {M} tokens; peak = {PEAK_TFLOPS:.0f} TF/s dense bf16 "
          "(AMD headline /2, no sparsity)\n")
    print("| site | GEMM | M x N x K | ms | TF/s | % peak |")
    print("|---|---|---|---|---|---|")
    for name, m, n, k in shapes:
        x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
        t_f = bench(lambda: torch.matmul(x, w.t()))
        t_dx = bench(lambda: torch.matmul(dy, w))
        t_dw = bench(lambda: torch.matmul(dy.t(), x))
        for g, t in (("fwd", t_f), ("bwd dX", t_dx), ("bwd dW", t_dw)):
            tf = gemm_tf(m, n, k, t)
            print(f"| {name} | {g} | {m}x{n}x{k} | {t * 1e3:.3f} | "
                  f"{tf:.0f} | {100 * tf / PEAK_TFLOPS:.1f}% |")
        if args.fp8:
            a8 = x.to(torch.float8_e4m3fn)
            b8 = w.to(torch.float8_e4m3fn).t()
            sc = torch.tensor(1.0, device="cuda")
            t8 = bench(lambda: torch._scaled_mm(
                a8, b8, scale_a=sc, scale_b=sc, out_dtype=torch.bfloat16))
            tf8 = gemm_tf(m, n, k, t8)
            print(f"| {name} | fwd FP8 e4m3 | {m}x{n}x{k} | {t8 * 1e3:.3f} | "
                  f"{tf8:.0f} | {100 * tf8 / (2 * PEAK_TFLOPS):.1f}% of 5PF |")
        del x, w, dy

    # epilogue-fused MLP A/B (bias+gelu in the GEMM vs separate kernels)
    from libai_amd.ops.fused_bias import bias_gelu
    from libai_amd.ops.fused_mlp import fused_mlp_available
    from libai_amd.ops._ext import ext
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    x = torch.randn(M, H, device="cuda", dtype=torch.bfloat16)
    w1 = torch.randn(4 * H, H, device="cuda", dtype=torch.bfloat16)
    b1 = torch.randn(4 * H, device="cuda", dtype=torch.bfloat16)
    w2 = torch.randn(H, 4 * H, device="cuda", dtype=torch.bfloat16)
    print("\n## MLP gelu-backward fusion A/B (DGELU_BGRAD epilogue)\n")
    if fused_mlp_available(x):
        import torch.nn.functional as FF

        pre = FF.linear(x, w1, b1)
        dy = torch.randn(M, H, device="cuda", dtype=torch.bfloat16)

        def separate():
            dy4h = torch.matmul(dy, w2)  # dX GEMM of the row linear
            from libai_amd.ops.fused_bias import bias_gelu as _bg  # noqa

            dpre, db = ext().bias_gelu_bwd(pre, torch.zeros_like(b1), dy4h, True)
            return dpre, db

        def fused():
            return ext().lt_dgelu_bgrad(dy, w2, pre)

        t_sep = bench(separate)
        t_fus = bench(fused)
        print(f"separate (dX GEMM + dgelu + colsum): {t_sep * 1e3:.3f} ms")
        print(f"fused DGELU_BGRAD GEMM:              {t_fus * 1e3:.3f} ms "
              f"({100 * (t_sep - t_fus) / t_sep:+.1f}%)")
    else:
        print("hipBLASLt DGELU_BGRAD unavailable on this stack")


if __name__ == "__main__":
    main()
