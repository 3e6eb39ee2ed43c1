#!/usr/bin/env python3
"""Micro A/B: where does the fp8 end-to-end regression come from?"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

def t(fn, n=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e3

def main():
    import bench as bm; bm._enable_tuned_gemms()
    from libai_amd.ops import fp8
    from libai_amd.utils import distributed as du
    du.setup_dist_util({})
    fp8.set_fp8_gemms(True)
    M, K, N = 49152, 1024, 3072
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    print(f"shape {M}x{N}x{K}")
    print(f"F.linear fwd (no bias):   {t(lambda: F.linear(x, w)):.3f} ms")
    print(f"F.linear fwd (bias):      {t(lambda: F.linear(x, w, b)):.3f} ms")
    with torch.no_grad():
        print(f"fp8_linear fwd (no bias): {t(lambda: fp8.fp8_linear(x, w)):.3f} ms")
        print(f"fp8_linear fwd (bias):    {t(lambda: fp8.fp8_linear(x, w, b)):.3f} ms")
        x8, sx = fp8._quant(x); w8, sw = fp8._quant(w)
        print(f"  _scaled_mm only:        {t(lambda: torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw, out_dtype=torch.bfloat16)):.3f} ms")
        print(f"  _scaled_mm + bias:      {t(lambda: torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw, bias=b, out_dtype=torch.bfloat16)):.3f} ms")
        print(f"  quant(x) only:          {t(lambda: fp8._quant(x)):.3f} ms")
        st = fp8.DelayedScale(); st.quant(x.detach())
        print(f"  fused delayed quant:    {t(lambda: st.quant(x.detach())):.3f} ms")
        stx, stw = fp8.DelayedScale(), fp8.DelayedScale()
        print(f"fp8_linear fwd (delayed): {t(lambda: fp8.fp8_linear(x, w, b, x_state=stx, w_state=stw)):.3f} ms")
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    def fl_full():
        y = F.linear(x, w, b); y.backward(dy); x.grad = w.grad = b.grad = None
    stx2, stw2 = fp8.DelayedScale(), fp8.DelayedScale()
    def f8_full():
        y = fp8.fp8_linear(x, w, b, x_state=stx2, w_state=stw2)
        y.backward(dy); x.grad = w.grad = b.grad = None
    print(f"F.linear fwd+bwd:         {t(fl_full):.3f} ms")
    print(f"fp8_linear fwd+bwd:       {t(f8_full):.3f} ms")
    dy2 = dy
    print(f"  bwd dX dy@w:            {t(lambda: dy2 @ w):.3f} ms")
    print(f"  bwd dW dy.t@x:          {t(lambda: dy2.t() @ x.detach()):.3f} ms")

if __name__ == "__main__":
    main()
