import sys, time, torch
sys.path.insert(0, "/root/repo")
import bench as bench_mod
bench_mod._enable_tuned_gemms()
from libai_amd.ops._ext import ext
torch.cuda.init()

def t(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

M = 49152
print("| shape | torch ms | custom ms | custom TF/s | speedup |")
for N, K in [(1024, 1024), (3072, 1024), (4096, 1024), (1024, 4096)]:
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    # correctness first
    got = ext().gemm_dw(dy, x, 0)
    ref = (dy.float().t() @ x.float())
    rel = (got.float() - ref).abs().max() / ref.abs().max()
    tt = t(lambda: torch.matmul(dy.t(), x))
    for s in (0, 4, 8, 16):
        tc = t(lambda: ext().gemm_dw(dy, x, s))
        tf = 2.0 * M * N * K / tc / 1e12
        print(f"| {N}x{K} s={s} | {tt*1e3:.3f} | {tc*1e3:.3f} | {tf:.0f} | {tt/tc:.2f}x | rel={rel:.4f}")
