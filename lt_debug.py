import sys, torch
sys.path.insert(0, "/root/repo")
from libai_amd.ops._ext import ext
torch.cuda.init()
for M in [2048, 4096, 8192, 16384, 32768, 49152]:
    for N, K in [(4096, 1024)]:
        dy = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w2 = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        aux = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        try:
            dpre, db = ext().lt_dgelu_bgrad(dy, w2, aux)
            torch.cuda.synchronize()
            # numerics vs torch tanh dgelu
            pre = aux.float().requires_grad_(True)
            torch.nn.functional.gelu(pre, approximate="tanh").backward(dy.float() @ w2.float())
            rel = (dpre.float() - pre.grad).abs().max() / pre.grad.abs().max()
            relb = (db - pre.grad.sum(0)).abs().max() / pre.grad.sum(0).abs().max().clamp(min=1e-3)
            print(f"M{M} N{N} K{K}: OK rel {rel:.4f} dbrel {relb:.5f}")
        except Exception as e:
            print(f"M{M} N{N} K{K}: FAIL {str(e)[:70]}")
