"""HIP kernel numerics vs plain PyTorch fp32 references (run on MI355X)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _setup():
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    from libai_amd.ops._ext import has_ext

    assert has_ext(), "HIP extension must be built+loaded on the GPU box"
    yield


def _rel_err(a, b):
    return ((a.float() - b.float()).abs().max() / (b.float().abs().max() + 1e-6)).item()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(8, 1024), (33, 768), (128, 4096), (5, 31)])
def test_layer_norm_fwd_bwd(dtype, shape):
    from libai_amd.ops.norm import layer_norm

    torch.manual_seed(0)
    x = torch.randn(*shape, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(shape[-1], device="cuda", dtype=dtype, requires_grad=True)
    b = torch.randn(shape[-1], device="cuda", dtype=dtype, requires_grad=True)
    y = layer_norm(x, w, b, 1e-5)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(xr, (shape[-1],), wr, br, 1e-5)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert _rel_err(y, yr) < tol

    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g.float())
    assert _rel_err(x.grad, xr.grad) < tol * 3
    assert _rel_err(w.grad, wr.grad) < tol * 3
    assert _rel_err(b.grad, br.grad) < tol * 3


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_rms_norm_fwd_bwd(dtype):
    from libai_amd.ops.norm import rms_norm

    torch.manual_seed(0)
    x = torch.randn(64, 1024, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(1024, device="cuda", dtype=dtype, requires_grad=True)
    y = rms_norm(x, w, 1e-6)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    yr = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-6) * wr
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert _rel_err(y, yr) < tol
    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g.float())
    assert _rel_err(x.grad, xr.grad) < tol * 3
    assert _rel_err(w.grad, wr.grad) < tol * 3


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_bias_gelu(dtype):
    from libai_amd.ops.fused_bias import bias_gelu

    torch.manual_seed(0)
    x = torch.randn(32, 512, device="cuda", dtype=dtype, requires_grad=True)
    b = torch.randn(512, device="cuda", dtype=dtype, requires_grad=True)
    y = bias_gelu(x, b)
    xr = x.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    yr = torch.nn.functional.gelu(xr + br)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert _rel_err(y, yr) < tol
    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g.float())
    assert _rel_err(x.grad, xr.grad) < tol * 3
    assert _rel_err(b.grad, br.grad) < tol * 3


def test_bias_dropout_add_statistics_and_determinism():
    from libai_amd.ops.fused_bias import bias_dropout_add

    torch.manual_seed(0)
    x = torch.randn(256, 1024, device="cuda", dtype=torch.bfloat16)
    b = torch.zeros(1024, device="cuda", dtype=torch.bfloat16)
    r = torch.zeros_like(x)
    p = 0.3
    y = bias_dropout_add(x, b, r, p=p, training=True)
    kept = (y != 0).float().mean().item()
    assert abs(kept - (1 - p)) < 0.02, f"keep rate {kept} vs {1 - p}"
    # surviving entries are scaled by 1/(1-p)
    mask = y != 0
    ratio = (y[mask].float() / x[mask].float()).mean().item()
    assert abs(ratio - 1 / (1 - p)) < 1e-2


def test_bias_dropout_add_backward_mask_matches_forward():
    from libai_amd.ops.fused_bias import bias_dropout_add

    torch.manual_seed(7)
    x = torch.randn(64, 512, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = bias_dropout_add(x, None, None, p=0.5, training=True)
    g = torch.ones_like(y)
    y.backward(g)
    # grad nonzero exactly where forward kept (bias=0, residual=None)
    kept_fwd = y.detach() != 0
    kept_bwd = x.grad != 0
    x_nonzero = x.detach() != 0
    assert torch.equal(kept_fwd & x_nonzero, kept_bwd & x_nonzero)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("causal,sq,sk", [(True, 128, 128), (False, 64, 64),
                                          (True, 1, 128), (False, 33, 96)])
def test_fused_softmax(dtype, causal, sq, sk):
    from libai_amd.ops.softmax import fused_scale_mask_softmax

    torch.manual_seed(0)
    b, nh = 2, 4
    s = torch.randn(b, nh, sq, sk, device="cuda", dtype=dtype, requires_grad=True)
    mask = None
    if not causal:
        mask = (torch.rand(b, sq, sk, device="cuda") < 0.2).to(torch.uint8)
    scale = 0.125
    y = fused_scale_mask_softmax(s, pad_mask=mask, scale=scale, p=0.0, causal=causal)

    sr = s.detach().float().requires_grad_(True)
    sf = sr * scale
    if causal:
        cm = torch.ones(sq, sk, dtype=torch.bool, device="cuda").tril_(sk - sq)
        sf = sf.masked_fill(~cm, float("-inf"))
    if mask is not None:
        sf = sf - 10000.0 * mask[:, None, :, :].float()
    yr = torch.softmax(sf, dim=-1)
    tol = 1e-5 if dtype == torch.float32 else 1e-2
    assert _rel_err(y, yr) < tol

    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g.float())
    assert _rel_err(s.grad, sr.grad) < tol * 3


def test_fused_softmax_dropout_backward_consistency():
    """With dropout on, compare against a manual recompute using the output's
    observed mask."""
    from libai_amd.ops.softmax import fused_scale_mask_softmax

    torch.manual_seed(0)
    s = torch.randn(2, 2, 64, 64, device="cuda", dtype=torch.float32,
                    requires_grad=True)
    y = fused_scale_mask_softmax(s, scale=0.2, p=0.4, causal=True)
    # reconstruct P and mask from the dropped output
    sr = s.detach().float() * 0.2
    cm = torch.ones(64, 64, dtype=torch.bool, device="cuda").tril_()
    p_ref = torch.softmax(sr.masked_fill(~cm, float("-inf")), dim=-1)
    keep = (y != 0) | (p_ref < 1e-12)
    scale_obs = torch.where(y != 0, y / p_ref.clamp_min(1e-30), torch.zeros_like(y))
    # kept entries should be P / (1-p)
    kept_vals = scale_obs[y != 0]
    assert (kept_vals - 1 / 0.6).abs().median() < 1e-3

    g = torch.ones_like(y)
    y.backward(g)
    assert torch.isfinite(s.grad).all()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_vocab_ce(dtype):
    from libai_amd.ops.cross_entropy import vocab_parallel_cross_entropy

    torch.manual_seed(0)
    logits = torch.randn(128, 1000, device="cuda", dtype=dtype, requires_grad=True)
    target = torch.randint(0, 1000, (128,), device="cuda")
    target[5] = -100  # ignore
    loss = vocab_parallel_cross_entropy(logits, target, ignore_index=-100)
    lr = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lr, target, reduction="none",
                                            ignore_index=-100)
    tol = 1e-4 if dtype == torch.float32 else 3e-2
    assert _rel_err(loss, ref) < tol
    loss.mean().backward()
    ref.mean().backward()
    assert _rel_err(logits.grad, lr.grad) < tol * 3
    assert (logits.grad[5] == 0).all(), "ignored row must have zero grad"


def test_fused_adamw_matches_torch_on_gpu():
    from libai_amd.optim import FusedAdamW

    torch.manual_seed(0)
    m1 = torch.nn.Linear(64, 64).cuda()
    m2 = torch.nn.Linear(64, 64).cuda()
    m2.load_state_dict(m1.state_dict())
    o1 = FusedAdamW(m1.parameters(), lr=1e-2, weight_decay=0.01)
    o2 = torch.optim.AdamW(m2.parameters(), lr=1e-2, weight_decay=0.01)
    for i in range(5):
        x = torch.randn(8, 64, device="cuda")
        o1.zero_grad()
        m1(x).pow(2).mean().backward()
        o1.step()
        o2.zero_grad()
        m2(x).pow(2).mean().backward()
        o2.step()
    assert _rel_err(m1.weight, m2.weight) < 1e-5


def test_fused_adamw_bf16_master_weights():
    from libai_amd.optim import FusedAdamW

    torch.manual_seed(0)
    m = torch.nn.Linear(128, 128).to(torch.bfloat16).cuda()
    opt = FusedAdamW(m.parameters(), lr=1e-3, weight_decay=0.0)
    w0 = m.weight.detach().clone()
    for _ in range(3):
        opt.zero_grad()
        m(torch.randn(8, 128, device="cuda", dtype=torch.bfloat16)).pow(2).mean().backward()
        opt.step()
    assert not torch.equal(m.weight, w0)
    # master is fp32 and in sync with bf16 copy
    _, b = opt.buckets[0]
    assert b.flat_master.dtype == torch.float32
    assert _rel_err(b.flat_param.float(), b.flat_master) < 1e-2


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_fused_embedding_matches_torch(dtype):
    """K10 gather fwd + atomic scatter bwd vs F.embedding (fp32 oracle)."""
    from libai_amd.ops.embedding import fused_embedding

    torch.manual_seed(0)
    V, H, N = 1000, 256, 4096
    w = torch.randn(V, H, device="cuda", dtype=dtype, requires_grad=True)
    ids = torch.randint(0, V, (8, N // 8), device="cuda")
    out = fused_embedding(ids, w)
    wr = w.detach().float().requires_grad_(True)
    ref = torch.nn.functional.embedding(ids, wr)
    assert torch.equal(out.float(), ref.detach().float())  # gather is exact

    g = torch.randn_like(ref)
    out.backward(g.to(dtype))
    ref.backward(g)
    err = (w.grad.float() - wr.grad).abs().max().item()
    tol = 0.05 if dtype == torch.bfloat16 else 1e-4
    assert err < tol, f"embedding bwd max err {err}"


def test_fused_embedding_vocab_shard_and_padding():
    from libai_amd.ops.embedding import fused_embedding

    torch.manual_seed(1)
    V, H = 512, 128
    half = V // 2
    w_full = torch.randn(V, H, device="cuda", dtype=torch.float32)
    ids = torch.randint(0, V, (64, 32), device="cuda")

    # shard 1 (rows half..V): OOV ids -> zero rows; sum of shard outputs == full
    outs = []
    grads = []
    for start in (0, half):
        shard = w_full[start:start + half].clone().requires_grad_(True)
        o = fused_embedding(ids, shard, vocab_start=start)
        o.sum().backward()
        outs.append(o)
        grads.append(shard.grad)
    full = outs[0] + outs[1]
    ref = torch.nn.functional.embedding(ids, w_full)
    assert torch.equal(full, ref)
    wr = w_full.clone().requires_grad_(True)
    torch.nn.functional.embedding(ids, wr).sum().backward()
    assert torch.allclose(torch.cat(grads), wr.grad, atol=1e-4)

    # padding_idx: row grad stays zero
    w = torch.randn(V, H, device="cuda", dtype=torch.float32, requires_grad=True)
    ids_pad = torch.full((4, 8), 7, device="cuda")
    o = fused_embedding(ids_pad, w, padding_idx=7)
    o.sum().backward()
    assert w.grad.abs().max().item() == 0.0


def test_lt_fused_mlp_matches_reference():
    """hipBLASLt epilogue MLP (gelu-aux fwd, dgelu+bgrad bwd) vs fp32 ref."""
    from libai_amd.ops.fused_mlp import fused_mlp, fused_mlp_available

    x_probe = torch.randn(2, 2, device="cuda", dtype=torch.bfloat16)
    if not fused_mlp_available(x_probe):
        pytest.skip("hipBLASLt epilogues unavailable on this stack")

    torch.manual_seed(0)
    M, H, F = 512, 256, 1024
    x = torch.randn(4, M // 4, H, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w1 = (torch.randn(F, H, device="cuda") * 0.05).to(torch.bfloat16).requires_grad_(True)
    b1 = (torch.randn(F, device="cuda") * 0.05).to(torch.bfloat16).requires_grad_(True)
    w2 = (torch.randn(H, F, device="cuda") * 0.05).to(torch.bfloat16).requires_grad_(True)

    y = fused_mlp(x, w1, b1, w2)
    g = torch.randn_like(y)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    w1r = w1.detach().float().requires_grad_(True)
    b1r = b1.detach().float().requires_grad_(True)
    w2r = w2.detach().float().requires_grad_(True)
    # hipBLASLt GELU is the tanh approximation
    ref = torch.nn.functional.gelu(xr @ w1r.t() + b1r, approximate="tanh") @ w2r.t()
    ref.backward(g.float())

    assert (y.float() - ref).abs().max().item() < 0.15, \
        (y.float() - ref).abs().max().item()
    for got, want, name, tol in (
        (x.grad, xr.grad, "dx", 0.3),
        (w1.grad, w1r.grad, "dw1", 0.5),
        (b1.grad, b1r.grad, "db1", 0.3),
        (w2.grad, w2r.grad, "dw2", 0.5),
    ):
        rel = (got.float() - want).abs().max() / want.abs().max().clamp(min=1e-3)
        assert rel.item() < tol, f"{name} rel err {rel.item()}"


def test_mlp_layer_uses_fused_path_gpu():
    """The MLP module's fused path trains and matches the unfused path
    closely (different gelu flavor: tanh vs erf)."""
    import libai_amd.ops.fused_mlp as fm
    from libai_amd.layers.mlp import MLP

    if not fm.fused_mlp_available(
        torch.randn(2, 2, device="cuda", dtype=torch.bfloat16)
    ):
        pytest.skip("hipBLASLt epilogues unavailable")

    torch.manual_seed(0)
    mlp = MLP(256, 1024, output_dropout_prob=0.0).to(torch.bfloat16).cuda()
    x = torch.randn(2, 64, 256, device="cuda", dtype=torch.bfloat16)
    res = torch.zeros_like(x)
    out_fused = mlp(x, residual=res)
    saved = fm._PROBED
    try:
        fm._PROBED = False  # force the unfused erf path
        out_ref = mlp(x, residual=res)
    finally:
        fm._PROBED = saved
    # tanh-vs-erf gelu differ by <3e-3 on typical activations (bf16 noise)
    assert (out_fused.float() - out_ref.float()).abs().max().item() < 0.1


def test_gemm_dw_matches_reference():
    """split-K dW GEMM (dY^T @ X) vs fp32 matmul, several shapes/splits."""
    from libai_amd.ops._ext import ext

    torch.manual_seed(0)
    for M, N, K in [(4096, 1024, 1024), (1000, 256, 384), (8192, 3072, 1024)]:
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        got = ext().gemm_dw(dy, x, 0)
        ref = dy.float().t() @ x.float()
        rel = (got.float() - ref).abs().max() / ref.abs().max()
        assert rel.item() < 2e-2, f"M{M} N{N} K{K}: rel {rel.item()}"
        # explicit split counts agree too
        for s in (1, 7):
            got_s = ext().gemm_dw(dy, x, s)
            rel = (got_s.float() - ref).abs().max() / ref.abs().max()
            assert rel.item() < 2e-2, f"splits={s}: rel {rel.item()}"


def test_rope_kv_insert_matches_reference():
    """Fused decode-step RoPE+KV-insert (K17 decode fusion) vs fp32 python
    rotation on STRIDED q/k/v views; untouched cache rows stay untouched."""
    from libai_amd.ops._ext import ext
    from libai_amd.ops.rope import _tables

    torch.manual_seed(0)
    B, NH, KVH, HD, MAX = 3, 8, 2, 64, 50
    pos_i = 7
    # strided views carved from a fused-projection-like buffer
    buf = torch.randn(B, 1, NH * HD + 2 * KVH * HD, device="cuda",
                      dtype=torch.bfloat16)
    q = buf[..., : NH * HD].view(B, 1, NH, HD)
    k = buf[..., NH * HD : NH * HD + KVH * HD].view(B, 1, KVH, HD)
    v = buf[..., NH * HD + KVH * HD :].view(B, 1, KVH, HD)
    ck = torch.randn(B, KVH, MAX, HD, device="cuda", dtype=torch.bfloat16)
    cv = torch.randn_like(ck)
    ck0, cv0 = ck.clone(), cv.clone()
    pos = torch.tensor([pos_i], device="cuda", dtype=torch.int64)
    cos_t, sin_t = _tables(MAX, HD, 10000.0, torch.device("cuda"))

    qo = ext().rope_kv_insert(q, k, v, ck, cv, cos_t, sin_t, pos, True)

    def rot(x):  # fp32 rotate_half at pos_i
        cos = cos_t[pos_i].float()
        sin = sin_t[pos_i].float()
        x1, x2 = x[..., : HD // 2].float(), x[..., HD // 2 :].float()
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin],
                         dim=-1).to(torch.bfloat16)

    # rotation compared with tight tolerance (kernel may contract to FMA;
    # 1-ulp fp32 difference can flip the bf16 rounding on exact ties)
    assert torch.allclose(qo.float(), rot(q).permute(0, 2, 1, 3).float(),
                          atol=2e-2, rtol=1e-2)
    assert torch.allclose(ck[:, :, pos_i].float(), rot(k).squeeze(1).float(),
                          atol=2e-2, rtol=1e-2)
    assert torch.equal(cv[:, :, pos_i], v.squeeze(1))
    # every other row untouched
    mask = torch.ones(MAX, dtype=torch.bool)
    mask[pos_i] = False
    assert torch.equal(ck[:, :, mask], ck0[:, :, mask])
    assert torch.equal(cv[:, :, mask], cv0[:, :, mask])

    # no-rotate variant (GPT path): q transposed, k/v copied verbatim
    ck.copy_(ck0)
    cv.copy_(cv0)
    dummy = torch.zeros(1, device="cuda", dtype=torch.float32)
    qo2 = ext().rope_kv_insert(q, k, v, ck, cv, dummy, dummy, pos, False)
    assert torch.equal(qo2, q.permute(0, 2, 1, 3))
    assert torch.equal(ck[:, :, pos_i], k.squeeze(1))
    assert torch.equal(cv[:, :, pos_i], v.squeeze(1))


def test_res_norm_fwd_matches_separate_ops():
    """Fused residual+norm == the bias_dropout_res(p=0) + norm pair it
    replaces in the captured decode step, BITWISE (same fp32 association)."""
    from libai_amd.ops._ext import ext

    torch.manual_seed(0)
    R, H = 32, 1024
    x = torch.randn(R, 1, H, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    res = torch.randn_like(x)
    g = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    b2 = torch.randn(H, device="cuda", dtype=torch.bfloat16)

    h_ref = ext().bias_dropout_res_fwd(x.contiguous(), bias, res.contiguous(),
                                       0.0, 0)
    y_ref, _, _ = ext().ln_fwd(h_ref, g, b2, False, 1e-5)
    h, y = ext().res_norm_fwd(x, bias, res, g, b2, 1e-5, False)
    assert torch.equal(h, h_ref)
    assert torch.equal(y, y_ref)

    # RMS variant, no bias (the Llama decode shape)
    h_ref2 = x + res
    y_ref2, _, _ = ext().ln_fwd(h_ref2.contiguous(), g, None, True, 1e-6)
    h2, y2 = ext().res_norm_fwd(x, None, res, g, None, 1e-6, True)
    assert torch.equal(h2, h_ref2)
    assert torch.equal(y2, y_ref2)
