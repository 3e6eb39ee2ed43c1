"""RoPE / SwiGLU HIP kernels vs fp32 torch references, plus Llama GPU smoke."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _setup():
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    yield


def _rel(a, b):
    return ((a.float() - b.float()).abs().max() / (b.float().abs().max() + 1e-6)).item()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("hs", [64, 128])
def test_rope_matches_reference(dtype, hs):
    from libai_amd.ops.rope import _ref_rope, _tables, apply_rotary_pos_emb

    torch.manual_seed(0)
    x = torch.randn(2, 33, 4, hs, device="cuda", dtype=dtype, requires_grad=True)
    y = apply_rotary_pos_emb(x, max_seq=64, pos0=3)
    cos_t, sin_t = _tables(64, hs, 10000.0, x.device)
    ref = _ref_rope(x.detach().float(), cos_t, sin_t, 3)
    tol = 1e-5 if dtype == torch.float32 else 1e-2
    assert _rel(y, ref) < tol
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().float().requires_grad_(True)
    _ref_rope(xr, cos_t, sin_t, 3).backward(g.float())
    assert _rel(x.grad, xr.grad) < tol * 3


def test_rope_strided_view_input():
    from libai_amd.ops.rope import apply_rotary_pos_emb

    torch.manual_seed(0)
    qkv = torch.randn(2, 16, 4, 3, 64, device="cuda", dtype=torch.bfloat16)
    q_view = qkv[..., 0, :]
    y1 = apply_rotary_pos_emb(q_view, max_seq=32)
    y2 = apply_rotary_pos_emb(q_view.contiguous(), max_seq=32)
    assert torch.equal(y1, y2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_swiglu_matches_reference(dtype):
    from libai_amd.ops.swiglu import swiglu

    torch.manual_seed(0)
    x = torch.randn(64, 256, device="cuda", dtype=dtype, requires_grad=True)
    y = swiglu(x)
    xr = x.detach().float().requires_grad_(True)
    g, u = xr.chunk(2, -1)
    ref = torch.nn.functional.silu(g) * u
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert _rel(y, ref) < tol
    gr = torch.randn_like(y)
    y.backward(gr)
    ref.backward(gr.float())
    assert _rel(x.grad, xr.grad) < tol * 3


def test_llama_gpu_train_step_and_flash128():
    from libai_amd.models import LlamaForCausalLM
    from libai_amd.optim import FusedAdamW, get_default_optimizer_params

    torch.manual_seed(0)
    m = LlamaForCausalLM(
        hidden_layers=2, vocab_size=1024, hidden_size=512, intermediate_size=1024,
        num_attention_heads=4, max_position_embeddings=256,  # head_dim 128 -> flash
    ).to(torch.bfloat16).cuda()
    opt = FusedAdamW(get_default_optimizer_params(m, base_lr=1e-3), lr=1e-3,
                     clip_grad=1.0)
    ids = torch.arange(257, device="cuda").remainder(64).unsqueeze(0).repeat(4, 1)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
        out["lm_loss"].backward()
        opt.step()
        losses.append(float(out["lm_loss"]))
    assert losses[-1] < losses[0] * 0.8, f"no learning: {losses}"


def test_llama_gpu_flash_matches_unfused():
    from libai_amd.models.llama import LlamaAttention
    from libai_amd.models.utils.weight_init import init_method_normal

    torch.manual_seed(0)
    attn = LlamaAttention(512, 4, 256, init_method_normal(0.02),
                          init_method_normal(0.02)).to(torch.bfloat16).cuda().eval()
    x = torch.randn(2, 128, 512, device="cuda", dtype=torch.bfloat16)
    y_flash = attn(x)
    y_unfused, _ = attn(x, use_cache=True)
    rel = (y_flash.float() - y_unfused.float()).abs().max() / (
        y_unfused.float().abs().max() + 1e-6
    )
    assert rel < 3e-2, f"flash vs unfused llama attention: {rel}"
