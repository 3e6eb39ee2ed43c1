"""fp8 e4m3 forward GEMMs on gfx950: numerics budget + training sanity."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _setup():
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    yield


def test_fp8_linear_numerics():
    from libai_amd.ops import fp8

    if not fp8.fp8_available():
        pytest.skip("no fp8 _scaled_mm on this stack")
    torch.manual_seed(0)
    x = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(128, 256, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.randn(128, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = fp8.fp8_linear(x, w, b)
    ref = torch.nn.functional.linear(x.float(), w.float(), b.float())
    rel = (y.float() - ref).abs().max() / ref.abs().max()
    assert rel.item() < 0.05, rel.item()  # e4m3 fwd rounding budget

    # backward is EXACT bf16 (same GEMMs as F.linear's backward)
    dy = torch.randn_like(y)
    y.backward(dy)
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    torch.nn.functional.linear(x2, w2, b2).backward(dy)
    assert torch.equal(x.grad, x2.grad)
    assert torch.equal(w.grad, w2.grad)
    assert torch.equal(b.grad, b2.grad)


def test_fp8_gpt_training_sanity():
    """Tiny GPT: 30 fp8 steps reduce the loss comparably to bf16 steps."""
    from libai_amd.models.gpt_model import GPTForPreTraining
    from libai_amd.ops import fp8

    if not fp8.fp8_available():
        pytest.skip("no fp8 _scaled_mm on this stack")

    def run(enabled):
        torch.manual_seed(0)
        m = GPTForPreTraining(
            hidden_layers=2, vocab_size=1024, hidden_size=256,
            ffn_hidden_size=1024, num_attention_heads=4, max_seq_length=128,
            embedding_dropout_prob=0.0, attention_dropout_prob=0.0,
            output_dropout_prob=0.0,
        ).to("cuda", torch.bfloat16)
        opt = torch.optim.SGD(m.parameters(), lr=0.05)
        torch.manual_seed(1)
        ids = torch.randint(0, 1024, (4, 65), device="cuda")
        fp8.set_fp8_gemms(enabled)
        try:
            losses = []
            for _ in range(30):
                loss = m(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"]
                opt.zero_grad()
                loss.backward()
                opt.step()
                losses.append(float(loss))
        finally:
            fp8.set_fp8_gemms(False)
        return losses

    bf16 = run(False)
    f8 = run(True)
    assert f8[-1] < 0.7 * f8[0], f8  # learning happens
    assert abs(f8[-1] - bf16[-1]) < 0.25 * bf16[0], (f8[-1], bf16[-1])


def test_quant_fp8_kernel_numerics():
    """Fused single-pass quantize: dequant error within e4m3 budget, amax
    feeds the next scale (delayed scaling)."""
    from libai_amd.ops import fp8

    if not fp8.fp8_available():
        pytest.skip("no fp8 on this stack")
    torch.manual_seed(0)
    x = torch.randn(4096, 1024, device="cuda", dtype=torch.bfloat16) * 3
    st = fp8.DelayedScale()
    st.quant(x)            # bootstrap (dynamic pass seeds the scale)
    t8, used = st.quant(x)  # fused kernel path
    deq = t8.float() * used
    xf = x.float()
    # e4m3: 3 mantissa bits -> relative error <= 2^-4 per element
    rel = ((deq - xf).abs() / xf.abs().clamp(min=1e-3)).max()
    assert rel.item() < 0.08, rel.item()
    # the observed amax became the next scale
    expect = (xf.abs().max() / 448.0).item()
    assert abs(st.scale.item() - expect) / expect < 1e-3

    # saturation: values that outgrow the stale scale clamp to +-448*scale
    x2 = x * 100
    t8b, used_b = st.quant(x2)
    assert used_b.item() == pytest.approx(st_prev_scale_check(st, expect),
                                          rel=1e-3)
    assert t8b.float().abs().max().item() <= 448.0
    # and the NEXT scale caught up to the new amax
    assert st.scale.item() == pytest.approx(
        (x2.float().abs().max() / 448.0).item(), rel=1e-3)


def st_prev_scale_check(st, expect):
    return expect  # used_b was the pre-update scale (== expect)
