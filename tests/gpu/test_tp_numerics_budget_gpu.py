"""bf16 TP numerics budget on one GPU (SURVEY §4(e) spirit): simulate the
tensor-parallel shard arithmetic — including the bf16 rounding the row
all-reduce introduces — and bound the divergence vs the single-GEMM path.

The gloo oracles prove EXACT fp32 equivalence of the collective wiring;
this quantifies the extra bf16 error a TP run accumulates per layer so the
'same loss within bf16 tolerance' contract has a measured budget behind it.
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _setup():
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    yield


def test_col_row_linear_tp2_bf16_budget():
    torch.manual_seed(0)
    M, H, F = 4096, 1024, 4096
    x = (torch.randn(M, H, device="cuda") / math.sqrt(H)).to(torch.bfloat16)
    w1 = (torch.randn(F, H, device="cuda") * 0.02).to(torch.bfloat16)
    w2 = (torch.randn(H, F, device="cuda") * 0.02).to(torch.bfloat16)

    # single-device reference (one GEMM pair, fp32 accumulate inside)
    ref = torch.matmul(torch.matmul(x, w1.t()), w2.t())

    # TP2 simulation: col shards produce halves; row shards produce partial
    # sums that the all-reduce ADDS IN BF16 — the extra rounding step
    w1a, w1b = w1.chunk(2, dim=0)
    w2a, w2b = w2.chunk(2, dim=1)
    inter_a = torch.matmul(x, w1a.t())
    inter_b = torch.matmul(x, w1b.t())
    part_a = torch.matmul(inter_a, w2a.t())
    part_b = torch.matmul(inter_b, w2b.t())
    tp = part_a + part_b  # bf16 add == the RCCL all-reduce rounding

    err = (tp.float() - ref.float()).abs()
    rel = err.max() / ref.float().abs().max()
    # budget: one bf16 rounding of same-magnitude partials ~ 2^-8 relative
    assert rel.item() < 1.5e-2, f"TP2 bf16 rel err {rel.item()}"


def test_attention_head_split_bf16_budget():
    from libai_amd.ops.attention import flash_attention

    torch.manual_seed(1)
    b, s, h, d = 2, 512, 8, 64
    q = torch.randn(b, s, h, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, s, h, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, s, h, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)
    full = flash_attention(q, k, v, scale, causal=True)
    # TP2 = heads 0-3 and 4-7 computed independently (bit-identical per
    # head: the kernel is head-parallel, so the split must be EXACT)
    ha = flash_attention(q[:, :, :4].contiguous(), k[:, :, :4].contiguous(),
                         v[:, :, :4].contiguous(), scale, causal=True)
    hb = flash_attention(q[:, :, 4:].contiguous(), k[:, :, 4:].contiguous(),
                         v[:, :, 4:].contiguous(), scale, causal=True)
    assert torch.equal(full[:, :, :4], ha)
    assert torch.equal(full[:, :, 4:], hb)


def test_vocab_parallel_ce_tp2_bf16_budget():
    """Vocab-split CE: per-shard max/sumexp combined like the TP kernel."""
    torch.manual_seed(2)
    n, v = 4096, 50304
    logits = (torch.randn(n, v, device="cuda") * 2).to(torch.bfloat16)
    tgt = torch.randint(0, v, (n,), device="cuda")
    ref = torch.nn.functional.cross_entropy(logits.float(), tgt,
                                            reduction="none")
    # shard halves, two-phase reduction in fp32 (what the kernel + TP
    # all-reduces compute)
    la, lb = logits.float().chunk(2, dim=1)
    m = torch.maximum(la.max(1).values, lb.max(1).values)
    se = ((la - m[:, None]).exp().sum(1) + (lb - m[:, None]).exp().sum(1))
    tl = logits.float()[torch.arange(n, device="cuda"), tgt]
    tp = m + se.log() - tl
    assert (tp - ref).abs().max().item() < 1e-4
