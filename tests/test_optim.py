import math

import pytest
import torch

from libai_amd.optim import FusedAdamW, get_default_optimizer_params
from libai_amd.scheduler import WarmupCosineLR, WarmupMultiStepLR, WarmupPolynomialLR
from libai_amd.utils import distributed as du

du.setup_dist_util({})


def _tiny_model():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.LayerNorm(16), torch.nn.Linear(16, 4)
    )


def test_fused_adamw_matches_torch_adamw():
    m1, m2 = _tiny_model(), _tiny_model()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)
    opt1 = FusedAdamW(m1.parameters(), lr=1e-2, weight_decay=0.0)
    opt2 = torch.optim.AdamW(m2.parameters(), lr=1e-2, weight_decay=0.0)
    torch.manual_seed(1)
    xs = [torch.randn(4, 8) for _ in range(5)]
    for x in xs:
        opt1.zero_grad()
        m1(x).pow(2).mean().backward()
        opt1.step()
        opt2.zero_grad()
        m2(x).pow(2).mean().backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()


def test_fused_adamw_weight_decay_and_clip():
    m = _tiny_model()
    opt = FusedAdamW(m.parameters(), lr=1e-2, weight_decay=0.1, clip_grad=0.001)
    (m(torch.randn(4, 8)).pow(2).mean() * 1000).backward()
    norm_before = opt.grad_norm().item()
    assert norm_before > 0.001
    opt.step()  # must not blow up params thanks to clip
    assert all(torch.isfinite(p).all() for p in m.parameters())


def test_param_groups_norm_bias_no_decay():
    m = _tiny_model()
    groups = get_default_optimizer_params(
        m, base_lr=1e-3, weight_decay=0.1, weight_decay_norm=0.0, weight_decay_bias=0.0
    )
    by_wd = {}
    for g in groups:
        by_wd.setdefault(g.get("weight_decay"), []).extend(g["params"])
    # 2 linear weights decay; LN weight + 3 biases don't
    assert sum(p.numel() for p in by_wd[0.1]) == 8 * 16 + 16 * 4
    assert 0.0 in by_wd


def test_grad_views_are_persistent():
    m = _tiny_model()
    opt = FusedAdamW(m.parameters(), lr=1e-3)
    _ = opt.buckets
    g0 = next(iter(m.parameters())).grad
    assert g0 is not None  # flat view installed
    m(torch.randn(2, 8)).sum().backward()
    assert next(iter(m.parameters())).grad.data_ptr() == g0.data_ptr()


def test_optimizer_state_roundtrip():
    m = _tiny_model()
    opt = FusedAdamW(m.parameters(), lr=1e-2)
    for _ in range(3):
        opt.zero_grad()
        m(torch.randn(4, 8)).pow(2).mean().backward()
        opt.step()
    state = opt.state_dict()

    m2 = _tiny_model()
    opt2 = FusedAdamW(m2.parameters(), lr=1e-2)
    opt2.load_state_dict(state)
    assert opt2._step == opt._step
    for (_, b1), (_, b2) in zip(opt.buckets, opt2.buckets):
        assert torch.allclose(b1.exp_avg, b2.exp_avg)


def test_warmup_cosine_schedule():
    m = _tiny_model()
    opt = FusedAdamW(m.parameters(), lr=1.0)
    sched = WarmupCosineLR(opt, max_iter=100, warmup_iter=10, warmup_factor=0.001)
    lrs = []
    for _ in range(100):
        lrs.append(opt.param_groups[0]["lr"])
        sched.step()
    assert lrs[0] == pytest.approx(0.001)
    assert lrs[10] == pytest.approx(1.0)
    assert lrs[99] < 0.01
    # monotone decay after warmup
    assert all(a >= b - 1e-9 for a, b in zip(lrs[10:], lrs[11:]))


def test_warmup_multistep_schedule():
    m = _tiny_model()
    opt = FusedAdamW(m.parameters(), lr=1.0)
    sched = WarmupMultiStepLR(opt, milestones=[5, 8], gamma=0.1, warmup_iter=0)
    lrs = []
    for _ in range(10):
        lrs.append(opt.param_groups[0]["lr"])
        sched.step()
    assert lrs[4] == pytest.approx(1.0)
    assert lrs[5] == pytest.approx(0.1)
    assert lrs[8] == pytest.approx(0.01)


def test_warmup_polynomial_schedule():
    m = _tiny_model()
    opt = FusedAdamW(m.parameters(), lr=1.0)
    sched = WarmupPolynomialLR(opt, max_iter=10, warmup_iter=2, power=1.0)
    for _ in range(10):
        sched.step()
    assert opt.param_groups[0]["lr"] == pytest.approx(0.0, abs=1e-6)
