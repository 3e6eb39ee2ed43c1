"""End-to-end convergence: the full 345M stack must memorize a repeating
pattern (exercises flash attention + all fused kernels + AdamW + clip +
scheduler gradients at the real benchmark shape)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpt2_345m_memorizes_pattern():
    from libai_amd.models import GPTForPreTraining
    from libai_amd.optim import FusedAdamW, get_default_optimizer_params
    from libai_amd.scheduler import WarmupCosineLR
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    m = GPTForPreTraining(
        hidden_layers=24, vocab_size=50304, hidden_size=1024,
        ffn_hidden_size=4096, num_attention_heads=16, max_seq_length=1024,
        embedding_dropout_prob=0.1, attention_dropout_prob=0.1,
        output_dropout_prob=0.1,
    ).to(torch.bfloat16).cuda()
    opt = FusedAdamW(get_default_optimizer_params(m, base_lr=3e-4), lr=3e-4,
                     weight_decay=0.01, clip_grad=1.0)
    sched = WarmupCosineLR(opt, max_iter=60, warmup_iter=10)
    g = torch.Generator().manual_seed(7)
    base = torch.randint(0, 256, (64,), generator=g)
    batch = base.repeat(17)[:1025].unsqueeze(0).repeat(16, 1).cuda()
    losses = []
    for _ in range(60):
        opt.zero_grad()
        out = m(input_ids=batch[:, :-1], labels=batch[:, 1:])
        out["lm_loss"].backward()
        opt.step()
        sched.step()
        losses.append(float(out["lm_loss"]))
    assert all(l == l for l in losses), "NaN loss"
    assert losses[-1] < 0.2, f"345M failed to memorize: {losses[::10]}"
