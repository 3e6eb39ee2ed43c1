"""Reproducibility contract (SURVEY §5 race-detection row): same seed =>
bit-identical losses across runs, and distinct DP seeds diverge dropout
while TP-group seeds agree."""

import torch

from libai_amd.models import GPTForPreTraining
from libai_amd.optim import FusedAdamW
from libai_amd.utils import distributed as du

du.setup_dist_util({})

KW = dict(hidden_layers=2, vocab_size=128, hidden_size=32, ffn_hidden_size=64,
          num_attention_heads=4, max_seq_length=32,
          embedding_dropout_prob=0.1, attention_dropout_prob=0.1,
          output_dropout_prob=0.1)


def _run(seed):
    torch.manual_seed(seed)
    m = GPTForPreTraining(**KW)
    opt = FusedAdamW(m.parameters(), lr=1e-3, clip_grad=1.0)
    g = torch.Generator().manual_seed(7)
    losses = []
    for _ in range(3):
        opt.zero_grad()
        ids = torch.randint(0, 128, (2, 17), generator=g)
        out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
        out["lm_loss"].backward()
        opt.step()
        losses.append(out["lm_loss"].item())
    return losses


def test_same_seed_same_losses():
    assert _run(1234) == _run(1234)


def test_different_seed_differs():
    assert _run(1234) != _run(999)


def test_tp_group_seed_policy():
    # same (pp, dp) coordinate -> same seed regardless of tp rank
    s = du.same_seed_for_tp_group(1234)
    assert isinstance(s, int)
