"""lm-eval harness bridge: loglikelihood math vs manual computation
(reference capability: projects/Eval_LLM/eval_harness.py)."""

import pytest
import torch
import torch.nn.functional as F

from libai_amd.inference.eval_harness import LibaiEvalHarnessLM
from libai_amd.models import GPTForPreTraining
from libai_amd.utils import distributed as du

du.setup_dist_util({})


class ToyTokenizer:
    """Byte-level: token = byte value (vocab 256); eod = 0."""

    eod = 0

    def encode(self, text):
        return list(text.encode("utf-8"))

    def decode(self, ids):
        return bytes(int(i) % 256 for i in ids).decode("utf-8", errors="ignore")


@pytest.fixture(scope="module")
def lm():
    torch.manual_seed(0)
    model = GPTForPreTraining(
        hidden_layers=2, vocab_size=256, hidden_size=32, ffn_hidden_size=64,
        num_attention_heads=4, max_seq_length=128, embedding_dropout_prob=0.0,
        attention_dropout_prob=0.0, output_dropout_prob=0.0,
    ).eval()
    return LibaiEvalHarnessLM(model, ToyTokenizer(), max_length=64,
                              device=torch.device("cpu"))


def test_loglikelihood_matches_manual(lm):
    ctx, cont = "hello ", "world"
    (ll, greedy), = lm.loglikelihood([(ctx, cont)])
    # manual: logprob of cont tokens under the model given ctx
    ids = lm._encode(ctx) + lm._encode(cont)
    inp = torch.tensor([ids[:-1]])
    with torch.no_grad():
        logits = lm.model(input_ids=inp)["prediction_scores"].float()
    n = len(lm._encode(cont))
    lp = F.log_softmax(logits[0, -n:], dim=-1)
    want = sum(lp[i, t] for i, t in enumerate(ids[-n:]))
    assert ll == pytest.approx(float(want), abs=1e-4)
    assert isinstance(greedy, bool)


def test_loglikelihood_additivity(lm):
    """ll(ctx, a+b) == ll(ctx, a) + ll(ctx+a, b) for an autoregressive LM."""
    (whole, _), = lm.loglikelihood([("ab", "cdef")])
    (p1, _), = lm.loglikelihood([("ab", "cd")])
    (p2, _), = lm.loglikelihood([("abcd", "ef")])
    assert whole == pytest.approx(p1 + p2, abs=1e-3)


def test_loglikelihood_rolling_and_empty_context(lm):
    (roll,) = lm.loglikelihood_rolling([("hello",)])
    assert roll == roll and roll < 0  # finite negative logprob
    (ll, _), = lm.loglikelihood([("", "hi")])  # empty ctx -> eot context
    assert ll == ll


def test_generate_until_stops(lm):
    (text,) = lm.generate_until([("abc", {"max_gen_toks": 8, "until": []})])
    assert isinstance(text, str)
    # stop-sequence cut
    (cut,) = lm.generate_until([("abc", {"max_gen_toks": 8,
                                         "until": [text[1:2]] if len(text) > 1 else []})])
    assert isinstance(cut, str)
