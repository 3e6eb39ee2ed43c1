"""Task pipelines end-to-end on CPU (reference: tests/inference/*).

Covers the preprocess/forward/postprocess protocol of all three task
pipelines with tiny random-init models and a word-level test tokenizer.
"""

import pytest
import torch

from libai_amd.inference.basic import (
    ImageClassificationPipeline,
    TextClassificationPipeline,
    TextGenerationPipeline,
)


class _ToyTok:
    """Minimal tokenizer satisfying the pipeline protocol."""

    pad_token = "<pad>"
    eos_token = "<eos>"

    def __init__(self, vocab_size=128):
        self.vocab_size = vocab_size

    def encode(self, text, add_special_tokens=False):
        ids = [3 + (hash(w) % (self.vocab_size - 4)) for w in text.split()]
        return ids or [3]

    def decode(self, ids, skip_special_tokens=True):
        return " ".join(f"tok{i}" for i in ids if not skip_special_tokens or i > 2)

    def convert_tokens_to_ids(self, tok):
        return {"<pad>": 0, "<eos>": 1}.get(tok, 2)


@pytest.fixture(scope="module")
def _dist():
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    yield


def test_text_generation_pipeline(_dist):
    from libai_amd.models import GPTForPreTraining

    torch.manual_seed(0)
    m = GPTForPreTraining(
        hidden_layers=2, vocab_size=128, hidden_size=64, ffn_hidden_size=128,
        num_attention_heads=4, max_seq_length=64,
    )
    pipe = TextGenerationPipeline(model=m, tokenizer=_ToyTok())
    out = pipe("hello world", max_length=12)
    assert isinstance(out, list) and "generated_text" in out[0]
    # deterministic greedy
    out2 = pipe("hello world", max_length=12)
    assert out == out2


def test_text_classification_pipeline(_dist):
    from libai_amd.models import BertForPreTraining

    torch.manual_seed(0)
    m = BertForPreTraining(
        vocab_size=128, hidden_size=64, hidden_layers=2, num_attention_heads=4,
        intermediate_size=128, max_position_embeddings=64,
    )
    pipe = TextClassificationPipeline(model=m, tokenizer=_ToyTok())
    out = pipe(["good movie", "bad movie with more words"])
    assert len(out) == 2
    for r in out:
        assert set(r) == {"label", "score"} and 0.0 <= r["score"] <= 1.0


def test_image_classification_pipeline(_dist):
    from libai_amd.models import VisionTransformer

    torch.manual_seed(0)
    m = VisionTransformer(img_size=32, patch_size=8, embed_dim=64, depth=2,
                          num_heads=4, num_classes=10)
    pipe = ImageClassificationPipeline(model=m, tokenizer=None)
    out = pipe(torch.randn(2, 3, 32, 32), topk=3)
    assert len(out) == 2
    assert len(out[0]["classes"]) == 3
    assert out[0]["scores"] == sorted(out[0]["scores"], reverse=True)
