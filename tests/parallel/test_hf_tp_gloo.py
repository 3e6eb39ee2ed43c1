"""TP-sharded HF model == unsharded HF model (2-process gloo)."""

import pytest
import torch

from tests.dist_helper import run_dist

transformers = pytest.importorskip("transformers")


def _worker(rank, world):
    import torch
    from transformers import LlamaConfig, LlamaForCausalLM

    from libai_amd.models.utils.hf_tp import tp_shard_hf_model
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=4, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg).eval()
    torch.manual_seed(0)
    ref = LlamaForCausalLM(cfg).eval()

    ids = torch.randint(0, 128, (2, 10))
    with torch.no_grad():
        expected = ref(input_ids=ids).logits

    tp_shard_hf_model(model)
    with torch.no_grad():
        got = model(input_ids=ids).logits
    assert torch.allclose(got, expected, atol=1e-4), (
        f"TP-sharded HF llama mismatch: {(got - expected).abs().max()}"
    )
    return True


def test_hf_llama_tp2_equivalence():
    assert all(run_dist(_worker, 2))
