import torch

from libai_amd.lora import apply_lora, merge_lora
from libai_amd.models import GPTForPreTraining
from libai_amd.utils import distributed as du

du.setup_dist_util({})

TINY = dict(hidden_layers=2, vocab_size=64, hidden_size=32, ffn_hidden_size=128,
            num_attention_heads=4, max_seq_length=32,
            embedding_dropout_prob=0.0, attention_dropout_prob=0.0,
            output_dropout_prob=0.0)


def test_lora_freezes_base_and_trains_adapters():
    torch.manual_seed(0)
    m = GPTForPreTraining(**TINY)
    apply_lora(m, r=4, alpha=8)
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    assert trainable and all("lora_" in n for n in trainable)
    n_trainable = sum(p.numel() for p in m.parameters() if p.requires_grad)
    n_total = sum(p.numel() for p in m.parameters())
    assert n_trainable < 0.2 * n_total

    ids = torch.randint(0, 64, (2, 17))
    out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
    out["lm_loss"].backward()
    grads = [p.grad for n, p in m.named_parameters() if "lora_" in n]
    assert all(g is not None for g in grads)


def test_lora_zero_init_is_identity_and_merge_matches():
    torch.manual_seed(0)
    m = GPTForPreTraining(**TINY).eval()
    ids = torch.randint(0, 64, (1, 16))
    with torch.no_grad():
        before = m(input_ids=ids)["prediction_scores"]
    apply_lora(m, r=4, alpha=8)
    m.eval()
    with torch.no_grad():
        after = m(input_ids=ids)["prediction_scores"]
    assert torch.allclose(before, after, atol=1e-5)  # B=0 -> identity

    # train a step so adapters are nonzero, then merge must preserve outputs
    for p in m.parameters():
        if p.requires_grad:
            torch.nn.init.normal_(p, std=0.02)
    m.eval()
    with torch.no_grad():
        lora_out = m(input_ids=ids)["prediction_scores"]
    merge_lora(m)
    with torch.no_grad():
        merged_out = m(input_ids=ids)["prediction_scores"]
    assert torch.allclose(lora_out, merged_out, atol=1e-4)
