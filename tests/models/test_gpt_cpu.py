import tempfile

import torch

from libai_amd.models import GPTForPreTraining
from libai_amd.optim import FusedAdamW, get_default_optimizer_params
from libai_amd.utils import distributed as du

du.setup_dist_util({})

TINY = dict(
    hidden_layers=2,
    vocab_size=64,
    hidden_size=32,
    ffn_hidden_size=128,
    num_attention_heads=4,
    max_seq_length=32,
    embedding_dropout_prob=0.0,
    attention_dropout_prob=0.0,
    output_dropout_prob=0.0,
)


def test_gpt_forward_shapes():
    model = GPTForPreTraining(**TINY)
    ids = torch.randint(0, 64, (2, 16))
    out = model(input_ids=ids)
    assert out["prediction_scores"].shape == (2, 16, 64)
    out = model(input_ids=ids, labels=ids)
    assert out["lm_loss"].ndim == 0


def test_gpt_loss_decreases_on_learnable_data():
    torch.manual_seed(0)
    model = GPTForPreTraining(**TINY)
    opt = FusedAdamW(get_default_optimizer_params(model, base_lr=1e-2), lr=1e-2)
    # learnable pattern: constant repeated sequence
    ids = torch.arange(17).remainder(8).unsqueeze(0).repeat(4, 1)
    first = last = None
    for i in range(30):
        opt.zero_grad()
        out = model(input_ids=ids[:, :-1], labels=ids[:, 1:])
        out["lm_loss"].backward()
        opt.step()
        v = float(out["lm_loss"])
        first = first if first is not None else v
        last = v
    assert last < first * 0.5, f"loss did not decrease: {first} -> {last}"


def test_gpt_kv_cache_generation_consistent():
    torch.manual_seed(0)
    model = GPTForPreTraining(**TINY)
    model.eval()
    gpt = model.GPT_model
    ids = torch.randint(0, 64, (1, 10))
    with torch.no_grad():
        full = gpt(ids)
        logits1, past = gpt(ids[:, :9], use_cache=True)
        logits2, _ = gpt(ids[:, 9:10], past_key_values=past, use_cache=True)
    assert torch.allclose(full[:, :9], logits1, atol=1e-4)
    assert torch.allclose(full[:, 9:10], logits2, atol=1e-4)


def test_checkpoint_save_resume_roundtrip():
    from libai_amd.utils.checkpoint import Checkpointer

    torch.manual_seed(0)
    model = GPTForPreTraining(**TINY)
    opt = FusedAdamW(get_default_optimizer_params(model, base_lr=1e-3), lr=1e-3)
    ids = torch.randint(0, 64, (2, 17))
    for _ in range(2):
        opt.zero_grad()
        model(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"].backward()
        opt.step()

    with tempfile.TemporaryDirectory() as d:
        ck = Checkpointer(model, d, optimizer=opt)
        ck.save("model_0000002", iteration=2)
        assert ck.has_checkpoint()

        torch.manual_seed(123)
        model2 = GPTForPreTraining(**TINY)
        opt2 = FusedAdamW(get_default_optimizer_params(model2, base_lr=1e-3), lr=1e-3)
        ck2 = Checkpointer(model2, d, optimizer=opt2)
        extra = ck2.resume_or_load("", resume=True)
        assert extra["iteration"] == 2
        for (n1, p1), (n2, p2) in zip(
            model.named_parameters(), model2.named_parameters()
        ):
            assert n1 == n2
            assert torch.allclose(p1, p2), f"param {n1} not restored"
        # same forward after restore
        out1 = model(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"]
        out2 = model2(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"]
        assert torch.allclose(out1, out2, atol=1e-6)


def test_activation_checkpoint_same_grads():
    torch.manual_seed(0)
    m1 = GPTForPreTraining(**TINY)
    torch.manual_seed(0)
    m2 = GPTForPreTraining(**TINY)
    m2.set_activation_checkpoint(True)
    ids = torch.randint(0, 64, (2, 17))
    m1(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"].backward()
    m2(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"].backward()
    for (n, p1), (_, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-6), f"grad mismatch {n}"
