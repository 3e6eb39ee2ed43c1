"""ZeRO stage 3 (parameter sharding) vs plain DP on 2-process gloo.

Reference capability: enable_zero stage 3 in the compiled graph
(reference graph_base.py:69-70, tested end-to-end in reference
tests/models/test_gpt.py:186-199).  Here: FSDP-style per-unit
gather/release over FusedAdamW's flat buckets (parallel/zero.py).
"""

import pytest
import torch

from tests.dist_helper import run_dist


def _make_model():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 33), torch.nn.LayerNorm(33), torch.nn.Linear(33, 8)
    )


def _train(rank, model, opt, steps=5, micros=1):
    torch.manual_seed(100 + rank)
    for _ in range(steps):
        opt.zero_grad()
        for _ in range(micros):
            x = torch.randn(4, 16)
            (model(x).pow(2).mean() / micros).backward()
        opt.grad_sync()
        opt.step()


def _flat_params(model, opt=None):
    if opt is not None and hasattr(opt, "materialize_all_params"):
        opt.materialize_all_params()
    flats = torch.cat([p.detach().reshape(-1).clone() for p in model.parameters()])
    if opt is not None and hasattr(opt, "release_all_params"):
        opt.release_all_params()
    return flats


def _plain_worker(rank, world, micros):
    from libai_amd.optim import FusedAdamW
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    model = _make_model()
    opt = FusedAdamW(model.parameters(), lr=1e-2, weight_decay=0.01, clip_grad=1.0)
    _train(rank, model, opt, micros=micros)
    return _flat_params(model)


def _zero3_worker(rank, world, micros):
    from libai_amd.optim import FusedAdamW
    from libai_amd.parallel.zero import setup_zero3
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    model = _make_model()
    opt = FusedAdamW(model.parameters(), lr=1e-2, weight_decay=0.01, clip_grad=1.0)
    setup_zero3(model, opt)
    assert opt._zero_eff == 3

    # params at rest are RELEASED: total live param memory is the 1/dp shards
    for _, b in opt.buckets:
        assert not b.params_live(), "params not released after setup"
        assert b.shard_param.numel() == b.numel // world

    _train(rank, model, opt, micros=micros)

    for _, b in opt.buckets:
        assert not b.params_live(), "params not released after step"
        assert not b.grads_live(), "grads not released after step"

    # an eval forward between steps works (gather + release, no grads)
    model.eval()
    with torch.no_grad():
        model(torch.randn(2, 16))
    model.train()
    for _, b in opt.buckets:
        assert not b.params_live()

    return _flat_params(model, opt)


@pytest.mark.parametrize("micros", [1, 2])
def test_zero3_matches_plain_dp(micros):
    plain = run_dist(_plain_worker, 2, args=(micros,))
    z3 = run_dist(_zero3_worker, 2, args=(micros,))
    assert torch.allclose(z3[0], plain[0], atol=1e-5), (
        (z3[0] - plain[0]).abs().max()
    )
    assert torch.allclose(z3[0], z3[1], atol=1e-6)  # ranks agree


def _zero3_gpt_worker(rank, world, act_ckpt):
    from libai_amd.models import GPTForPreTraining
    from libai_amd.optim import FusedAdamW
    from libai_amd.parallel.zero import setup_zero3
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    tiny = dict(hidden_layers=2, vocab_size=64, hidden_size=32,
                ffn_hidden_size=128, num_attention_heads=4, max_seq_length=32,
                embedding_dropout_prob=0.0, attention_dropout_prob=0.0,
                output_dropout_prob=0.0)
    torch.manual_seed(0)
    model = GPTForPreTraining(**tiny)
    if act_ckpt:
        model.set_activation_checkpoint(True)
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.01)
    if act_ckpt == "ref":
        pass  # plain DP reference
    else:
        setup_zero3(model, opt)

    torch.manual_seed(500 + rank)
    for _ in range(3):
        opt.zero_grad()
        ids = torch.randint(0, 64, (2, 17))
        out = model(input_ids=ids[:, :-1], labels=ids[:, 1:])
        out["lm_loss"].backward()
        opt.grad_sync()
        opt.step()
    return _flat_params(model, opt)


@pytest.mark.parametrize("act_ckpt", [False, True])
def test_zero3_gpt_transformer_units(act_ckpt):
    """GPT with per-TransformerLayer units (+ activation checkpointing
    recompute composes with the gather/release lifecycle)."""
    ref = run_dist(_zero3_gpt_worker, 2, args=("ref",))
    z3 = run_dist(_zero3_gpt_worker, 2, args=(act_ckpt,))
    assert torch.allclose(z3[0], ref[0], atol=1e-5), (
        (z3[0] - ref[0]).abs().max()
    )


def _zero3_ckpt_worker(rank, world, tmpdir):
    import os

    from libai_amd.optim import FusedAdamW
    from libai_amd.parallel.zero import setup_zero3
    from libai_amd.utils import distributed as du
    from libai_amd.utils.checkpoint import Checkpointer

    du.setup_dist_util({})
    model = _make_model()
    opt = FusedAdamW(model.parameters(), lr=1e-2, weight_decay=0.01)
    opt.set_param_names(model.named_parameters())
    setup_zero3(model, opt)
    _train(rank, model, opt, steps=2)
    ck = Checkpointer(model, tmpdir, optimizer=opt)
    ck.save("mid")
    _train(rank, model, opt, steps=2)
    final = _flat_params(model, opt)

    # fresh replica resumes from "mid" and must land on the same weights
    model2 = _make_model()
    with torch.no_grad():  # perturb so the load is observable
        for p in model2.parameters():
            p.add_(1.0)
    opt2 = FusedAdamW(model2.parameters(), lr=1e-2, weight_decay=0.01)
    opt2.set_param_names(model2.named_parameters())
    setup_zero3(model2, opt2)
    ck2 = Checkpointer(model2, tmpdir, optimizer=opt2)
    ck2.load(os.path.join(tmpdir, "mid"))
    _train(rank, model2, opt2, steps=2)
    final2 = _flat_params(model2, opt2)
    assert torch.allclose(final, final2, atol=1e-5), (final - final2).abs().max()
    return True


@pytest.mark.timeout(300)
def test_zero3_checkpoint_roundtrip(tmp_path):
    run_dist(_zero3_ckpt_worker, 2, args=(str(tmp_path),))


def _llama_mem_worker(rank, world):
    """Llama-shaped ZeRO-3 memory accounting: live param bytes at rest are
    ~1/dp of the replicated model (the VERDICT's Llama-shape memory check,
    CPU-sized shapes; the sharding arithmetic is size-independent)."""
    import torch

    from libai_amd.models import LlamaForCausalLM
    from libai_amd.optim import FusedAdamW
    from libai_amd.parallel.zero import setup_zero3
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    model = LlamaForCausalLM(hidden_layers=4, vocab_size=1024, hidden_size=512,
                             intermediate_size=1408, num_attention_heads=8,
                             max_position_embeddings=128)
    total = sum(p.numel() for p in model.parameters())
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    setup_zero3(model, opt)

    def live_param_elems():
        n = 0
        for _, b in opt.buckets:
            n += b.shard_param.numel()
            n += b.flat_param.untyped_storage().size() // b.flat_param.element_size()
        return n

    at_rest = live_param_elems()
    assert at_rest <= total / world * 1.05, (at_rest, total)

    # one step: grads at rest are sharded too; peak full buffers are freed
    ids = torch.randint(0, 1024, (2, 33))
    opt.zero_grad()
    model(input_ids=ids[:, :-1], labels=ids[:, 1:])["lm_loss"].backward()
    opt.grad_sync()
    opt.step()
    assert live_param_elems() <= total / world * 1.05
    grad_live = sum(
        b.shard_grad.numel()
        + b.flat_grad.untyped_storage().size() // b.flat_grad.element_size()
        for _, b in opt.buckets
    )
    assert grad_live <= total / world * 1.05
    # optimizer state (masters + moments) is sharded by construction
    state = sum(b.flat_master.numel() + b.exp_avg.numel() + b.exp_avg_sq.numel()
                for _, b in opt.buckets)
    assert state <= 3 * total / world * 1.05
    return True


def test_zero3_llama_shape_memory():
    run_dist(_llama_mem_worker, 2)
