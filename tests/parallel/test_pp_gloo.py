"""1F1B pipeline correctness on 2-process gloo CPU: loss and grads must match
a single-process run of the same model (same seed => same init)."""

import pytest
import torch

from tests.dist_helper import run_dist

MODEL_KW = dict(
    hidden_layers=4,
    vocab_size=128,
    hidden_size=32,
    ffn_hidden_size=128,
    num_attention_heads=4,
    max_seq_length=32,
    embedding_dropout_prob=0.0,
    attention_dropout_prob=0.0,
    output_dropout_prob=0.0,
)


def _single_process_reference(num_micro):
    from libai_amd.models import GPTForPreTraining
    from libai_amd.utils import distributed as du

    du._DIST_UTIL = None
    du.setup_dist_util({})
    torch.manual_seed(123)
    model = GPTForPreTraining(**MODEL_KW)
    torch.manual_seed(99)
    losses = []
    for i in range(num_micro):
        ids = torch.randint(0, 128, (2, 33))
        out = model(input_ids=ids[:, :-1], labels=ids[:, 1:])
        loss = out["lm_loss"] / num_micro
        loss.backward()
        losses.append(float(out["lm_loss"]))
    grads = {
        n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None
    }
    return losses, grads


def _pp2_worker(rank, world, num_micro):
    import torch

    from libai_amd.models import GPTForPreTraining
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    model = GPTForPreTraining(**MODEL_KW)
    model.hidden_size = 32
    sched = PipelineScheduler(model, dtype=torch.float32)

    torch.manual_seed(99)
    batches = []
    for _ in range(num_micro):
        ids = torch.randint(0, 128, (2, 33))
        batches.append({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    loss_dict = sched.run_1f1b(batches)
    grads = {
        n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None
    }
    loss = float(loss_dict["lm_loss"]) if loss_dict else None
    return loss, {k: v for k, v in grads.items()}


@pytest.mark.parametrize("num_micro", [1, 4])
def test_pp2_matches_single_process(num_micro):
    ref_losses, ref_grads = _single_process_reference(num_micro)
    results = run_dist(_pp2_worker, 2, args=(num_micro,))
    pp_loss = results[1][0]  # last stage
    assert pp_loss == pytest.approx(sum(ref_losses) / num_micro, abs=1e-4)

    # gradient parity: each stage's local grads match the reference's
    seen = set()
    for rank, (loss, grads) in enumerate(results):
        for name, g in grads.items():
            assert name in ref_grads, f"unexpected grad {name} on rank {rank}"
            ref = ref_grads[name]
            assert torch.allclose(g, ref, atol=1e-4), (
                f"grad mismatch {name} on rank {rank}: "
                f"max diff {(g - ref).abs().max()}"
            )
            seen.add(name)
    assert seen == set(ref_grads.keys()), (
        f"missing grads: {set(ref_grads) - seen}"
    )


def test_pipeline_stage_batch_keys():
    """Middle stages consume no batch tensors; first/last declare theirs."""
    from libai_amd.models import GPTForPreTraining

    m = GPTForPreTraining(**MODEL_KW)
    assert m.pipeline_stage_batch_keys(True, False) == {"input_ids"}
    assert m.pipeline_stage_batch_keys(False, True) == {"labels"}
    assert m.pipeline_stage_batch_keys(False, False) == set()
