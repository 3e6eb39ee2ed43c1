"""Pipeline parallelism for T5 (tuple activation boundary) and RoBERTa, plus
eval under PP — 2-process gloo vs single-process reference.

Reference capability: t5_model.py:450+ / roberta_model.py assign pipeline
stage ids; evaluator.py:119-278 evaluates pipelined models.
"""

import pytest
import torch

from tests.dist_helper import run_dist

T5_KW = dict(
    vocab_size=128,
    hidden_size=32,
    hidden_layers=2,   # 2 enc + 2 dec -> pipeline_num_layers=4
    num_attention_heads=4,
    intermediate_size=128,
    hidden_dropout_prob=0.0,
    attention_probs_dropout_prob=0.0,
    embedding_dropout_prob=0.0,
    max_position_embeddings=64,
)

ROBERTA_KW = dict(
    vocab_size=128,
    hidden_size=32,
    hidden_layers=4,
    num_attention_heads=4,
    intermediate_size=128,
    hidden_dropout_prob=0.0,
    attention_probs_dropout_prob=0.0,
    max_position_embeddings=64,
    add_pooling_layer=False,
)


def _t5_batches(num_micro):
    torch.manual_seed(99)
    batches = []
    for _ in range(num_micro):
        enc = torch.randint(3, 128, (2, 16))
        dec = torch.randint(3, 128, (2, 12))
        labels = torch.randint(0, 128, (2, 12))
        enc_mask = (torch.arange(16)[None, :] < torch.tensor([[16], [9]])).long()
        loss_mask = torch.ones(2, 12, dtype=torch.long)
        batches.append({
            "encoder_input_ids": enc, "decoder_input_ids": dec,
            "encoder_attn_mask": enc_mask, "lm_labels": labels,
            "loss_mask": loss_mask,
        })
    return batches


def _t5_reference(num_micro):
    from libai_amd.models.t5_model import T5ForPreTraining
    from libai_amd.utils import distributed as du

    du._DIST_UTIL = None
    du.setup_dist_util({})
    torch.manual_seed(123)
    model = T5ForPreTraining(**T5_KW)
    losses = []
    for b in _t5_batches(num_micro):
        out = model(
            encoder_input_ids=b["encoder_input_ids"],
            decoder_input_ids=b["decoder_input_ids"],
            encoder_attn_mask=b["encoder_attn_mask"],
            lm_labels=b["lm_labels"], loss_mask=b["loss_mask"],
        )
        (out["masked_lm_loss"] / num_micro).backward()
        losses.append(float(out["masked_lm_loss"]))
    grads = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    return losses, grads


def _t5_pp2_worker(rank, world, num_micro):
    import torch

    from libai_amd.models.t5_model import T5ForPreTraining
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    # stage 0 = encoder (layers 0-1), stage 1 = decoder (layers 2-3) + head
    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    model = T5ForPreTraining(**T5_KW)
    sched = PipelineScheduler(model, dtype=torch.float32)
    loss_dict = sched.run_1f1b(_t5_batches(num_micro))
    grads = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    loss = float(loss_dict["masked_lm_loss"]) if loss_dict else None
    return loss, grads


@pytest.mark.parametrize("num_micro", [1, 4])
def test_t5_pp2_matches_single_process(num_micro):
    ref_losses, ref_grads = _t5_reference(num_micro)
    results = run_dist(_t5_pp2_worker, 2, args=(num_micro,))
    pp_loss = results[1][0]
    assert pp_loss == pytest.approx(sum(ref_losses) / num_micro, abs=1e-4)
    seen = set()
    for rank, (loss, grads) in enumerate(results):
        for name, g in grads.items():
            assert name in ref_grads, name
            assert torch.allclose(g, ref_grads[name], atol=1e-4), (
                f"grad mismatch {name} (rank {rank}): "
                f"{(g - ref_grads[name]).abs().max()}"
            )
            seen.add(name)
    missing = set(ref_grads) - seen
    assert not missing, f"params with no grads anywhere: {missing}"


def _roberta_batches(num_micro):
    torch.manual_seed(77)
    batches = []
    for _ in range(num_micro):
        ids = torch.randint(3, 128, (2, 16))
        mask = (torch.arange(16)[None, :] < torch.tensor([[16], [11]])).long()
        labels = torch.randint(0, 128, (2, 16))
        loss_mask = (torch.rand(2, 16) < 0.3).long()
        loss_mask[:, 0] = 1
        batches.append({"input_ids": ids, "attention_mask": mask,
                        "lm_labels": labels, "loss_mask": loss_mask})
    return batches


def _roberta_reference(num_micro):
    from libai_amd.models.roberta_model import RobertaForPreTraining
    from libai_amd.utils import distributed as du

    du._DIST_UTIL = None
    du.setup_dist_util({})
    torch.manual_seed(123)
    model = RobertaForPreTraining(**ROBERTA_KW)
    losses = []
    for b in _roberta_batches(num_micro):
        out = model(**b)
        (out["lm_loss"] / num_micro).backward()
        losses.append(float(out["lm_loss"]))
    grads = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    return losses, grads


def _roberta_pp2_worker(rank, world, num_micro):
    import torch

    from libai_amd.models.roberta_model import RobertaForPreTraining
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    model = RobertaForPreTraining(**ROBERTA_KW)
    model.hidden_size = 32
    sched = PipelineScheduler(model, dtype=torch.float32)
    loss_dict = sched.run_1f1b(_roberta_batches(num_micro))
    grads = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    loss = float(loss_dict["lm_loss"]) if loss_dict else None
    return loss, grads


def test_roberta_pp2_matches_single_process():
    num_micro = 2
    ref_losses, ref_grads = _roberta_reference(num_micro)
    results = run_dist(_roberta_pp2_worker, 2, args=(num_micro,))
    pp_loss = results[1][0]
    assert pp_loss == pytest.approx(sum(ref_losses) / num_micro, abs=1e-4)
    for rank, (loss, grads) in enumerate(results):
        for name, g in grads.items():
            assert torch.allclose(g, ref_grads[name], atol=1e-4), (
                f"grad mismatch {name} (rank {rank})"
            )


def _eval_pp2_worker(rank, world):
    import torch

    from libai_amd.evaluation import inference_on_dataset
    from libai_amd.evaluation.ppl_evaluator import PPLEvaluator
    from libai_amd.models import GPTForPreTraining
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    kw = dict(hidden_layers=4, vocab_size=128, hidden_size=32,
              ffn_hidden_size=128, num_attention_heads=4, max_seq_length=32,
              embedding_dropout_prob=0.0, attention_dropout_prob=0.0,
              output_dropout_prob=0.0)
    model = GPTForPreTraining(**kw)
    model.hidden_size = 32
    sched = PipelineScheduler(model, dtype=torch.float32)

    torch.manual_seed(5)
    data = []
    for _ in range(4):
        ids = torch.randint(0, 128, (2, 33))
        data.append({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    res = inference_on_dataset(model, data, PPLEvaluator(),
                               pipeline_scheduler=sched)
    return res


def test_eval_under_pp2():
    """inference_on_dataset runs through run_eval; both ranks get results."""
    results = run_dist(_eval_pp2_worker, 2)
    assert results[0] == results[1]
    assert results[0], "empty eval results"
    for v in results[0].values():
        assert v == v, "NaN metric"


def _bloom_pp2_worker(rank, world):
    import torch

    from libai_amd.models import BloomForCausalLM
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    model = BloomForCausalLM(vocab_size=128, hidden_size=32, hidden_layers=4,
                             num_attention_heads=4)
    model.hidden_size = 32
    sched = PipelineScheduler(model, dtype=torch.float32)
    torch.manual_seed(7)
    batches = []
    for _ in range(2):
        ids = torch.randint(0, 128, (2, 17))
        batches.append({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    loss_dict = sched.run_1f1b(batches)
    grads = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    return (float(loss_dict["lm_loss"]) if loss_dict else None), grads


def test_bloom_pp2_matches_single_process():
    from libai_amd.models import BloomForCausalLM
    from libai_amd.utils import distributed as du

    du._DIST_UTIL = None
    du.setup_dist_util({})
    torch.manual_seed(123)
    model = BloomForCausalLM(vocab_size=128, hidden_size=32, hidden_layers=4,
                             num_attention_heads=4)
    torch.manual_seed(7)
    losses = []
    for _ in range(2):
        ids = torch.randint(0, 128, (2, 17))
        out = model(input_ids=ids[:, :-1], labels=ids[:, 1:])
        (out["lm_loss"] / 2).backward()
        losses.append(float(out["lm_loss"]))
    ref_grads = {n: p.grad.clone() for n, p in model.named_parameters()
                 if p.grad is not None}
    results = run_dist(_bloom_pp2_worker, 2)
    assert results[1][0] == pytest.approx(sum(losses) / 2, abs=1e-4)
    for rank, (loss, grads) in enumerate(results):
        for name, g in grads.items():
            assert torch.allclose(g, ref_grads[name], atol=1e-4), \
                f"grad mismatch {name} (rank {rank})"


def _vit_pp2_worker(rank, world):
    import torch

    from libai_amd.engine.trainer import EagerTrainer
    from libai_amd.models import VisionTransformer
    from libai_amd.optim import FusedAdamW
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    model = VisionTransformer(img_size=32, patch_size=8, embed_dim=32, depth=4,
                              num_heads=4, num_classes=10, drop_rate=0.0,
                              attn_drop_rate=0.0)
    sched = PipelineScheduler(model, dtype=torch.float32)
    opt = FusedAdamW(model.parameters(), lr=1e-3)

    torch.manual_seed(7)
    data = []
    for _ in range(8):
        data.append({"images": torch.randn(2, 3, 32, 32),
                     "labels": torch.randint(0, 10, (2,))})
    tr = EagerTrainer(model, data, opt, grad_acc_steps=2,
                      pipeline_scheduler=sched)
    # the trainer path exercises pipeline_stage_batch_keys + boundary shapes
    tr.train(0, 3)
    # stage-1 (non-first) batches must NOT carry the image tensor
    b = tr.get_batch(data[0])
    if rank == 1:
        assert "images" not in b and "labels" in b
    else:
        assert "images" in b
    return True


def test_vit_pp2_end_to_end():
    """ViT under PP through the TRAINER (catches the boundary-shape
    derivation and the stage batch-key filtering)."""
    run_dist(_vit_pp2_worker, 2)


def _t5_eval_pp2_worker(rank, world):
    import torch

    from libai_amd.evaluation import inference_on_dataset
    from libai_amd.evaluation.ppl_evaluator import PPLEvaluator
    from libai_amd.models.t5_model import T5ForPreTraining
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"pipeline_parallel_size": 2, "pipeline_num_layers": 4})
    torch.manual_seed(123)
    model = T5ForPreTraining(**T5_KW)
    sched = PipelineScheduler(model, dtype=torch.float32)
    res = inference_on_dataset(model, _t5_batches(3), PPLEvaluator(),
                               pipeline_scheduler=sched)
    return res


def test_t5_eval_under_pp2():
    """Eval through the TUPLE-boundary pipeline (enc-dec): both ranks get
    identical, finite results."""
    results = run_dist(_t5_eval_pp2_worker, 2)
    assert results[0] == results[1] and results[0]
