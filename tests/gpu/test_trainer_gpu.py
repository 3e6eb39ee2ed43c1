"""DefaultTrainer end-to-end on one MI355X: bf16 train -> checkpoint ->
resume -> eval through the real engine stack (config, hooks, fused AdamW,
checkpointer, evaluator)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _cfg(tmp_path, train_iter=8):
    from libai_amd.config import LazyCall
    from libai_amd.config.lazy import ConfigDict
    from libai_amd.data import build_nlp_train_loader
    from libai_amd.data.datasets import SyntheticGPTDataset
    from libai_amd.models import GPTForPreTraining
    from libai_amd.scheduler import WarmupCosineLR
    from libai_amd.optim import FusedAdamW, get_default_optimizer_params

    cfg = ConfigDict()
    cfg.model = LazyCall(GPTForPreTraining)(
        hidden_layers=2, vocab_size=1024, hidden_size=256,
        ffn_hidden_size=512, num_attention_heads=4, max_seq_length=128,
        embedding_dropout_prob=0.1, attention_dropout_prob=0.1,
        output_dropout_prob=0.1,
    )
    cfg.dataloader = ConfigDict()
    cfg.dataloader.train = LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticGPTDataset)(
            vocab_size=1024, seq_length=128, size=512),
        train_batch_size=4, num_workers=0,
    )
    cfg.optim = LazyCall(FusedAdamW)(
        params=LazyCall(get_default_optimizer_params)(base_lr=1e-3),
        lr=1e-3, weight_decay=0.01, clip_grad=1.0,
    )
    cfg.train = ConfigDict(
        output_dir=str(tmp_path),
        train_micro_batch_size=4,
        num_accumulation_steps=1,
        train_iter=train_iter,
        log_period=2,
        amp=dict(enabled=True),
        checkpointer=dict(period=4, max_to_keep=2),
        evaluation=dict(enabled=False, eval_period=0),
        dist=dict(data_parallel_size=None, tensor_parallel_size=1,
                  pipeline_parallel_size=1),
        seed=1234,
    )
    return cfg


def test_trainer_train_ckpt_resume_gpu(tmp_path):
    from libai_amd.engine.default import DefaultTrainer
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    cfg = _cfg(tmp_path)
    tr = DefaultTrainer(cfg)
    tr.train()
    ckpts = os.listdir(os.path.join(str(tmp_path), "checkpoints"))
    assert "model_final" in ckpts, ckpts

    # resume continues from the saved iteration with restored state
    cfg2 = _cfg(tmp_path, train_iter=12)
    cfg2.train.resume = True
    tr2 = DefaultTrainer(cfg2)
    assert tr2.start_iter == 8, tr2.start_iter
    tr2.train()

    # weights-only load into a fresh trainer must not revert on step
    # (the resync_masters contract) — loss stays finite throughout
    cfg3 = _cfg(tmp_path, train_iter=2)
    cfg3.train.output_dir = str(tmp_path / "ft")
    cfg3.train.load_weight = os.path.join(str(tmp_path), "checkpoints",
                                          "model_final")
    tr3 = DefaultTrainer(cfg3)
    before = [p.detach().clone() for p in tr3.model.parameters()]
    loaded = torch.load(os.path.join(str(tmp_path), "checkpoints",
                                     "model_final", "model.pt"),
                        map_location="cpu", weights_only=False)
    name0, p0 = next(iter(tr3.model.named_parameters()))
    assert torch.equal(p0.detach().cpu(), loaded[name0].to(p0.dtype)), \
        "load_weight did not apply"
    tr3.train()
    assert any(not torch.equal(b, p.detach())
               for b, p in zip(before, tr3.model.parameters())), \
        "fine-tune step did not train"
