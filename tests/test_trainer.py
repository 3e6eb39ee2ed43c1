"""DefaultTrainer harness tests (reference: tests/test_trainer.py + the
trainer fixture model tests/layers/test_trainer_model.py): end-to-end config
-> train -> checkpoint -> resume, single-process and 2-process gloo DP."""

import os
import tempfile

import pytest
import torch

from tests.dist_helper import run_dist

TINY_OVERRIDES = [
    "model.cfg.hidden_layers=2",
    "model.cfg.hidden_size=64",
    "model.cfg.ffn_hidden_size=256",
    "model.cfg.num_attention_heads=4",
    "model.cfg.max_seq_length=32",
    "model.cfg.vocab_size=128",
    "dataloader.train.dataset.vocab_size=128",
    "dataloader.train.dataset.seq_length=32",
    "dataloader.train.dataset.size=256",
    "dataloader.train.num_workers=0",
    "train.train_micro_batch_size=2",
    "train.amp.enabled=False",
    "train.train_iter=6",
    "train.log_period=2",
    "train.checkpointer.period=3",
]


def _make_cfg(output_dir, extra=()):
    from libai_amd.config import LazyConfig

    cfg = LazyConfig.load(os.path.join(os.path.dirname(__file__), "..", "configs",
                                       "gpt2_pretrain.py"))
    LazyConfig.apply_overrides(cfg, list(TINY_OVERRIDES) + list(extra))
    cfg.train.output_dir = output_dir
    return cfg


def test_default_trainer_runs_and_checkpoints(tmp_path):
    from libai_amd.engine import DefaultTrainer, default_setup

    cfg = _make_cfg(str(tmp_path))
    default_setup(cfg)
    trainer = DefaultTrainer(cfg)
    trainer.train()
    ckpt_dir = os.path.join(str(tmp_path), "checkpoints")
    assert os.path.exists(os.path.join(ckpt_dir, "last_checkpoint"))
    assert os.path.exists(os.path.join(ckpt_dir, "model_final", "model.pt"))
    # metrics written
    assert os.path.exists(os.path.join(str(tmp_path), "metrics.json"))


def test_default_trainer_resume(tmp_path):
    from libai_amd.config import LazyConfig
    from libai_amd.engine import DefaultTrainer, default_setup

    cfg = _make_cfg(str(tmp_path))
    default_setup(cfg)
    DefaultTrainer(cfg).train()

    cfg2 = _make_cfg(str(tmp_path), extra=["train.train_iter=9", "train.resume=True"])
    trainer2 = DefaultTrainer(cfg2)
    assert trainer2.start_iter == 6, trainer2.start_iter
    trainer2.train()
    assert trainer2.iter == 9


def _dp2_worker(rank, world, output_dir):
    import torch

    from libai_amd.engine import DefaultTrainer, default_setup

    cfg = _make_cfg(output_dir)
    default_setup(cfg)
    trainer = DefaultTrainer(cfg)
    trainer.train()
    # all ranks end with identical parameters (DP sync)
    import torch.distributed as dist

    flat = torch.cat([p.detach().reshape(-1) for p in trainer.model.parameters()])
    gathered = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    assert torch.allclose(gathered[0], gathered[1], atol=1e-6)
    return True


def test_default_trainer_dp2_gloo(tmp_path):
    assert all(run_dist(_dp2_worker, 2, args=(str(tmp_path),)))


def _pp2_trainer_worker(rank, world, output_dir):
    from libai_amd.engine import DefaultTrainer, default_setup

    cfg = _make_cfg(
        output_dir,
        extra=[
            "train.dist.pipeline_parallel_size=2",
            "train.dist.pipeline_num_layers=2",
            "train.num_accumulation_steps=2",
        ],
    )
    default_setup(cfg)
    trainer = DefaultTrainer(cfg)
    trainer.train()
    return True


def test_default_trainer_pp2_gloo(tmp_path):
    assert all(run_dist(_pp2_trainer_worker, 2, args=(str(tmp_path),)))
