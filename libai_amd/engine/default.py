"""DefaultTrainer: config-driven end-to-end training orchestration.

Reference behavior: libai/engine/default.py:62-848 — batch-size arithmetic,
default_setup, build order (tokenizer -> loaders -> model -> optimizer ->
scheduler -> checkpointer -> hooks), resume logic, auto-scaling, test().
"""

import logging
import os

import torch

from ..config import instantiate, try_get_key
from ..data.structures import Instance
from ..optim.build import build_optimizer
from ..utils import distributed as du
from ..utils.checkpoint import Checkpointer, PeriodicCheckpointer
from ..utils.events import CommonMetricPrinter, JSONWriter, TensorboardXWriter
from ..utils.logger import setup_logger
from . import hooks
from .trainer import EagerTrainer, TrainerBase

__all__ = ["DefaultTrainer", "default_setup"]


def _check_batch_size(cfg):
    """global = micro x dp x acc (reference: default.py:62-119)."""
    dutil = du.get_dist_util()
    dp = dutil.data_parallel_size
    train = cfg.train
    micro = try_get_key(train, "train_micro_batch_size", default=None)
    glob = try_get_key(train, "global_batch_size", default=None)
    acc = try_get_key(train, "num_accumulation_steps", default=None)

    if micro is not None and glob is not None:
        if acc is None:
            assert glob % (micro * dp) == 0, (
                f"global_batch_size {glob} must be divisible by "
                f"micro_batch_size*dp ({micro}*{dp})"
            )
            acc = glob // (micro * dp)
        else:
            assert glob == micro * dp * acc, (
                f"global_batch_size {glob} != micro({micro}) x dp({dp}) x acc({acc})"
            )
    elif micro is not None:
        acc = acc or 1
        glob = micro * dp * acc
    elif glob is not None:
        acc = acc or 1
        assert glob % (dp * acc) == 0
        micro = glob // (dp * acc)
    else:
        raise ValueError("set train.train_micro_batch_size or train.global_batch_size")

    # pipeline parallelism executes grad-acc as its micro-batch count
    if dutil.pipeline_parallel_size > 1 and acc < dutil.pipeline_parallel_size:
        logging.getLogger(__name__).warning(
            f"num_accumulation_steps {acc} < pipeline stages "
            f"{dutil.pipeline_parallel_size}: the 1F1B bubble will dominate"
        )
    train.train_micro_batch_size = micro
    train.global_batch_size = glob
    train.num_accumulation_steps = acc
    return micro, glob, acc


def default_setup(cfg, args=None):
    """Logger + dist + seed + config snapshot (reference: default.py:147-201)."""
    output_dir = try_get_key(cfg, "train.output_dir", default="./output")
    if du.is_main_process():
        os.makedirs(output_dir, exist_ok=True)

    dist_cfg = try_get_key(cfg, "train.dist", default={})
    du.setup_dist_util(dist_cfg)
    rank = du.get_rank()
    logger = setup_logger(output_dir, distributed_rank=rank)
    logger.info(f"Rank of current process: {rank}. World size: {du.get_world_size()}")
    if args is not None:
        logger.info(f"Command line arguments: {args}")

    # model init must be IDENTICAL across DP ranks (and TP ranks slice one
    # full init): seed with the BASE seed here; DefaultTrainer re-seeds with
    # the per-dp-rank stream after the model is built so dropout diverges.
    seed = try_get_key(cfg, "train.seed", default=1234)
    torch.manual_seed(seed)
    import random

    import numpy as np

    random.seed(seed + rank)
    np.random.seed(seed + rank)

    _check_batch_size(cfg)

    fp8_cfg = try_get_key(cfg, "train.fp8", default=None)
    if fp8_cfg and fp8_cfg.get("enabled", False):
        from ..ops import fp8 as fp8_ops

        fp8_ops.set_fp8_gemms(True)
        logger.info("fp8 e4m3 forward GEMMs enabled (experimental; "
                    "backward stays bf16)")

    if du.is_main_process():
        from ..config import LazyConfig

        LazyConfig.save(cfg, os.path.join(output_dir, "config.yaml"))
    return cfg


class DefaultTrainer(TrainerBase):
    def __init__(self, cfg):
        super().__init__()
        self.cfg = cfg
        logger = logging.getLogger(__name__)
        self.logger = logger

        dutil = du.get_dist_util()
        if (
            try_get_key(cfg, "train.global_batch_size", default=None) is None
            or try_get_key(cfg, "train.num_accumulation_steps", default=None) is None
        ):
            _check_batch_size(cfg)
        micro, glob, acc = (
            cfg.train.train_micro_batch_size,
            cfg.train.global_batch_size,
            cfg.train.num_accumulation_steps,
        )
        logger.info(
            f"Batch sizes: micro={micro} global={glob} acc={acc} "
            f"dp={dutil.data_parallel_size} tp={dutil.tensor_parallel_size} "
            f"pp={dutil.pipeline_parallel_size}"
        )

        self.tokenizer = None
        if try_get_key(cfg, "tokenization", default=None) is not None:
            from ..tokenizer import build_tokenizer

            self.tokenizer = build_tokenizer(cfg)

        # dataloaders
        self.train_loader, self.valid_loader, self.test_loader = self.build_train_loader(cfg)

        # model (CPU init -> dtype -> device)
        self.model = self.build_model(cfg)

        # pipeline engine
        self.pipeline_scheduler = None
        if dutil.pipeline_parallel_size > 1:
            from ..parallel.pipeline import PipelineScheduler

            hidden = try_get_key(cfg, "model.cfg.hidden_size", "model.hidden_size",
                                 default=None)
            self.model.hidden_size = hidden
            self.pipeline_scheduler = PipelineScheduler(
                self.model, dtype=self._model_dtype(cfg)
            )
        self.model.to(du.get_device())

        # optimizer + scheduler (after model is on its final device: the
        # flat-bucket optimizer freezes param storage)
        self.optimizer = self.build_optimizer(cfg, self.model)
        if hasattr(self.optimizer, "set_param_names"):
            self.optimizer.set_param_names(self.model.named_parameters())
        if getattr(self.optimizer, "zero_stage", 0) == 3:
            from ..parallel.zero import setup_zero3

            self._zero3_manager = setup_zero3(self.model, self.optimizer)
        self.lr_scheduler = self.build_lr_scheduler(cfg, self.optimizer)

        # diverge the RNG stream per DP rank now that init is done (dropout
        # masks must differ across DP, stay identical within a TP group)
        torch.manual_seed(
            du.same_seed_for_tp_group(try_get_key(cfg, "train.seed", default=1234))
        )

        self._trainer = EagerTrainer(
            self.model, self.train_loader, self.optimizer, acc,
            pipeline_scheduler=self.pipeline_scheduler,
            comm_bucket_mb=try_get_key(cfg, "train.comm.bucket_mb", default=None),
        )

        self.checkpointer = Checkpointer(
            self.model,
            os.path.join(cfg.train.output_dir, "checkpoints"),
            optimizer=self.optimizer,
            lr_scheduler=self.lr_scheduler,
        )
        # reference auto-scales epochs -> iters (default.py:695-774)
        train_epoch = try_get_key(cfg, "train.train_epoch", default=0) or 0
        if train_epoch > 0 and self.train_loader is not None and hasattr(
            self.train_loader, "dataset"
        ):
            iters_per_epoch = max(len(self.train_loader.dataset) // glob, 1)
            cfg.train.train_iter = max(
                int(train_epoch * iters_per_epoch), cfg.train.train_iter or 0
            )
            logger.info(
                f"auto-scaled train_epoch={train_epoch} -> train_iter="
                f"{cfg.train.train_iter} ({iters_per_epoch} iters/epoch)"
            )
        self.max_iter = cfg.train.train_iter
        self.global_batch_size = glob
        self.start_iter = 0

        if try_get_key(cfg, "train.resume", default=False):
            extra = self.checkpointer.resume_or_load(
                try_get_key(cfg, "train.load_weight", default=""), resume=True
            )
            self.start_iter = int(extra.get("iteration", -1)) + 1
            if hasattr(self.train_loader, "batch_sampler") and hasattr(
                self.train_loader.batch_sampler, "set_consumed_samples"
            ):
                self.train_loader.batch_sampler.set_consumed_samples(
                    self.start_iter * glob
                )
        elif try_get_key(cfg, "train.load_weight", default=""):
            self.checkpointer.resume_or_load(cfg.train.load_weight, resume=False)

        self.register_hooks(self.build_hooks())

    # -- builders -----------------------------------------------------------

    def _model_dtype(self, cfg):
        if try_get_key(cfg, "train.amp.enabled", default=False):
            return torch.bfloat16
        return torch.float32

    @classmethod
    def build_model(cls, cfg):
        model = instantiate(cfg.model)
        if try_get_key(cfg, "train.amp.enabled", default=False):
            model = model.to(torch.bfloat16)
        if try_get_key(cfg, "train.activation_checkpoint.enabled", default=False):
            if hasattr(model, "set_activation_checkpoint"):
                model.set_activation_checkpoint(True)
        if try_get_key(cfg, "train.lora.enabled", default=False):
            from ..lora import apply_lora

            apply_lora(
                model,
                r=try_get_key(cfg, "train.lora.r", default=8),
                alpha=try_get_key(cfg, "train.lora.alpha", default=16),
                dropout=try_get_key(cfg, "train.lora.dropout", default=0.0),
            )
        logger = logging.getLogger(__name__)
        n_params = sum(p.numel() for p in model.parameters())
        logger.info(f"Model built: {n_params / 1e6:.1f}M local parameters")
        return model

    @classmethod
    def build_optimizer(cls, cfg, model):
        opt = build_optimizer(cfg.optim, model)
        from ..parallel.zero import zero_stage_from_config

        stage = zero_stage_from_config(cfg)
        if stage and hasattr(opt, "zero_stage"):
            opt.zero_stage = stage  # buckets are built lazily, so this applies
        return opt

    @classmethod
    def build_lr_scheduler(cls, cfg, optimizer):
        sched_cfg = try_get_key(cfg, "train.scheduler", default=None)
        if sched_cfg is None:
            return None
        import copy

        sched_cfg = copy.deepcopy(sched_cfg)
        sched_cfg["optimizer"] = optimizer
        return instantiate(sched_cfg)

    @classmethod
    def build_train_loader(cls, cfg):
        dl_cfg = try_get_key(cfg, "dataloader.train", default=None)
        if dl_cfg is None:
            return None, None, None
        result = instantiate(dl_cfg)
        if isinstance(result, tuple):
            train, valid, test = (list(result) + [None, None])[:3]
            return train, valid, test
        return result, None, None

    @classmethod
    def build_test_loader(cls, cfg):
        dl_cfg = try_get_key(cfg, "dataloader.test", default=None)
        if dl_cfg is None:
            return []
        result = instantiate(dl_cfg)
        return result if isinstance(result, list) else [result]

    def build_hooks(self):
        cfg = self.cfg
        ckpt_period = try_get_key(cfg, "train.checkpointer.period", default=5000)
        max_to_keep = try_get_key(cfg, "train.checkpointer.max_to_keep", default=None)
        log_period = try_get_key(cfg, "train.log_period", default=20)
        ret = [
            hooks.IterationTimer(),
            hooks.LRScheduler(),
            hooks.PeriodicCheckpointerHook(
                PeriodicCheckpointer(self.checkpointer, ckpt_period,
                                     max_iter=self.max_iter, max_to_keep=max_to_keep)
            ),
        ]
        prof = try_get_key(cfg, "train.profiler", default=None)
        if prof:
            ret.append(hooks.TorchProfilerHook(
                os.path.join(cfg.train.output_dir, "profiler"),
                start_iter=prof.get("start_iter", 10),
                end_iter=prof.get("end_iter", 13),
                with_stack=prof.get("with_stack", False),
            ))
        eval_period = try_get_key(cfg, "train.evaluation.eval_period", default=0)
        if eval_period and try_get_key(cfg, "train.evaluation.enabled", default=True):
            def _eval():
                return self.test(self.cfg, model=self.model,
                                 pipeline_scheduler=self.pipeline_scheduler)

            ret.append(hooks.EvalHook(eval_period, _eval))
        if du.is_main_process():
            ret.append(hooks.PeriodicWriter(self.build_writers(), period=log_period))
        return ret

    def build_writers(self):
        return [
            CommonMetricPrinter(self.global_batch_size, self.max_iter),
            JSONWriter(os.path.join(self.cfg.train.output_dir, "metrics.json")),
            TensorboardXWriter(os.path.join(self.cfg.train.output_dir, "tb")),
        ]

    # -- run ----------------------------------------------------------------

    def train(self):
        super().train(self.start_iter, self.max_iter)

    def run_step(self):
        self._trainer.iter = self.iter
        self._trainer.storage = self.storage
        self._trainer.run_step()

    def get_batch(self, data):
        if isinstance(data, Instance):
            data = data.to_dict()
        return self._trainer.get_batch(data)

    @classmethod
    def test(cls, cfg, model=None, evaluator=None, pipeline_scheduler=None):
        """Run evaluators over the test loaders (reference: default.py:781-848).

        With ``pipeline_scheduler`` the eval forward runs through the 1F1B
        engine's run_eval (every stage participates; reference evaluates
        pipelined graphs too, evaluator.py:119)."""
        from ..evaluation import inference_on_dataset

        test_loaders = cls.build_test_loader(cfg)
        if not test_loaders or model is None:
            return {}
        results = {}
        for i, loader in enumerate(test_loaders):
            ev = evaluator or instantiate(
                try_get_key(cfg, "train.evaluation.evaluator", default=None)
            )
            if ev is None:
                continue
            results[f"dataset_{i}"] = inference_on_dataset(
                model, loader, ev, pipeline_scheduler=pipeline_scheduler)
        return results
