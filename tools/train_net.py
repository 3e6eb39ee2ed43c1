#!/usr/bin/env python3
"""Training entry point (reference: tools/train_net.py:32-71)."""

import random
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import numpy as np
import torch

from libai_amd.config import LazyConfig, default_argument_parser, try_get_key
from libai_amd.engine import DefaultTrainer, default_setup


def main(args):
    cfg = LazyConfig.load(args.config_file)
    cfg = LazyConfig.apply_overrides(cfg, args.opts)
    if args.fast_dev_run:
        cfg.train.train_epoch = 0
        cfg.train.train_iter = 20
        cfg.train.evaluation.eval_period = 10
        cfg.train.log_period = 1
    if args.resume:
        cfg.train.resume = True
    default_setup(cfg, args)

    if args.eval_only:
        trainer = DefaultTrainer(cfg)
        return DefaultTrainer.test(cfg, model=trainer.model)

    trainer = DefaultTrainer(cfg)
    return trainer.train()


if __name__ == "__main__":
    args = default_argument_parser().parse_args()
    main(args)
