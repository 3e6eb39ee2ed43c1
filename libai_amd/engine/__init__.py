from .default import DefaultTrainer, default_setup
from .trainer import EagerTrainer, HookBase, TrainerBase

__all__ = ["DefaultTrainer", "default_setup", "EagerTrainer", "HookBase", "TrainerBase"]
