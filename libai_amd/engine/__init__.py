from .default import DefaultTrainer, default_setup
from .trainer import EagerTrainer, GraphTrainer, HookBase, TrainerBase

__all__ = ["DefaultTrainer", "default_setup", "EagerTrainer",
    "GraphTrainer", "HookBase", "TrainerBase"]
