"""DefaultTrainer.test(): evaluator plumbing over a test dataloader."""

import torch

from libai_amd.config import ConfigDict, LazyCall
from libai_amd.data import build_nlp_test_loader
from libai_amd.data.datasets import SyntheticImageDataset
from libai_amd.engine import DefaultTrainer
from libai_amd.evaluation import ClsEvaluator
from libai_amd.models import VisionTransformer
from libai_amd.utils import distributed as du


def test_default_trainer_test_with_evaluator():
    du.setup_dist_util({})
    cfg = ConfigDict(
        {
            "dataloader": {
                "test": LazyCall(build_nlp_test_loader)(
                    dataset=LazyCall(SyntheticImageDataset)(
                        img_size=32, num_classes=10, size=16
                    ),
                    test_batch_size=8,
                    num_workers=0,
                ),
            },
            "train": {
                "evaluation": {
                    "enabled": True,
                    "evaluator": LazyCall(ClsEvaluator)(topk=(1, 5)),
                    "eval_iter": 10,
                },
            },
        }
    )
    torch.manual_seed(0)
    model = VisionTransformer(img_size=32, patch_size=8, embed_dim=32, depth=1,
                              num_heads=2, num_classes=10).eval()
    results = DefaultTrainer.test(cfg, model=model)
    assert "dataset_0" in results
    assert "cls" in results["dataset_0"]
