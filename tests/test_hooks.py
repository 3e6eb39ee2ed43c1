"""Engine hooks: callback lifecycle, best-checkpointer, periodic GC.

Reference behavior: libai/engine/hooks.py:46-293 + utils/checkpoint.py:309-390.
"""

import os

import torch
from torch import nn

from libai_amd.engine.hooks import BestCheckpointer, CallbackHook
from libai_amd.engine.trainer import EagerTrainer
from libai_amd.optim import FusedAdamW
from libai_amd.utils.checkpoint import Checkpointer, PeriodicCheckpointer
from libai_amd.utils import distributed as du

du.setup_dist_util({})


class _ToyModel(nn.Module):
    def __init__(self):
        super().__init__()
        self.w = nn.Linear(4, 4)

    def forward(self, x):
        return {"loss": self.w(x).pow(2).mean()}


def _loader():
    while True:
        yield {"x": torch.randn(2, 4)}


def test_callback_hook_lifecycle(tmp_path):
    model = _ToyModel()
    opt = FusedAdamW(model.parameters(), lr=1e-2)
    trainer = EagerTrainer(model, _loader(), opt, grad_acc_steps=1)
    calls = []
    trainer.register_hooks([
        CallbackHook(
            before_train=lambda t: calls.append("bt"),
            after_train=lambda: calls.append("at"),
            before_step=lambda t: calls.append("bs"),
            after_step=lambda t: calls.append("as"),
        )
    ])
    trainer.train(0, 3)
    assert calls == ["bt", "bs", "as", "bs", "as", "bs", "as", "at"]


def test_best_checkpointer_saves_on_improvement(tmp_path):
    model = _ToyModel()
    ckpt = Checkpointer(model, save_dir=str(tmp_path))
    opt = FusedAdamW(model.parameters(), lr=1e-2)
    trainer = EagerTrainer(model, _loader(), opt, grad_acc_steps=1)
    hook = BestCheckpointer(eval_period=1, checkpointer=ckpt, val_metric="acc",
                            mode="max")
    trainer.register_hooks([hook])

    # drive the hook manually through the storage protocol
    from libai_amd.utils.events import EventStorage

    with EventStorage(0) as storage:
        trainer.storage = storage
        storage.put_scalar("acc", 0.5, smoothing_hint=False)
        hook.after_step()
        assert hook.best_value == 0.5
        storage.step()
        storage.put_scalar("acc", 0.4, smoothing_hint=False)
        hook.after_step()
        assert hook.best_value == 0.5  # no regression save
        storage.step()
        storage.put_scalar("acc", 0.9, smoothing_hint=False)
        hook.after_step()
        assert hook.best_value == 0.9
    assert os.path.exists(os.path.join(tmp_path, "model_best"))


def test_periodic_checkpointer_gc(tmp_path):
    model = _ToyModel()
    ckpt = Checkpointer(model, save_dir=str(tmp_path))
    pc = PeriodicCheckpointer(ckpt, period=1, max_iter=100, max_to_keep=2)
    for it in range(4):
        pc.step(it)
    kept = sorted(d for d in os.listdir(tmp_path) if d.startswith("model_"))
    # at most 2 periodic checkpoints retained
    assert len(kept) == 2, kept
    assert kept[-1] == "model_0000003"


def test_torch_profiler_hook(tmp_path):
    """TorchProfilerHook captures [start, end) and writes a chrome trace."""
    import torch
    from libai_amd.engine.hooks import TorchProfilerHook
    from libai_amd.engine.trainer import HookBase  # noqa: F401

    class _T:
        iter = 0

    hook = TorchProfilerHook(str(tmp_path), start_iter=1, end_iter=3)
    hook.trainer = _T()
    x = torch.randn(8, 8)
    for it in range(5):
        hook.trainer.iter = it
        hook.before_step()
        (x @ x).sum()
        hook.after_step()
    traces = list(tmp_path.glob("rank*_trace.json"))
    assert len(traces) == 1 and traces[0].stat().st_size > 0
    assert hook._prof is None  # closed, one-shot
