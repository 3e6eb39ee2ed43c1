"""Topology-independent checkpointing.

Reference behavior: libai/utils/checkpoint.py:87-390 — checkpoints store FULL
(unsharded) tensors so they reshard onto any (dp, tp, pp) on load
(checkpoint.py:289-306), with a ``last_checkpoint`` tag file and
``max_to_keep`` GC in PeriodicCheckpointer.

Save: every TP-sharded parameter (marked ``tensor_parallel`` +
``tp_shard_dim`` by the layers) is all-gathered over the TP group and
concatenated; pp stages contribute their local keys; the (tp0, dp0) rank of
each stage writes its shard of the key space and rank 0 merges + writes the
single ``model.pt``.  Load: every rank reads the full file (shared fs) and
slices its own TP shard — no broadcast needed, any topology.
"""

import logging
import os
import shutil

import torch
import torch.distributed as dist

from . import distributed as du

__all__ = ["Checkpointer", "PeriodicCheckpointer"]


def _consolidate_model_state(model):
    """Full (unsharded) state_dict on this rank (TP gathers included)."""
    dutil = du.get_dist_util()
    tp = dutil.tensor_parallel_size
    params = dict(model.named_parameters())
    out = {}
    for name, t in model.state_dict().items():
        p = params.get(name)
        if p is not None and getattr(p, "tensor_parallel", False) and tp > 1:
            from ..layers.linear import tp_merge

            dim = getattr(p, "tp_shard_dim", 0)
            shards = [torch.empty_like(t) for _ in range(tp)]
            dist.all_gather(shards, t.contiguous(), group=dutil.tensor_parallel_group)
            out[name] = tp_merge(shards, dim,
                                 getattr(p, "tp_fused_chunks", 1)).cpu()
        elif p is not None and getattr(p, "expert_parallel", False) \
                and dutil.data_parallel_size > 1:
            # expert-parallel params: every EP(=DP) rank owns DIFFERENT
            # experts — gather the full expert dim so the checkpoint is
            # EP-topology-independent like the TP shards
            shards = [torch.empty_like(t) for _ in range(dutil.data_parallel_size)]
            dist.all_gather(shards, t.contiguous(),
                            group=dutil.data_parallel_group)
            out[name] = torch.cat(shards, dim=0).cpu()
        else:
            t = t.detach()
            # clone on CPU: .cpu() would alias the flat bucket storage, which
            # ZeRO-3 releases before torch.save serializes the state
            out[name] = t.clone() if t.device.type == "cpu" else t.cpu()
    return out


def _shard_for_load(model, full_state):
    """Slice full tensors into this rank's TP shards; returns local state."""
    dutil = du.get_dist_util()
    tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
    params = dict(model.named_parameters())
    local = {}
    for name, cur in model.state_dict().items():
        if name not in full_state:
            continue
        full = full_state[name]
        p = params.get(name)
        if p is not None and getattr(p, "tensor_parallel", False) and tp > 1:
            from ..layers.linear import tp_slice

            dim = getattr(p, "tp_shard_dim", 0)
            full = tp_slice(full, tp, tpr, dim, getattr(p, "tp_fused_chunks", 1))
        elif p is not None and getattr(p, "expert_parallel", False) \
                and full.shape[0] != cur.shape[0]:
            ep = dutil.data_parallel_size
            if full.shape[0] == cur.shape[0] * ep:
                full = full.chunk(ep, dim=0)[dutil.data_parallel_rank]
        if tuple(full.shape) != tuple(cur.shape):
            logging.getLogger(__name__).warning(
                f"checkpoint key {name}: shape {tuple(full.shape)} != model "
                f"{tuple(cur.shape)}; skipped"
            )
            continue
        local[name] = full
    return local


class Checkpointer:
    """Save/load checkpoints; one directory per checkpoint name."""

    def __init__(self, model, save_dir="", *, save_to_disk=None, **checkpointables):
        self.model = model
        self.save_dir = save_dir
        self.checkpointables = dict(checkpointables)
        self.logger = logging.getLogger(__name__)
        self.save_to_disk = (
            save_to_disk if save_to_disk is not None else du.is_main_process()
        )

    def add_checkpointable(self, key, obj):
        self.checkpointables[key] = obj

    def _zero3_opt(self):
        """The ZeRO-3 optimizer managing this model's released param storage
        (p.data is unreadable at rest; gather around state_dict/load)."""
        opt = self.checkpointables.get("optimizer") or getattr(
            self.model, "_zero3_optimizer", None)
        return opt if hasattr(opt, "materialize_all_params") else None

    def save(self, name, **kwargs):
        if not self.save_dir:
            return
        dutil = du.get_dist_util()
        ckpt_dir = os.path.join(self.save_dir, name)
        if du.is_main_process():
            os.makedirs(ckpt_dir, exist_ok=True)
        du.synchronize()

        # model: consolidate over TP; PP stages write stage files merged below
        z3 = self._zero3_opt()
        if z3 is not None:
            z3.materialize_all_params()
        model_state = _consolidate_model_state(self.model)
        if z3 is not None:
            z3.release_all_params()
        pp = dutil.pipeline_parallel_size
        writer = dutil.data_parallel_rank == 0 and dutil.tensor_parallel_rank == 0
        if pp > 1:
            if writer:
                torch.save(
                    model_state,
                    os.path.join(ckpt_dir, f"model_stage{dutil.pipeline_parallel_rank}.pt"),
                )
            du.synchronize()
            if du.is_main_process():
                merged = {}
                for s in range(pp):
                    path = os.path.join(ckpt_dir, f"model_stage{s}.pt")
                    merged.update(torch.load(path, map_location="cpu", weights_only=False))
                torch.save(merged, os.path.join(ckpt_dir, "model.pt"))
                for s in range(pp):
                    os.remove(os.path.join(ckpt_dir, f"model_stage{s}.pt"))
        elif writer and du.is_main_process():
            torch.save(model_state, os.path.join(ckpt_dir, "model.pt"))

        # other checkpointables: per-rank topology-tagged states for PP/TP-local
        # objects (optimizer/scheduler); rank-local shard files.
        # state_dict() may be COLLECTIVE (FusedAdamW gathers ZeRO shards over
        # DP and TP shards over TP) — every rank must call it; only the
        # writer rank saves the result.
        for key, obj in self.checkpointables.items():
            if not hasattr(obj, "state_dict"):
                continue
            state = obj.state_dict()
            if key == "optimizer" and (pp > 1 or dutil.tensor_parallel_size > 1):
                if dutil.data_parallel_rank == 0:
                    torch.save(
                        state,
                        os.path.join(
                            ckpt_dir,
                            f"{key}_tp{dutil.tensor_parallel_rank}"
                            f"_pp{dutil.pipeline_parallel_rank}.pt",
                        ),
                    )
            elif du.is_main_process():
                torch.save(state, os.path.join(ckpt_dir, f"{key}.pt"))

        if du.is_main_process():
            extra = dict(kwargs)
            torch.save(extra, os.path.join(ckpt_dir, "extra.pt"))
            self.tag_last_checkpoint(name)
        du.synchronize()
        self.logger.info(f"Saved checkpoint to {ckpt_dir}")

    def load(self, path, checkpointables=None):
        """Load a checkpoint directory; reshards model weights to the current
        topology.  Returns the extra metadata dict."""
        if not path:
            return {}
        dutil = du.get_dist_util()
        self.logger.info(f"Loading checkpoint from {path}")
        model_file = os.path.join(path, "model.pt")
        full_state = torch.load(model_file, map_location="cpu", weights_only=False)
        local = _shard_for_load(self.model, full_state)
        z3 = self._zero3_opt()
        if z3 is not None:
            z3.materialize_all_params()  # p.data must be writable
        missing, unexpected = self.model.load_state_dict(local, strict=False)
        if z3 is not None:
            z3.refresh_shards_from_params()
            z3.release_all_params()
        missing = [m for m in missing if m in dict(self.model.named_parameters())]
        if missing:
            self.logger.warning(f"missing keys in checkpoint: {missing[:10]}...")

        # The flat-bucket optimizer aliases p.data — after rewriting the
        # weights its fp32 masters are stale; refresh them (a subsequent
        # optimizer state load below simply overwrites them again).
        for obj in self.checkpointables.values():
            if hasattr(obj, "resync_masters"):
                obj.resync_masters()

        keys = (
            self.checkpointables.keys() if checkpointables is None else checkpointables
        )
        for key in keys:
            obj = self.checkpointables.get(key)
            if obj is None or not hasattr(obj, "load_state_dict"):
                continue
            f_sharded = os.path.join(
                path,
                f"{key}_tp{dutil.tensor_parallel_rank}"
                f"_pp{dutil.pipeline_parallel_rank}.pt",
            )
            f_plain = os.path.join(path, f"{key}.pt")
            f = f_sharded if os.path.exists(f_sharded) else f_plain
            if os.path.exists(f):
                obj.load_state_dict(torch.load(f, map_location="cpu", weights_only=False))
            else:
                self.logger.warning(f"no state for checkpointable {key!r} in {path}")
        extra_f = os.path.join(path, "extra.pt")
        if os.path.exists(extra_f):
            return torch.load(extra_f, map_location="cpu", weights_only=False)
        return {}

    def resume_or_load(self, path, *, resume=True):
        if resume and self.has_checkpoint():
            path = self.get_checkpoint_file()
            return self.load(path)
        if path:
            return self.load(path, checkpointables=[])  # weights only
        return {}

    def has_checkpoint(self):
        return os.path.exists(os.path.join(self.save_dir, "last_checkpoint"))

    def get_checkpoint_file(self):
        try:
            with open(os.path.join(self.save_dir, "last_checkpoint")) as f:
                return os.path.join(self.save_dir, f.read().strip())
        except OSError:
            return ""

    def tag_last_checkpoint(self, name):
        with open(os.path.join(self.save_dir, "last_checkpoint"), "w") as f:
            f.write(name)


class PeriodicCheckpointer:
    """Save every ``period`` iterations, keep at most ``max_to_keep``
    (reference: checkpoint.py:309-390)."""

    def __init__(self, checkpointer, period, max_iter=None, max_to_keep=None):
        self.checkpointer = checkpointer
        self.period = int(period)
        self.max_iter = max_iter
        self.max_to_keep = max_to_keep
        self._recent = []

    def step(self, iteration, **kwargs):
        iteration = int(iteration)
        extra = {"iteration": iteration}
        extra.update(kwargs)
        if (iteration + 1) % self.period == 0 or (
            self.max_iter is not None and iteration >= self.max_iter - 1
        ):
            name = f"model_{iteration:07d}"
            if self.max_iter is not None and iteration >= self.max_iter - 1:
                name = "model_final"
            self.checkpointer.save(name, **extra)
            if self.max_to_keep:
                self._recent.append(name)
                while len(self._recent) > self.max_to_keep:
                    old = self._recent.pop(0)
                    if du.is_main_process() and old != "model_final":
                        shutil.rmtree(
                            os.path.join(self.checkpointer.save_dir, old),
                            ignore_errors=True,
                        )

    def save(self, name, **kwargs):
        self.checkpointer.save(name, **kwargs)
