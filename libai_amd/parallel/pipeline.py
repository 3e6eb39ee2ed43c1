"""Explicit 1F1B pipeline-parallel engine.

Re-materializes what the reference got from OneFlow's nn.Graph compiler
(SURVEY.md §2.5 PP row; reference: libai/models/utils/graph_base.py:63-64 set
grad-accumulation => internal 1F1B; libai/layers/transformer_layer.py:158
to_global stage hops => explicit RCCL P2P over xGMI here).

Schedule: classic 1F1B — stage s of p runs min(p-1-s, m) warmup forwards,
then steady 1F1B, then cooldown backwards.  Per directed (stage, stage+1)
pair the messages are homogeneous (activations down, gradients up) and
strictly micro-batch ordered, so blocking irecv + isend pairs cannot
mismatch on the RCCL communicator.

The engine owns stage-local execution: it takes the model's
``pipeline_units()`` (ordered (layer_idx, name, fn)), keeps the units of the
local stage, and prunes every non-local module (frees its parameters).
Modules shared between stages (tied embeddings: reference
libai/layers/lm_logits.py:44 moves the embedding to the last stage) stay
materialized on every owning stage and their grads are all-reduced over a
dedicated tied-parameter group after each step.
"""

import logging

import torch
import torch.distributed as dist
from torch import nn

from ..utils import distributed as du

__all__ = ["PipelineScheduler", "prune_non_local_stages"]

logger = logging.getLogger(__name__)


def _module_param_ids(module):
    return {id(p) for p in module.parameters()}


def prune_non_local_stages(model, stage_modules, local_stage, dutil):
    """Free parameters owned by other stages by replacing their modules with
    Identity.  A module shared between stages (tied embeddings) survives on
    every owning stage: replacement recurses and only drops submodules that
    own NO locally-kept parameter.

    stage_modules: {layer_idx: [modules]} from model.pipeline_stage_modules().
    Returns the set of param ids kept locally.
    """
    keep_ids = set()
    drop_modules = []
    for layer_idx, mods in stage_modules.items():
        stage = dutil.layer_stage_id(layer_idx)
        for m in mods:
            if stage == local_stage:
                keep_ids |= _module_param_ids(m)
            else:
                drop_modules.append(m)

    def drop_or_recurse(module):
        """Replace children owning no kept params; recurse otherwise."""
        for name, child in list(module.named_children()):
            child_ids = _module_param_ids(child)
            if child_ids & keep_ids:
                drop_or_recurse(child)
            else:
                if isinstance(module, nn.ModuleList):
                    module[int(name)] = nn.Identity()
                else:
                    setattr(module, name, nn.Identity())

    name_of = {id(m): n for n, m in model.named_modules()}
    for m in drop_modules:
        if not (_module_param_ids(m) - keep_ids):
            # everything in it is kept (or it is empty): nothing to drop
            if not _module_param_ids(m):
                continue
            continue
        name = name_of.get(id(m))
        if name is None:
            continue
        if _module_param_ids(m) & keep_ids:
            drop_or_recurse(m)
        else:
            parent_name, _, attr = name.rpartition(".")
            parent = model.get_submodule(parent_name) if parent_name else model
            if isinstance(parent, nn.ModuleList):
                parent[int(attr)] = nn.Identity()
            else:
                setattr(parent, attr, nn.Identity())
    return keep_ids


class PipelineScheduler:
    def __init__(self, model, activation_shape=None, dtype=torch.float32):
        self.dutil = du.get_dist_util()
        self.model = model
        self.dtype = dtype
        self.pp = self.dutil.pipeline_parallel_size
        self.stage = self.dutil.pipeline_parallel_rank
        assert self.pp > 1, "PipelineScheduler requires pipeline_parallel_size > 1"

        units = model.pipeline_units()
        self.local_units = [
            (idx, name, fn)
            for idx, name, fn in units
            if self.dutil.layer_stage_id(idx) == self.stage
        ]
        assert self.local_units, f"stage {self.stage} owns no pipeline units"
        self.is_first = self.stage == 0
        self.is_last = self.stage == self.pp - 1

        stage_modules = model.pipeline_stage_modules()
        prune_non_local_stages(model, stage_modules, self.stage, self.dutil)
        self._build_tied_groups(stage_modules)

        self.activation_shape = activation_shape or self._default_activation_shape
        self._hidden_size = getattr(model, "hidden_size", None)

    # -- tied parameters ----------------------------------------------------

    def _build_tied_groups(self, stage_modules):
        """Params owned by >1 stage (tied embeddings) need grad all-reduce
        across those stages.  Every rank constructs every group (same order)."""
        dutil = self.dutil
        owner_stages = {}  # param id -> sorted set of stages
        param_obj = {}
        for layer_idx, mods in stage_modules.items():
            stage = dutil.layer_stage_id(layer_idx)
            for m in mods:
                for p in m.parameters():
                    owner_stages.setdefault(id(p), set()).add(stage)
                    param_obj[id(p)] = p
        tied = {
            pid: sorted(stages)
            for pid, stages in owner_stages.items()
            if len(stages) > 1
        }
        self.tied_params = []  # (param, group) for local tied params
        groups_made = {}
        dp, tp = dutil.data_parallel_size, dutil.tensor_parallel_size
        for pid, stages in sorted(tied.items(), key=lambda kv: kv[1][0]):
            key = tuple(stages)
            if key not in groups_made:
                mygroup = None
                for d in range(dp):
                    for t in range(tp):
                        ranks = [s * dp * tp + d * tp + t for s in stages]
                        g = dist.new_group(ranks)
                        if d == dutil.data_parallel_rank and t == dutil.tensor_parallel_rank:
                            mygroup = g
                groups_made[key] = mygroup
            if self.stage in stages:
                self.tied_params.append((param_obj[pid], groups_made[key]))

    def sync_tied_grads(self):
        for p, group in self.tied_params:
            if p.grad is not None:
                dist.all_reduce(p.grad.data, group=group)

    # -- p2p ----------------------------------------------------------------
    #
    # A pipeline boundary carries a TUPLE of activation tensors (usually one;
    # enc-dec models like T5 carry (encoder_out, decoder_hidden) between
    # decoder stages — reference t5_model.py:450+ relies on OneFlow moving
    # both global tensors).  Single-tensor models keep the legacy
    # fn(hidden, batch) unit signature; tuple-state models take/return tuples.

    @staticmethod
    def _to_tuple(x):
        if x is None:
            return None
        if isinstance(x, (tuple, list)):
            return tuple(x)
        return (x,)

    def _default_activation_shape(self, batch):
        ref = batch.get("input_ids")
        if ref is None:
            ref = next(v for v in batch.values() if torch.is_tensor(v))
        h = self._hidden_size
        if h is None:
            raise RuntimeError(
                "PipelineScheduler needs model.hidden_size or an activation_shape fn"
            )
        return (*ref.shape[:2], h)

    def boundary_shapes(self, batch):
        """Shapes of the activation tuple entering this stage's first unit."""
        if hasattr(self.model, "pipeline_boundary_shapes"):
            first_idx = self.local_units[0][0]
            return [tuple(s) for s in
                    self.model.pipeline_boundary_shapes(batch, first_idx)]
        return [tuple(self.activation_shape(batch))]

    def _comm(self, send_prev=None, send_next=None, recv_prev_shapes=None,
              recv_next_shapes=None, device=None):
        """One fused P2P exchange (RCCL group): any subset of
        {send to prev, send to next, recv from prev, recv from next}; each
        side is a LIST of tensors/shapes (the activation tuple).

        Fusing the steady-state send_forward+recv_backward (and the mirror
        send_backward+recv_forward) into ONE batch_isend_irecv is what makes
        the 1F1B kernel ordering deadlock-free on RCCL — the unfused blocking
        ops form a cycle between adjacent stages.
        """
        ops = []
        recv_prev = recv_next = None
        if recv_prev_shapes is not None:
            recv_prev = [torch.empty(s, dtype=self.dtype, device=device)
                         for s in recv_prev_shapes]
            ops += [dist.P2POp(dist.irecv, t, self.dutil.prev_pipeline_rank())
                    for t in recv_prev]
        if recv_next_shapes is not None:
            recv_next = [torch.empty(s, dtype=self.dtype, device=device)
                         for s in recv_next_shapes]
            ops += [dist.P2POp(dist.irecv, t, self.dutil.next_pipeline_rank())
                    for t in recv_next]
        if send_prev is not None:
            ops += [dist.P2POp(dist.isend, t.contiguous(),
                               self.dutil.prev_pipeline_rank())
                    for t in send_prev]
        if send_next is not None:
            ops += [dist.P2POp(dist.isend, t.contiguous(),
                               self.dutil.next_pipeline_rank())
                    for t in send_next]
        if ops:
            for req in dist.batch_isend_irecv(ops):
                req.wait()
        return recv_prev, recv_next

    # -- schedule ------------------------------------------------------------

    def _run_units(self, xs, batch):
        """xs: tuple of recv'd activations (or None on the first stage).
        Units take/return a bare tensor when the state is a single tensor
        (legacy protocol) and a tuple otherwise."""
        h = xs[0] if (xs is not None and len(xs) == 1) else xs
        for idx, name, fn in self.local_units:
            h = fn(h, batch)
        return h

    def _loss_backward(self, out, num_micro):
        losses = {k: v for k, v in out.items() if v.requires_grad}
        total = sum(losses.values()) / num_micro
        total.backward()

    @staticmethod
    def _grads_of(xs):
        return [t.grad if t.grad is not None else torch.zeros_like(t) for t in xs]

    def run_1f1b(self, micro_batches):
        """Run fwd+bwd over the micro-batches; returns the averaged loss dict
        on the last stage, None elsewhere."""
        device = du.get_device()
        m = len(micro_batches)
        num_warmup = min(self.pp - 1 - self.stage, m)
        num_steady = m - num_warmup

        in_flight = []  # (x_in, out) queue, oldest first
        loss_acc = {}
        it = iter(micro_batches)

        def accumulate_loss(out):
            if self.is_last:
                for k, v in out.items():
                    if torch.is_tensor(v) and v.ndim == 0:
                        loss_acc[k] = loss_acc.get(k, 0.0) + v.detach() / m

        def recv_fwd(batch):
            if self.is_first:
                return None
            xs, _ = self._comm(recv_prev_shapes=self.boundary_shapes(batch),
                               device=device)
            return tuple(t.requires_grad_(True) for t in xs)

        def detached(out):
            return [t.detach() for t in self._to_tuple(out)]

        # warmup forwards
        for _ in range(num_warmup):
            batch = next(it)
            x = recv_fwd(batch)
            out = self._run_units(x, batch)
            accumulate_loss(out)
            if not self.is_last:
                self._comm(send_next=detached(out))
            in_flight.append((x, out))

        # steady 1F1B
        x_next_batch = None
        if num_steady > 0:
            x_next_batch = next(it)
            x_next = recv_fwd(x_next_batch)
        for i in range(num_steady):
            out = self._run_units(x_next, x_next_batch)
            accumulate_loss(out)
            in_flight.append((x_next, out))
            # fused: send fwd output down, recv bwd grad up
            if self.is_last:
                grad_out = None
            else:
                out_t = self._to_tuple(out)
                _, grad_out = self._comm(send_next=detached(out),
                                         recv_next_shapes=[t.shape for t in out_t],
                                         device=device)
            bx, bout = in_flight.pop(0)
            if self.is_last:
                self._loss_backward(bout, m)
            else:
                torch.autograd.backward(self._to_tuple(bout),
                                        grad_tensors=grad_out)
            last_iter = i == num_steady - 1
            if self.is_first:
                if not last_iter:
                    x_next_batch = next(it)
                    x_next = None
            else:
                if last_iter:
                    self._comm(send_prev=self._grads_of(bx))
                else:
                    x_next_batch = next(it)
                    x_next, _ = self._comm(
                        send_prev=self._grads_of(bx),
                        recv_prev_shapes=self.boundary_shapes(x_next_batch),
                        device=device,
                    )
                    x_next = tuple(t.requires_grad_(True) for t in x_next)

        # cooldown backwards
        while in_flight:
            bx, bout = in_flight.pop(0)
            if self.is_last:
                self._loss_backward(bout, m)
            else:
                bout_t = self._to_tuple(bout)
                _, grad_out = self._comm(
                    recv_next_shapes=[t.shape for t in bout_t], device=device)
                torch.autograd.backward(bout_t, grad_tensors=grad_out)
            if not self.is_first:
                self._comm(send_prev=self._grads_of(bx))

        self.sync_tied_grads()
        return loss_acc if self.is_last else None

    # -- inference ----------------------------------------------------------

    @torch.no_grad()
    def run_eval(self, batch):
        """Single-batch forward through the pipeline; last stage returns the
        output dict, which is then broadcast nowhere (caller decides)."""
        device = du.get_device()
        x = None
        if not self.is_first:
            x, _ = self._comm(recv_prev_shapes=self.boundary_shapes(batch),
                              device=device)
            x = tuple(x)
        h = self._run_units(x, batch)
        if not self.is_last:
            self._comm(send_next=list(self._to_tuple(h)))
            return None
        return h
