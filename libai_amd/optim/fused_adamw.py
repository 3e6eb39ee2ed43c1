"""Flat-buffer fused AdamW for MI355X.

Design (replaces the reference's OneFlow fused model update + param-group
clip_grad, reference: libai/models/utils/graph_base.py:74-76,
libai/optim/build.py:86-126):

  * Parameters are grouped into buckets by (dtype, weight-decay on/off,
    tp-sharded) and each bucket's storage is FLATTENED: ``p.data`` becomes a
    view into one contiguous bf16 buffer, and ``p.grad`` a view into a flat
    grad buffer.  DP all-reduce, grad zeroing, clipping and the AdamW update
    each touch a handful of contiguous buffers instead of hundreds of
    tensors.
  * bf16 training keeps fp32 master weights; the HIP kernel updates master
    and rewrites the bf16 copy in one pass (csrc/kernels/adamw.hip).
  * Gradient clipping computes the global L2 norm with a multi-tensor HIP
    kernel; TP-sharded buckets contribute from every TP rank, replicated
    buckets only from tp_rank 0, then the squared norm is all-reduced over
    the model-parallel (TP x PP) axes — SURVEY.md §7 hard part 7.
  * On CPU (tests) the same class runs a pure-torch reference update with
    identical flat-buffer semantics.
"""

import math

import torch
import torch.distributed as dist

from ..ops._ext import ext, has_ext
from ..utils import distributed as du

__all__ = ["FusedAdamW"]


class _Bucket:
    def __init__(self, params, dtype, device, weight_decay_on, tp_sharded):
        self.params = params
        self.dtype = dtype
        self.weight_decay_on = weight_decay_on
        self.tp_sharded = tp_sharded
        self.numel = sum(p.numel() for p in params)
        self.flat_param = torch.empty(self.numel, dtype=dtype, device=device)
        self.flat_grad = torch.zeros(self.numel, dtype=dtype, device=device)
        # flatten param storage and attach grad views
        off = 0
        for p in params:
            n = p.numel()
            self.flat_param[off : off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off : off + n].view(p.shape)
            if p.grad is not None:  # preserve grads accumulated before flattening
                self.flat_grad[off : off + n].copy_(p.grad.reshape(-1))
            p.grad = self.flat_grad[off : off + n].view(p.shape)
            off += n
        self.flat_master = self.flat_param.float() if dtype != torch.float32 else self.flat_param
        self.exp_avg = torch.zeros(self.numel, dtype=torch.float32, device=device)
        self.exp_avg_sq = torch.zeros(self.numel, dtype=torch.float32, device=device)
        self._adam_desc = None
        self._norm_desc = None

    def adam_desc(self, chunk):
        if self._adam_desc is None:
            esz = self.flat_param.element_size()
            rows = []
            for off in range(0, self.numel, chunk):
                n = min(chunk, self.numel - off)
                rows.append(
                    [
                        self.flat_param.data_ptr() + off * esz
                        if self.dtype != torch.float32
                        else 0,
                        self.flat_master.data_ptr() + off * 4,
                        self.flat_grad.data_ptr() + off * esz,
                        self.exp_avg.data_ptr() + off * 4,
                        self.exp_avg_sq.data_ptr() + off * 4,
                        n,
                    ]
                )
            self._adam_desc = torch.tensor(
                rows, dtype=torch.int64, device=self.flat_param.device
            )
        return self._adam_desc

    def norm_desc(self, chunk):
        if self._norm_desc is None:
            esz = self.flat_grad.element_size()
            rows = []
            for off in range(0, self.numel, chunk):
                n = min(chunk, self.numel - off)
                rows.append([self.flat_grad.data_ptr() + off * esz, n])
            self._norm_desc = torch.tensor(
                rows, dtype=torch.int64, device=self.flat_grad.device
            )
        return self._norm_desc


def _is_tp_sharded(p):
    return bool(getattr(p, "tensor_parallel", False))


class FusedAdamW(torch.optim.Optimizer):
    """AdamW over flat buckets.  Extra kwargs:

    clip_grad: max global grad norm (0 = off), applied inside step()
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01, clip_grad=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.clip_grad = clip_grad
        self._step = 0
        self._buckets = None  # list of (group_idx, _Bucket)

    # -- bucket construction (lazy: after model is on its final device) -----

    def _build_buckets(self):
        self._buckets = []
        for gi, group in enumerate(self.param_groups):
            by_key = {}
            for p in group["params"]:
                if not p.requires_grad:
                    continue
                key = (p.dtype, p.device, _is_tp_sharded(p))
                by_key.setdefault(key, []).append(p)
            wd_on = group["weight_decay"] > 0
            for (dtype, device, tp_sharded), plist in by_key.items():
                self._buckets.append(
                    (gi, _Bucket(plist, dtype, device, wd_on, tp_sharded))
                )

    @property
    def buckets(self):
        if self._buckets is None:
            self._build_buckets()
        return self._buckets

    def zero_grad(self, set_to_none=False):
        # grads are persistent flat views -> zero in place, never free
        for _, b in self.buckets:
            b.flat_grad.zero_()

    # -- grad norm / clip ---------------------------------------------------

    def _grad_norm_sq(self):
        dutil = du.get_dist_util()
        device = self.buckets[0][1].flat_grad.device
        total = torch.zeros(1, dtype=torch.float32, device=device)
        use_hip = device.type == "cuda" and has_ext()
        chunk = ext().adamw_chunk_elems() if use_hip else 0
        for _, b in self.buckets:
            if not b.tp_sharded and dutil.tensor_parallel_rank != 0:
                continue  # replicated grads counted once per TP group
            if use_hip:
                ext().l2norm_sq(b.norm_desc(chunk), b.dtype == torch.bfloat16, total)
            else:
                total += b.flat_grad.float().pow(2).sum()
        if dist.is_initialized():
            if dutil.tensor_parallel_size > 1:
                dist.all_reduce(total, group=dutil.tensor_parallel_group)
            if dutil.pipeline_parallel_size > 1:
                dist.all_reduce(total, group=dutil.pipeline_parallel_group)
        return total

    def grad_norm(self):
        return self._grad_norm_sq().sqrt()

    # -- step ---------------------------------------------------------------

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._step += 1
        t = self._step

        grad_scale = 1.0
        clip_coef_t = None
        if self.clip_grad and self.clip_grad > 0:
            norm = self._grad_norm_sq().sqrt()
            # compute on host only on CPU; on GPU keep device-side (one sync ok)
            clip_coef_t = (self.clip_grad / (norm + 1e-6)).clamp(max=1.0)
            grad_scale = float(clip_coef_t.item())

        for gi, b in self.buckets:
            group = self.param_groups[gi]
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"] if b.weight_decay_on else 0.0
            bc1 = 1.0 - beta1**t
            bc2 = 1.0 - beta2**t
            if b.flat_grad.device.type == "cuda" and has_ext():
                chunk = ext().adamw_chunk_elems()
                ext().adamw_step(
                    b.adam_desc(chunk), b.dtype == torch.bfloat16, lr, beta1, beta2,
                    eps, wd, bc1, bc2, grad_scale,
                )
            else:
                g = b.flat_grad.float() * grad_scale
                b.exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
                b.exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                mhat = b.exp_avg / bc1
                vhat = b.exp_avg_sq / bc2
                b.flat_master.add_(
                    mhat / (vhat.sqrt() + eps) + wd * b.flat_master, alpha=-lr
                )
                if b.dtype != torch.float32:
                    b.flat_param.copy_(b.flat_master.to(b.dtype))
        return loss

    # -- state dict (topology-independent per-param tensors) ----------------

    def state_dict(self):
        per_param = []
        for gi, b in self.buckets:
            off = 0
            for p in b.params:
                n = p.numel()
                per_param.append(
                    {
                        "master": b.flat_master[off : off + n].clone(),
                        "exp_avg": b.exp_avg[off : off + n].clone(),
                        "exp_avg_sq": b.exp_avg_sq[off : off + n].clone(),
                        "shape": list(p.shape),
                    }
                )
                off += n
        return {
            "step": self._step,
            "param_groups": [
                {k: v for k, v in g.items() if k != "params"} for g in self.param_groups
            ],
            "per_param": per_param,
        }

    def load_state_dict(self, state_dict):
        self._step = state_dict["step"]
        for g, saved in zip(self.param_groups, state_dict["param_groups"]):
            g.update(saved)
        idx = 0
        for gi, b in self.buckets:
            off = 0
            for p in b.params:
                n = p.numel()
                entry = state_dict["per_param"][idx]
                b.flat_master[off : off + n].copy_(entry["master"])
                b.exp_avg[off : off + n].copy_(entry["exp_avg"])
                b.exp_avg_sq[off : off + n].copy_(entry["exp_avg_sq"])
                idx += 1
                off += n
            if b.dtype != torch.float32:
                b.flat_param.copy_(b.flat_master.to(b.dtype))
