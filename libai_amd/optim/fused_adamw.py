"""Flat-buffer fused AdamW for MI355X, with ZeRO-1/2 sharding.

Design (replaces the reference's OneFlow fused model update + param-group
clip_grad + enable_zero, reference: libai/models/utils/graph_base.py:69-76,
libai/optim/build.py:86-126):

  * Parameters are grouped into buckets by (dtype, weight-decay on/off,
    tp-sharded) and each bucket's storage is FLATTENED: ``p.data`` becomes a
    view into one contiguous bf16 buffer, and ``p.grad`` a view into a flat
    grad buffer.  DP all-reduce / reduce-scatter, grad zeroing, clipping and
    the AdamW update each touch a handful of contiguous buffers.
  * bf16 training keeps fp32 master weights; the HIP kernel updates master
    and rewrites the bf16 copy in one pass (csrc/kernels/adamw.hip).
  * ZeRO (flat buckets make DP slices contiguous):
      - stage 1: optimizer state (master/m/v) exists only for this DP rank's
        1/dp slice; grads are all-reduced in full; after the slice update the
        bf16 params are all-gathered over DP.
      - stage 2: gradients are reduce-scattered so each rank only receives
        its own slice's reduction (C6 in SURVEY.md §2.4).
  * Gradient clipping computes the global L2 norm (TP-aware: replicated
    buckets counted once per TP group; under ZeRO the slice norms are
    additionally reduced over DP) and applies the clip scale inside the
    fused update — one pass over grads total.
"""

import logging

import torch
import torch.distributed as dist

from ..ops._ext import ext, has_ext
from ..utils import distributed as du

__all__ = ["FusedAdamW"]


def _pad_to(n, mult):
    return (n + mult - 1) // mult * mult


class _Bucket:
    def __init__(self, params, dtype, device, weight_decay_on, tp_sharded, dp_size,
                 dp_rank, zero_stage, unit=-1):
        self.params = params
        self.dtype = dtype
        self.weight_decay_on = weight_decay_on
        self.tp_sharded = tp_sharded
        self.zero = zero_stage
        self.unit = unit  # ZeRO-3 gather/release granularity (module unit id)
        self.dp_size = dp_size
        self.dp_rank = dp_rank
        raw = sum(p.numel() for p in params)
        self.numel = _pad_to(raw, dp_size * 64) if zero_stage > 0 else raw
        self.shard = self.numel // dp_size if zero_stage > 0 else self.numel
        self.shard_off = self.shard * dp_rank if zero_stage > 0 else 0

        self.flat_param = torch.zeros(self.numel, dtype=dtype, device=device)
        self.flat_grad = torch.zeros(self.numel, dtype=dtype, device=device)
        off = 0
        for p in params:
            n = p.numel()
            self.flat_param[off : off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off : off + n].view(p.shape)
            if p.grad is not None:  # preserve grads accumulated before flattening
                self.flat_grad[off : off + n].copy_(p.grad.reshape(-1))
            p.grad = self.flat_grad[off : off + n].view(p.shape)
            off += n

        # optimizer state exists only for the local shard under ZeRO
        owned = self.flat_param[self.shard_off : self.shard_off + self.shard]
        if zero_stage == 3:
            # persistent shard storage; the full flat buffers are released
            # between materializations (finalize_zero3 after the DP broadcast)
            self.shard_param = owned.detach().clone()
            self.shard_grad = torch.zeros(self.shard, dtype=dtype, device=device)
            self.flat_master = (
                self.shard_param.float() if dtype != torch.float32
                else self.shard_param
            )
        elif zero_stage > 0:
            self.flat_master = owned.float() if dtype != torch.float32 else owned
        else:
            self.flat_master = (
                self.flat_param.float() if dtype != torch.float32 else self.flat_param
            )
        state_n = self.shard if zero_stage > 0 else self.numel
        self.exp_avg = torch.zeros(state_n, dtype=torch.float32, device=device)
        self.exp_avg_sq = torch.zeros(state_n, dtype=torch.float32, device=device)
        self._adam_desc = None
        self._norm_desc = None

    # region updated by the fused kernel (local shard under ZeRO)
    def _upd_param(self):
        if self.zero == 3:
            return self.shard_param
        return self.flat_param[self.shard_off : self.shard_off + self.shard] \
            if self.zero > 0 else self.flat_param

    def _upd_grad(self):
        if self.zero == 3:
            return self.shard_grad
        return self.flat_grad[self.shard_off : self.shard_off + self.shard] \
            if self.zero > 0 else self.flat_grad

    # -- ZeRO-3 param/grad lifecycle ----------------------------------------
    # flat_param/flat_grad keep their Storage objects forever (p.data/p.grad
    # are views into them) but the backing memory is freed between uses via
    # untyped_storage().resize_(0) and re-allocated on materialize — the
    # FSDP idiom, sized here for per-transformer-layer units.

    def params_live(self):
        return self.flat_param.untyped_storage().size() > 0

    def grads_live(self):
        return self.flat_grad.untyped_storage().size() > 0

    def finalize_zero3(self):
        """After flatten + DP broadcast: snapshot the owned shard and free
        the full buffers (the first forward re-gathers them)."""
        self.shard_param.copy_(
            self.flat_param[self.shard_off : self.shard_off + self.shard])
        if self.dtype != torch.float32:
            self.flat_master.copy_(self.shard_param.float())
        self.release_params()
        self.flat_grad.untyped_storage().resize_(0)

    def materialize_params(self, group):
        if self.params_live():
            return
        esz = self.flat_param.element_size()
        self.flat_param.untyped_storage().resize_(self.numel * esz)
        if dist.is_initialized() and self.dp_size > 1:
            dist.all_gather_into_tensor(self.flat_param, self.shard_param,
                                        group=group)
        else:
            self.flat_param[self.shard_off : self.shard_off + self.shard].copy_(
                self.shard_param)

    def release_params(self):
        self.flat_param.untyped_storage().resize_(0)

    def materialize_grads(self):
        """(Re-)allocate the full grad buffer zeroed, and re-arm the
        pending-param set for this backward."""
        if not self.grads_live():
            esz = self.flat_grad.element_size()
            self.flat_grad.untyped_storage().resize_(self.numel * esz)
            self.flat_grad.zero_()
        self.pending = {id(p) for p in self.params}

    def reduce_release_grads(self, group, dp):
        """Average-reduce-scatter the full grads into the owned shard
        (accumulating across micro-batches) and free the full buffer."""
        self.flat_grad.div_(dp)
        if dist.is_initialized() and self.dp_size > 1:
            tmp = torch.empty_like(self.shard_grad)
            dist.reduce_scatter_tensor(tmp, self.flat_grad, group=group)
            self.shard_grad.add_(tmp)
        else:
            self.shard_grad.add_(
                self.flat_grad[self.shard_off : self.shard_off + self.shard])
        self.flat_grad.untyped_storage().resize_(0)

    def adam_desc(self, chunk):
        if self._adam_desc is None:
            esz = self.flat_param.element_size()
            up, ug = self._upd_param(), self._upd_grad()
            n_total = up.numel()
            rows = []
            for off in range(0, n_total, chunk):
                n = min(chunk, n_total - off)
                rows.append(
                    [
                        up.data_ptr() + off * esz if self.dtype != torch.float32 else 0,
                        self.flat_master.data_ptr() + off * 4,
                        ug.data_ptr() + off * esz,
                        self.exp_avg.data_ptr() + off * 4,
                        self.exp_avg_sq.data_ptr() + off * 4,
                        n,
                    ]
                )
            self._adam_desc = torch.tensor(rows, dtype=torch.int64,
                                           device=self.flat_param.device)
        return self._adam_desc

    def norm_desc(self, chunk):
        if self._norm_desc is None:
            esz = self.flat_grad.element_size()
            ug = self._upd_grad()
            rows = []
            for off in range(0, ug.numel(), chunk):
                n = min(chunk, ug.numel() - off)
                rows.append([ug.data_ptr() + off * esz, n])
            self._norm_desc = torch.tensor(rows, dtype=torch.int64,
                                           device=self.flat_grad.device)
        return self._norm_desc


def _is_tp_sharded(p):
    return bool(getattr(p, "tensor_parallel", False))


class FusedAdamW(torch.optim.Optimizer):
    """AdamW over flat buckets.  Extra kwargs:

    clip_grad: max global grad norm (0 = off), applied inside step()
    zero_stage: 0 (off) / 1 (state sharding) / 2 (+grad reduce-scatter)
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01, clip_grad=0.0, zero_stage=0):
        self._param_names = {}
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.clip_grad = clip_grad
        self.zero_stage = zero_stage
        self._step = 0
        self._buckets = None

    # -- bucket construction (lazy: after model is on its final device) -----

    def _build_buckets(self):
        dutil = du.get_dist_util()
        dp, dpr = dutil.data_parallel_size, dutil.data_parallel_rank
        zero = self.zero_stage if dp > 1 else 0
        self._zero_eff = zero
        self._buckets = []
        for gi, group in enumerate(self.param_groups):
            by_key = {}
            for p in group["params"]:
                if not p.requires_grad:
                    continue
                # stage 3 additionally splits buckets by module UNIT so the
                # gather/release lifecycle is per-transformer-layer
                unit = getattr(p, "_zero3_unit", -1) if zero == 3 else -1
                ep = bool(getattr(p, "expert_parallel", False))
                key = (p.dtype, p.device, _is_tp_sharded(p), unit, ep)
                by_key.setdefault(key, []).append(p)
            wd_on = group["weight_decay"] > 0
            for (dtype, device, tp_sharded, unit, ep), plist in by_key.items():
                if ep and zero > 0:
                    raise NotImplementedError(
                        "ZeRO is not composed with expert-parallel params: "
                        "the EP group IS the DP group, and each rank's expert "
                        "state is already unique"
                    )
                b = _Bucket(plist, dtype, device, wd_on, tp_sharded, dp, dpr,
                            zero, unit=unit)
                b.expert_parallel = ep
                self._buckets.append((gi, b))
        # DP replicas must start bit-identical; broadcast once from dp rank 0
        # (expert buckets are INTENTIONALLY different per EP(=DP) rank)
        if dp > 1 and dist.is_initialized():
            for _, b in self._buckets:
                if getattr(b, "expert_parallel", False):
                    continue
                src_rank = dist.get_global_rank(dutil.data_parallel_group, 0)
                dist.broadcast(b.flat_param, src=src_rank,
                               group=dutil.data_parallel_group)
        if zero == 3:
            for _, b in self._buckets:
                b.finalize_zero3()

    @property
    def buckets(self):
        if self._buckets is None:
            self._build_buckets()
        return self._buckets

    def zero_grad(self, set_to_none=False):
        for _, b in self.buckets:
            if b.zero == 3:
                b.shard_grad.zero_()  # flat grads are zeroed on materialize
            else:
                b.flat_grad.zero_()

    @torch.no_grad()
    def resync_masters(self):
        """Refresh the fp32 masters from the (possibly just-loaded) bf16
        params.  Buckets alias ``p.data`` into ``flat_param``, so a
        weights-only checkpoint load updates ``flat_param`` but not
        ``flat_master`` — without this, the first step() would rewrite the
        loaded weights from the stale random-init masters.  The Checkpointer
        calls this after every model-weight load; a subsequent optimizer
        state load simply overwrites the masters again."""
        if self._buckets is None:
            return  # not built yet: masters will be created from current params
        for _, b in self._buckets:
            if b.dtype != torch.float32:
                b.flat_master.copy_(b._upd_param().float())
            # fp32 masters alias flat_param (non-ZeRO) or its owned slice

    # -- DP gradient communication (all-reduce / ZeRO reduce-scatter) --------

    def _sync_sp_grads(self, dutil):
        """Sequence parallelism: LN weights/biases and row-linear biases in
        the SP region see only this rank's seq-shard of tokens — their grads
        are PARTIAL and must be summed over TP (Megatron's
        sequence_parallel param marking)."""
        if dutil.tensor_parallel_size == 1 or not dist.is_initialized():
            return
        for _, b in self.buckets:
            for p in b.params:
                if getattr(p, "sequence_parallel_grad", False) and \
                        p.grad is not None:
                    dist.all_reduce(p.grad.data,
                                    group=dutil.tensor_parallel_group)

    def grad_sync(self):
        """Average gradients over the DP group; with ZeRO-2, reduce-scatter so
        only the local slice is received.  If comm/compute overlap is active
        (register_overlap_hooks + begin_overlap_step), this just drains the
        in-flight chunk all-reduces and handles stragglers."""
        dutil = du.get_dist_util()
        self._sync_sp_grads(dutil)  # BEFORE any DP reduce (full views valid)
        dp = dutil.data_parallel_size
        if dp == 1 or not dist.is_initialized():
            return
        group = dutil.data_parallel_group
        if getattr(self, "_overlap_active", False):
            self._finish_overlap(group, dp)
            return
        if self._zero_eff == 3:
            # the backward hooks reduce-scatter each unit as its grads land;
            # finish stragglers (buckets whose pending set never emptied)
            # and release any still-materialized params
            for _, b in self.buckets:
                if b.grads_live():
                    b.reduce_release_grads(group, dp)
                if b.params_live():
                    b.release_params()
            return
        for _, b in self.buckets:
            if getattr(b, "expert_parallel", False):
                # complete already (tokens arrived via all-to-all), but the
                # per-rank losses are means — match the dense params' DP
                # average normalization without a collective
                b.flat_grad.div_(dp)
                continue
            b.flat_grad.div_(dp)
            if self._zero_eff >= 2:
                out = b._upd_grad()
                dist.reduce_scatter_tensor(out, b.flat_grad, group=group)
            else:
                dist.all_reduce(b.flat_grad, group=group)

    # -- backward/comm overlap (bucketed async all-reduce, C5+C11) ----------
    #
    # Flat buckets are split into ~32M-element comm chunks following model
    # order; a post-accumulate-grad hook fires the chunk's async all-reduce as
    # soon as its last parameter's gradient lands, overlapping the remaining
    # backward.  Enabled per-step by the trainer (the LAST micro-batch only,
    # so gradient accumulation sees full sums).  ZeRO-2 keeps the simpler
    # post-backward reduce-scatter.

    OVERLAP_CHUNK = 32 * 1024 * 1024  # elements (= 64 MB bf16 per collective)

    def register_overlap_hooks(self, bucket_mb=None):
        """``bucket_mb`` overrides the per-collective fusion size -- the
        explicit analog of the reference's NCCL fusion-threshold knob
        (engine/default.py:194-199 / C11): bigger buckets amortize the
        xGMI ring latency, smaller ones start overlapping earlier in the
        backward.  Env LIBAI_BUCKET_MB / LIBAI_NO_OVERLAP also apply."""
        import os

        dutil = du.get_dist_util()
        if (dutil.data_parallel_size == 1 or not dist.is_initialized()
                or self.zero_stage >= 2
                or os.environ.get("LIBAI_NO_OVERLAP", "0") == "1"):
            return False
        mb = bucket_mb or os.environ.get("LIBAI_BUCKET_MB")
        if mb:
            # elements assuming 2-byte grads (bf16); fp32 buckets just fuse
            # half as many bytes per collective
            self.OVERLAP_CHUNK = max(1, int(float(mb) * 1024 * 1024 // 2))
        self._overlap_active = False
        self._chunks = []  # (bucket, start, end, param_ids)
        self._param_chunk = {}
        for _, b in self.buckets:
            if getattr(b, "expert_parallel", False):
                continue  # no DP collective for expert grads
            off = 0
            cur_params, cur_start = [], 0
            for p in b.params:
                cur_params.append(id(p))
                off += p.numel()
                if off - cur_start >= self.OVERLAP_CHUNK:
                    self._chunks.append([b, cur_start, off, set(cur_params)])
                    cur_start, cur_params = off, []
            if cur_params:
                self._chunks.append([b, cur_start, off, set(cur_params)])
        for ci, (bkt, s, e, pids) in enumerate(self._chunks):
            for p in bkt.params:
                if id(p) in pids:
                    self._param_chunk[id(p)] = ci
        self._pending = [set(c[3]) for c in self._chunks]
        self._handles = {}

        def make_hook():
            def hook(p):
                if not self._overlap_active:
                    return
                ci = self._param_chunk.get(id(p))
                if ci is None:
                    return
                pend = self._pending[ci]
                pend.discard(id(p))
                if not pend and ci not in self._handles:
                    dutil2 = du.get_dist_util()
                    bkt, s, e, _ = self._chunks[ci]
                    view = bkt.flat_grad[s:e]
                    view.div_(dutil2.data_parallel_size)
                    self._handles[ci] = dist.all_reduce(
                        view, group=dutil2.data_parallel_group, async_op=True
                    )
            return hook

        for _, b in self.buckets:
            for p in b.params:
                p.register_post_accumulate_grad_hook(make_hook())
        return True

    def begin_overlap_step(self):
        """Arm the hooks for the FINAL micro-batch's backward."""
        if not hasattr(self, "_chunks"):
            return
        self._pending = [set(c[3]) for c in self._chunks]
        self._handles = {}
        self._overlap_active = True

    def _finish_overlap(self, group, dp):
        for ci, (bkt, s, e, _) in enumerate(self._chunks):
            if ci not in self._handles:  # straggler (param without grad flow)
                view = bkt.flat_grad[s:e]
                view.div_(dp)
                self._handles[ci] = dist.all_reduce(view, group=group,
                                                    async_op=True)
        for h in self._handles.values():
            h.wait()
        self._overlap_active = False
        self._handles = {}

    def _gather_params(self):
        dutil = du.get_dist_util()
        if self._zero_eff in (0, 3) or dutil.data_parallel_size == 1:
            return  # stage 3 re-gathers lazily at the next forward
        group = dutil.data_parallel_group
        for _, b in self.buckets:
            dist.all_gather_into_tensor(b.flat_param, b._upd_param().contiguous(),
                                        group=group)

    # -- ZeRO-3 whole-model materialization (checkpointing / export) --------

    def materialize_all_params(self):
        """Gather every stage-3 bucket's full params (p.data becomes readable
        again).  No-op outside stage 3."""
        if getattr(self, "_zero_eff", None) != 3:
            return
        group = du.get_dist_util().data_parallel_group
        for _, b in self.buckets:
            b.materialize_params(group)

    def release_all_params(self):
        if getattr(self, "_zero_eff", None) != 3:
            return
        for _, b in self.buckets:
            b.release_params()

    @torch.no_grad()
    def refresh_shards_from_params(self):
        """After an external write into p.data (checkpoint load), re-snapshot
        the owned shards + masters from the materialized flat params."""
        if getattr(self, "_zero_eff", None) != 3:
            return
        for _, b in self.buckets:
            b.shard_param.copy_(
                b.flat_param[b.shard_off : b.shard_off + b.shard])
            if b.dtype != torch.float32:
                b.flat_master.copy_(b.shard_param.float())

    # -- grad norm / clip ---------------------------------------------------

    def _grad_norm_sq(self):
        dutil = du.get_dist_util()
        device = self.buckets[0][1].flat_grad.device
        total = torch.zeros(1, dtype=torch.float32, device=device)
        use_hip = device.type == "cuda" and has_ext()
        chunk = ext().adamw_chunk_elems() if use_hip else 0
        total_ep = torch.zeros(1, dtype=torch.float32, device=device)
        for _, b in self.buckets:
            if not b.tp_sharded and dutil.tensor_parallel_rank != 0:
                continue  # replicated grads counted once per TP group
            dst = total_ep if getattr(b, "expert_parallel", False) else total
            if use_hip:
                ext().l2norm_sq(b.norm_desc(chunk), b.dtype == torch.bfloat16, dst)
            else:
                dst += b._upd_grad().float().pow(2).sum()
        if dist.is_initialized():
            # structural (rank-symmetric) decision — a value-dependent branch
            # could diverge across ranks and wedge the collective
            has_ep = any(getattr(b, "expert_parallel", False)
                         for _, b in self.buckets)
            if has_ep and dutil.data_parallel_size > 1:
                # expert grads are per-EP-rank unique: sum their norms over DP
                dist.all_reduce(total_ep, group=dutil.data_parallel_group)
            total += total_ep
            if self._zero_eff > 0 and dutil.data_parallel_size > 1:
                dist.all_reduce(total, group=dutil.data_parallel_group)
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
        if self.clip_grad and self.clip_grad > 0:
            norm = self._grad_norm_sq().sqrt()
            grad_scale = float((self.clip_grad / (norm + 1e-6)).clamp(max=1.0).item())

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
                g = b._upd_grad().float() * grad_scale
                b.exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
                b.exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                mhat = b.exp_avg / bc1
                vhat = b.exp_avg_sq / bc2
                b.flat_master.add_(
                    mhat / (vhat.sqrt() + eps) + wd * b.flat_master, alpha=-lr
                )
                if b.dtype != torch.float32:
                    b._upd_param().copy_(b.flat_master.to(b.dtype))
                elif self._zero_eff > 0:
                    b._upd_param().copy_(b.flat_master)
        self._gather_params()
        return loss

    # -- state dict (topology-independent per-param tensors) ----------------

    def _full_state(self, b):
        """(master, exp_avg, exp_avg_sq) covering the WHOLE bucket (gathers
        the DP shards under ZeRO)."""
        dutil = du.get_dist_util()
        if self._zero_eff == 0 or dutil.data_parallel_size == 1:
            return b.flat_master, b.exp_avg, b.exp_avg_sq
        group = dutil.data_parallel_group
        outs = []
        for t in (b.flat_master, b.exp_avg, b.exp_avg_sq):
            full = torch.empty(b.numel, dtype=torch.float32, device=t.device)
            dist.all_gather_into_tensor(full, t.contiguous(), group=group)
            outs.append(full)
        return outs

    def set_param_names(self, named_parameters):
        """Register param names so optimizer checkpoints match by NAME:
        resuming on a different pipeline split (different local param list)
        then restores state for every overlapping parameter instead of
        relying on positional order."""
        self._param_names = {id(p): n for n, p in named_parameters}

    @staticmethod
    def _tp_gather_param_state(p, flat):
        """TP-sharded param state -> canonical full tensor (flattened).

        Makes the optimizer checkpoint TP-topology-independent like the
        model weights (reference: OneFlow global-tensor save)."""
        if not (getattr(p, "tensor_parallel", False) and dist.is_initialized()):
            return flat.clone(), list(p.shape)
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        if tp == 1:
            return flat.clone(), list(p.shape)
        from ..layers.linear import tp_merge

        local = flat.view(p.shape)
        shards = [torch.empty_like(local) for _ in range(tp)]
        dist.all_gather(shards, local.contiguous(),
                        group=dutil.tensor_parallel_group)
        full = tp_merge(shards, getattr(p, "tp_shard_dim", 0),
                        getattr(p, "tp_fused_chunks", 1))
        return full.reshape(-1), list(full.shape)

    @staticmethod
    def _ep_gather_param_state(p, flat, shape):
        """Expert-parallel param state -> full expert dim over DP(=EP),
        making the optimizer checkpoint EP-topology-independent."""
        if not (getattr(p, "expert_parallel", False) and dist.is_initialized()):
            return flat, shape
        dutil = du.get_dist_util()
        dp = dutil.data_parallel_size
        if dp == 1:
            return flat, shape
        local = flat.view(shape)
        shards = [torch.empty_like(local) for _ in range(dp)]
        dist.all_gather(shards, local.contiguous(),
                        group=dutil.data_parallel_group)
        full = torch.cat(shards, dim=0)
        return full.reshape(-1), list(full.shape)

    @staticmethod
    def _ep_slice_param_state(p, full_flat, full_shape):
        if not (getattr(p, "expert_parallel", False) and dist.is_initialized()):
            return full_flat, full_shape
        dutil = du.get_dist_util()
        dp = dutil.data_parallel_size
        if dp == 1 or list(full_shape) == list(p.shape):
            return full_flat, full_shape
        if full_shape[0] == p.shape[0] * dp:
            sl = full_flat.view(full_shape).chunk(dp, dim=0)[
                dutil.data_parallel_rank]
            return sl.reshape(-1), list(p.shape)
        return full_flat, full_shape

    @staticmethod
    def _tp_slice_param_state(p, full_flat, full_shape):
        if not (getattr(p, "tensor_parallel", False) and dist.is_initialized()):
            return full_flat
        dutil = du.get_dist_util()
        tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
        if tp == 1 or list(full_shape) == list(p.shape):
            return full_flat
        from ..layers.linear import tp_slice

        full = full_flat.view(full_shape)
        return tp_slice(full, tp, tpr, getattr(p, "tp_shard_dim", 0),
                        getattr(p, "tp_fused_chunks", 1)).reshape(-1)

    def state_dict(self):
        per_param = []
        for gi, b in self.buckets:
            master, m, v = self._full_state(b)
            off = 0
            for p in b.params:
                n = p.numel()
                mast, full_shape = self._tp_gather_param_state(p, master[off:off + n])
                ea, _ = self._tp_gather_param_state(p, m[off:off + n])
                es, _ = self._tp_gather_param_state(p, v[off:off + n])
                mast, full_shape = self._ep_gather_param_state(p, mast, full_shape)
                ea, _ = self._ep_gather_param_state(p, ea, list(p.shape))
                es, _ = self._ep_gather_param_state(p, es, list(p.shape))
                per_param.append(
                    {
                        "master": mast,
                        "exp_avg": ea,
                        "exp_avg_sq": es,
                        "shape": full_shape,
                        "name": self._param_names.get(id(p)),
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
        saved = state_dict["per_param"]
        by_name = {e["name"]: e for e in saved if e.get("name")}
        # Trust positional order only for legacy checkpoints without names:
        # with names present, per_param[idx] can belong to a DIFFERENT param
        # (pipeline-resplit resume changes the local param list), so unmatched
        # params keep fresh state instead of loading a stranger's.
        positional_ok = not by_name
        logger = logging.getLogger(__name__)
        idx = 0
        for gi, b in self.buckets:
            device = b.flat_param.device
            if b.zero == 3:  # collective: all DP ranks load together
                b.materialize_params(du.get_dist_util().data_parallel_group)
            # start masters from CURRENT weights so unmatched params keep
            # their values (not zeros) when flat_param is rewritten below
            master = b.flat_param.detach().float().clone()
            m = torch.zeros_like(master)
            v = torch.zeros_like(master)
            off = 0
            for p in b.params:
                n = p.numel()
                name = self._param_names.get(id(p))
                if name is not None and name in by_name:
                    entry = by_name[name]
                elif positional_ok and idx < len(saved):
                    entry = saved[idx]
                else:
                    logger.warning(
                        f"optimizer checkpoint has no state for {name!r}; "
                        "keeping fresh state"
                    )
                    idx += 1
                    off += n
                    continue
                shp = entry.get("shape", list(p.shape))
                em, shp_e = self._ep_slice_param_state(p, entry["master"], shp)
                ea2, _ = self._ep_slice_param_state(p, entry["exp_avg"], shp)
                es2, _ = self._ep_slice_param_state(p, entry["exp_avg_sq"], shp)
                master[off : off + n].copy_(
                    self._tp_slice_param_state(p, em, shp_e))
                m[off : off + n].copy_(
                    self._tp_slice_param_state(p, ea2, shp_e))
                v[off : off + n].copy_(
                    self._tp_slice_param_state(p, es2, shp_e))
                idx += 1
                off += n
            sl = slice(b.shard_off, b.shard_off + b.shard) if self._zero_eff > 0 \
                else slice(0, b.numel)
            b.flat_master.copy_(master[sl])
            b.exp_avg.copy_(m[sl])
            b.exp_avg_sq.copy_(v[sl])
            if b.zero == 3:
                # full flat_param storage is released at rest; write the
                # persistent shard (next forward gathers the fresh values)
                b.shard_param.copy_(master[sl].to(b.dtype))
                b.release_params()
            elif b.dtype != torch.float32:
                b.flat_param.copy_(master.to(b.dtype))
            else:
                b.flat_param.copy_(master)
