"""Expert-parallel Mixture-of-Experts FFN (top-k routing, all-to-all
dispatch over the DP/EP group).

BEYOND the reference's feature set (SURVEY §2.5 marks EP/MoE absent
upstream); the MI355X brief's RCCL axis list names EP explicitly — on a
node, the token exchange is two ``all_to_all_single`` calls per layer over
xGMI point-to-point links.

Design (correctness-first, dropless):
  * EP group == the DATA-parallel group: rank r owns experts
    [r*E/ep, (r+1)*E/ep).  Expert weights are drawn from a FULL [E, ...]
    init and sliced, so every EP layout computes the same function.
  * Router (replicated linear) picks top-k experts per token; gates are
    the softmax over the top-k logits.  Aux load-balancing loss =
    E * sum_i f_i * P_i (Switch/GShard), exposed as ``last_aux_loss``.
  * Dispatch: tokens are permuted by destination rank, exchanged with
    all_to_all_single (uneven splits; the counts travel in a small int64
    all_to_all first), run through the local experts, exchanged back, and
    combined with their gates.  The exchange is an autograd.Function whose
    backward is the reverse exchange.
  * Expert params carry ``expert_parallel=True``: FusedAdamW buckets them
    separately, skips the DP grad all-reduce (each rank's expert grads are
    already complete), skips the DP weight broadcast, and DP-reduces their
    grad-norm contribution.
"""

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch import nn

from ..utils import distributed as du

__all__ = ["MoELayer"]


class _AllToAll(torch.autograd.Function):
    """Uneven all_to_all_single; backward reverses the exchange."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.splits = (out_splits, in_splits)
        ctx.group = group
        if group is None or not dist.is_initialized():
            return x
        out = x.new_empty(sum(out_splits), *x.shape[1:])
        dist.all_to_all_single(out, x.contiguous(), out_splits, in_splits,
                               group=group)
        return out

    @staticmethod
    def backward(ctx, grad):
        out_splits, in_splits = ctx.splits
        if ctx.group is None or not dist.is_initialized():
            return grad, None, None, None
        back = grad.new_empty(sum(in_splits), *grad.shape[1:])
        dist.all_to_all_single(back, grad.contiguous(), in_splits, out_splits,
                               group=ctx.group)
        return back, None, None, None


class MoELayer(nn.Module):
    def __init__(self, hidden_size, ffn_hidden_size, num_experts, top_k=2,
                 activation="gelu", init_method=nn.init.xavier_normal_,
                 output_layer_init_method=None, aux_loss_coef=0.01,
                 *, layer_idx=0):
        super().__init__()
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        assert tp == 1, "MoE v1 runs with tensor_parallel_size == 1"
        self.ep = dutil.data_parallel_size
        self.ep_rank = dutil.data_parallel_rank
        assert num_experts % max(self.ep, 1) == 0, (num_experts, self.ep)
        self.num_experts = num_experts
        self.experts_per_rank = num_experts // max(self.ep, 1)
        self.top_k = top_k
        self.hidden_size = hidden_size
        self.ffn = ffn_hidden_size
        self.activation = activation
        self.aux_loss_coef = aux_loss_coef
        self.layer_idx = layer_idx
        self.last_aux_loss = None
        output_layer_init_method = output_layer_init_method or init_method

        self.router = nn.Linear(hidden_size, num_experts, bias=False)
        init_method(self.router.weight.data)

        # experts: slice of the FULL [E, ...] init (EP-layout independent)
        def full_slice(shape_full, init):
            full = torch.empty(*shape_full)
            for e in range(shape_full[0]):
                init(full[e])
            lo = self.ep_rank * self.experts_per_rank
            return full[lo : lo + self.experts_per_rank].clone()

        self.w1 = nn.Parameter(
            full_slice((num_experts, ffn_hidden_size, hidden_size), init_method))
        self.b1 = nn.Parameter(
            torch.zeros(self.experts_per_rank, ffn_hidden_size))
        self.w2 = nn.Parameter(
            full_slice((num_experts, hidden_size, ffn_hidden_size),
                       output_layer_init_method))
        self.b2 = nn.Parameter(torch.zeros(self.experts_per_rank, hidden_size))
        for p in (self.w1, self.b1, self.w2, self.b2):
            p.expert_parallel = True

    def _expert_ffn(self, x, e):
        h = F.linear(x, self.w1[e], self.b1[e])
        h = F.gelu(h, approximate="tanh") if self.activation == "gelu" \
            else F.silu(h)
        return F.linear(h, self.w2[e], self.b2[e])

    def forward(self, hidden_states, residual=None):
        orig_shape = hidden_states.shape
        x = hidden_states.reshape(-1, self.hidden_size)
        n = x.shape[0]

        # routing decisions in fp32 regardless of the model dtype
        logits = F.linear(x.float(), self.router.weight.float())  # [n, E]
        probs = torch.softmax(logits, dim=-1)
        gates, idx = probs.topk(self.top_k, dim=-1)  # [n, k]
        gates = gates / gates.sum(dim=-1, keepdim=True)

        # Switch/GShard aux loss: E * sum_i (token fraction_i * mean prob_i)
        with torch.no_grad():
            frac = torch.zeros(self.num_experts, device=x.device)
            frac.scatter_add_(0, idx.reshape(-1),
                              torch.full((n * self.top_k,), 1.0 / (n * self.top_k),
                                         device=x.device))
        self.last_aux_loss = self.num_experts * (frac * probs.mean(dim=0)).sum() \
            * self.aux_loss_coef

        # flatten the k copies; sort by destination expert
        flat_idx = idx.reshape(-1)              # [n*k]
        flat_gate = gates.reshape(-1)
        order = torch.argsort(flat_idx, stable=True)
        xk = x.repeat_interleave(self.top_k, dim=0)[order]
        sorted_expert = flat_idx[order]

        counts = torch.bincount(flat_idx, minlength=self.num_experts)
        if self.ep > 1 and dist.is_initialized():
            group = du.get_dist_util().data_parallel_group
            send = counts.reshape(self.ep, self.experts_per_rank).sum(-1)
            recv = torch.empty_like(send)
            dist.all_to_all_single(recv, send.contiguous(), group=group)
            in_splits = send.tolist()
            out_splits = recv.tolist()
            xr = _AllToAll.apply(xk, out_splits, in_splits, group)
            # received tokens: rank-major, expert-sorted within each source;
            # exchange per-(src, local expert) counts to regroup by expert
            cpe = counts.reshape(self.ep, self.experts_per_rank).contiguous()
            cpe_recv = torch.empty_like(cpe)
            dist.all_to_all_single(cpe_recv, cpe, group=group)  # [src, le]
            # order received tokens by local expert: build the permutation
            seg_expert = torch.repeat_interleave(
                torch.arange(self.experts_per_rank, device=x.device)
                .repeat(self.ep), cpe_recv.reshape(-1))
            regroup = torch.argsort(seg_expert, stable=True)
            xr = xr[regroup]
            per_expert = cpe_recv.sum(0)  # tokens per local expert
        else:
            xr = xk
            regroup = None
            per_expert = counts

        # run each local expert on its contiguous token slice
        outs = []
        off = 0
        for e in range(self.experts_per_rank):
            ne = int(per_expert[e])
            outs.append(self._expert_ffn(xr[off : off + ne], e))
            off += ne
        yr = torch.cat(outs, dim=0) if outs else xr[:0]

        if self.ep > 1 and dist.is_initialized():
            inv = torch.empty_like(regroup)
            inv[regroup] = torch.arange(regroup.numel(), device=x.device)
            yr = yr[inv]
            yk = _AllToAll.apply(yr, in_splits, out_splits, group)
        else:
            yk = yr

        # un-sort the k copies and gate-combine
        inv_order = torch.empty_like(order)
        inv_order[order] = torch.arange(order.numel(), device=x.device)
        yk = yk[inv_order]
        y = (yk.reshape(n, self.top_k, self.hidden_size)
             * flat_gate.reshape(n, self.top_k, 1).to(yk.dtype)).sum(dim=1)
        y = y.reshape(orig_shape)
        if residual is not None:
            y = y + residual
        return y
