"""Multi-head attention with TP-sharded heads and fused softmax.

Explicit re-design of the reference MultiheadAttention
(reference: libai/layers/attention.py:88-281):

  * fused QKV column-parallel linear (hidden -> 3*hidden/tp, attention.py:118-124)
  * batched QK^T and PV through rocBLAS (torch.matmul), scores shaped
    [b, nh/tp, s, s]
  * fused scale+causal/padding softmax(+dropout) HIP kernel (replaces
    flow._C.fused_scale_tril_softmax_mask_scale / fused_scale_mask_softmax_dropout,
    attention.py:221-246)
  * row-parallel output projection; its bias/dropout/residual-add is fused by
    the caller (TransformerLayer) via bias_dropout_add, matching the
    reference's fused_bias_add_dropout site (attention.py:265-267)
  * KV-cache incremental decoding (attention.py:201-208) and the
    cross-attention Q + KV split (attention.py:102-116).
"""

import math

import torch
from torch import nn

from ..ops.fused_bias import bias_dropout_add
from ..ops.softmax import fused_scale_mask_softmax

_DUMMY_TABS = {}


def _dummy_tab(device):
    """1-element fp32 placeholder for the no-rotate kv_insert call (the
    kernel never reads cos/sin when rotate=False)."""
    key = str(device)
    if key not in _DUMMY_TABS:
        _DUMMY_TABS[key] = torch.zeros(1, dtype=torch.float32, device=device)
    return _DUMMY_TABS[key]
from ..utils import distributed as du
from .linear import Linear1D

__all__ = ["MultiheadAttention", "AttnMaskType"]


class AttnMaskType:
    padding = "padding"
    causal = "causal"


class MultiheadAttention(nn.Module):
    def __init__(
        self,
        hidden_size,
        num_attention_heads,
        is_cross_attention=False,
        attention_dropout_prob=0.0,
        output_dropout_prob=0.0,
        init_method=nn.init.xavier_normal_,
        output_layer_init_method=None,
        bias_dropout_fusion=True,
        scale_mask_softmax_fusion=True,
        apply_query_key_layer_scaling=False,
        attn_mask_type=AttnMaskType.padding,
        sequence_parallel=False,
        *,
        layer_idx=0,
    ):
        super().__init__()
        self.hidden_size = hidden_size
        self.num_heads = num_attention_heads
        assert hidden_size % num_attention_heads == 0
        self.head_size = hidden_size // num_attention_heads
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        assert num_attention_heads % tp == 0, (num_attention_heads, tp)
        self.num_heads_local = num_attention_heads // tp
        self.attn_mask_type = attn_mask_type
        self.attention_dropout_prob = attention_dropout_prob
        self.output_dropout_prob = output_dropout_prob
        self.layer_idx = layer_idx
        self.is_cross_attention = is_cross_attention
        self.sequence_parallel = sequence_parallel
        assert not (sequence_parallel and is_cross_attention), \
            "sequence parallelism is wired for self-attention blocks only"

        self.norm_factor = 1.0 / math.sqrt(self.head_size)
        self.coeff = None
        if apply_query_key_layer_scaling:
            self.coeff = float(layer_idx + 1)
            self.norm_factor /= self.coeff

        output_layer_init_method = output_layer_init_method or init_method
        if is_cross_attention:
            self.query = Linear1D(hidden_size, hidden_size, parallel="col",
                                  init_method=init_method, layer_idx=layer_idx)
            self.key_value = Linear1D(hidden_size, 2 * hidden_size, parallel="col",
                                      init_method=init_method, layer_idx=layer_idx)
        else:
            self.query_key_value = Linear1D(hidden_size, 3 * hidden_size, parallel="col",
                                            init_method=init_method,
                                            sequence_parallel=sequence_parallel,
                                            layer_idx=layer_idx)
        self.dense = Linear1D(hidden_size, hidden_size, parallel="row",
                              init_method=output_layer_init_method,
                              skip_bias_add=True,
                              sequence_parallel=sequence_parallel,
                              layer_idx=layer_idx)

    def _split_heads(self, x, n):
        # [b, s, n*hs] -> [b, nh_local, s, hs] per chunk
        b, s, _ = x.shape
        x = x.view(b, s, self.num_heads_local, n * self.head_size)
        return x.permute(0, 2, 1, 3).chunk(n, dim=-1)

    def forward(
        self,
        hidden_states,
        encoder_states=None,
        attention_mask=None,
        past_key_value=None,
        use_cache=False,
        residual=None,
        position_bias=None,
        static_cache=None,
        position=None,
    ):
        """attention_mask: [b, sq, sk] bool/uint8, 1 = MASKED (reference semantics).

        ``position_bias``: optional additive score bias broadcastable to
        [b, nh_local, sq, sk] — ALiBi slopes (BLOOM, reference
        projects/BLOOM/modeling/) or T5 relative-position bias (reference
        projects/MT5/layers/attention_layer.py:118-123).  Added AFTER the
        1/sqrt(hs) scaling, matching both conventions.

        If ``residual`` is given, returns hidden + dropout(out + bias) (the
        fused TransformerLayer path); otherwise applies bias+dropout only.
        """
        if static_cache is not None:
            # hipGraph-capturable decode: preallocated [b, nh, MAX, hs] caches,
            # device position/length tensors; ONE kv_insert kernel (no-rotate
            # variant of K17's decode fusion) moves q to flash layout and
            # writes k/v into the cache row at the device position; fused
            # flash_decode masks by per-batch kv_len (all shapes static)
            from ..ops._ext import ext
            from ..ops.attention import flash_decode_attn

            ck, cv, kv_len32 = static_cache
            qkv = self.query_key_value(hidden_states)
            b, s, _ = hidden_states.shape
            nh, hs = self.num_heads_local, self.head_size
            qkv4 = qkv.view(b, s, nh, 3 * hs)
            q, k, v = (qkv4[..., :hs], qkv4[..., hs:2 * hs], qkv4[..., 2 * hs:])
            dummy = _dummy_tab(hidden_states.device)
            qo = ext().rope_kv_insert(q, k, v, ck, cv, dummy, dummy,
                                      position, False)
            scale_c = self.norm_factor * (self.coeff if self.coeff else 1.0)
            ctx = flash_decode_attn(qo, ck, cv, scale_c, kv_len=kv_len32)
            context = ctx.permute(0, 2, 1, 3).reshape(b, 1, nh * hs)
            # returns RAW (out, bias): the layer fuses bias+residual+norm
            return self.dense(context)

        # fused flash path: consumes the qkv buffer with ZERO copies (strided
        # [b, s, nh, hs] views), O lands directly in [b, s, h] layout.
        if (
            not self.is_cross_attention
            and past_key_value is None
            and not use_cache
            and position_bias is None
        ):
            from ..ops.attention import (
                flash_attention_available,
                flash_attention_qkv,
                mask_kv_len,
            )

            b, s, _ = hidden_states.shape
            if self.sequence_parallel:  # input is the seq shard; qkv gathers
                s = s * du.get_dist_util().tensor_parallel_size
            if flash_attention_available(
                self.head_size, hidden_states.dtype, hidden_states.device, s, s,
                attention_mask,
            ):
                qkv = self.query_key_value(hidden_states)
                qkv5 = qkv.view(b, s, self.num_heads_local, 3, self.head_size)
                o = flash_attention_qkv(
                    qkv5,
                    scale=self.norm_factor * (self.coeff if self.coeff else 1.0),
                    p_drop=self.attention_dropout_prob,
                    causal=self.attn_mask_type == AttnMaskType.causal,
                    training=self.training,
                    kv_len=mask_kv_len(attention_mask),
                )
                context = o.reshape(b, s, self.num_heads_local * self.head_size)
                out, bias = self.dense(context)
                return bias_dropout_add(
                    out, bias=bias, residual=residual, p=self.output_dropout_prob,
                    training=self.training,
                )

        if self.is_cross_attention:
            q = self._split_heads(self.query(hidden_states), 1)[0]
            if past_key_value is not None:
                k, v = past_key_value
            else:
                k, v = self._split_heads(self.key_value(encoder_states), 2)
        else:
            q, k, v = self._split_heads(self.query_key_value(hidden_states), 3)
            if past_key_value is not None:
                pk, pv = past_key_value
                k = torch.cat([pk, k], dim=2)
                v = torch.cat([pv, v], dim=2)
        present = (k, v) if use_cache else None

        scale_d = self.norm_factor * (self.coeff if self.coeff else 1.0)
        if (
            past_key_value is not None
            and attention_mask is None
            and position_bias is None
            and not self.is_cross_attention
        ):
            from ..ops.attention import (
                decode_attention_available,
                flash_decode_attn,
            )

            if decode_attention_available(q, self.head_size):
                # fused single-token decode (K16): no [b, nh, 1, skv] scores
                ctx = flash_decode_attn(q.contiguous(), k, v, scale_d)
                b, nh, _, hs = ctx.shape
                context = ctx.permute(0, 2, 1, 3).reshape(b, 1, nh * hs)
                out, bias = self.dense(context)
                out = bias_dropout_add(
                    out, bias=bias, residual=residual,
                    p=self.output_dropout_prob, training=self.training,
                )
                return (out, present) if use_cache else out

        # [b, nh, sq, hs] x [b, nh, hs, sk] -> [b, nh, sq, sk]  (rocBLAS bmm)
        scores = torch.matmul(q, k.transpose(-1, -2))
        # the reference's query_key_layer_scaling splits the scale between the
        # matmul alpha and the softmax input (attention.py:211, :240); in bf16 we
        # fold the whole 1/sqrt(hs) (incl. coeff round-trip = identity) into the
        # fused softmax's scale argument.
        scale = self.norm_factor * (self.coeff if self.coeff else 1.0)
        if position_bias is not None:
            # additive score bias (ALiBi / T5 relative positions): fold the
            # scale here so the fused softmax runs with scale=1
            scores = scores * scale + position_bias
            scale = 1.0
        causal = self.attn_mask_type == AttnMaskType.causal and past_key_value is None
        probs = fused_scale_mask_softmax(
            scores,
            pad_mask=attention_mask,
            scale=scale,
            p=self.attention_dropout_prob,
            causal=causal,
            training=self.training,
        )
        context = torch.matmul(probs, v)  # [b, nh, sq, hs]
        b, nh, sq, hs = context.shape
        context = context.permute(0, 2, 1, 3).reshape(b, sq, nh * hs)

        out, bias = self.dense(context)
        out = bias_dropout_add(
            out, bias=bias, residual=residual, p=self.output_dropout_prob,
            training=self.training,
        )
        if use_cache:
            return out, present
        return out

    def extra_repr(self):
        return (
            f"hidden_size={self.hidden_size}, num_heads={self.num_heads}, "
            f"layer_idx={self.layer_idx}, mask={self.attn_mask_type}"
        )
