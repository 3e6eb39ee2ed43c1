"""hipGraph-captured greedy decode (serving hot loop).

The per-token decode step is LAUNCH-BOUND: ~9 kernels per layer x 24 layers
at ~5 us each of launch overhead.  This module captures ONE fully static
decode step — embedding lookup at a device position tensor, per-layer
in-place KV insert + fused flash_decode (K16), greedy argmax, token/position
advance — into a single hipGraph and replays it per token with ZERO host
work and zero synchronisation inside the loop.

Static-shape contract (what makes the step capturable):
  - KV caches preallocated at [b, kv_heads, max_len, hs]; writes via
    index_copy_ at a DEVICE position tensor (no host ints in the step)
  - per-batch kv_len is a device int32 tensor the flash_decode kernel masks
    by, so the grid is sized once for max_len
  - RoPE (Llama) gathers its cos/sin rows by index_select at the device
    position instead of a host slice
  - the argmax result is copied back into the static input token buffer and
    recorded into a preallocated output buffer at a device step counter, so
    the graph is fully self-advancing: N replays = N tokens

Reference capability: libai/inference/generator/generation_utils.py greedy
search + oneflow nn.Graph capture of the decode step; rebuilt here on
torch.cuda.CUDAGraph (hipGraph on ROCm) + the K16 decode kernel.

GPU-only (bf16, head_size in {64, 128}, TP=1): raises on anything else —
no silent eager fallback.  Measured at b32 ctx1024
(profiles/decode_captured.md): GPT-2 345M 3.20 ms/tok captured vs 6.90
eager; Llama-1B (GQA+RoPE via the fused rope_kv_insert kernel) 3.83 vs
4.91.
"""

import torch

__all__ = ["CapturedGPTDecoder", "CapturedLlamaDecoder",
           "SamplingMixin", "CapturedGPTSampler", "CapturedLlamaSampler",
           "ContinuousGPTDecoder", "ContinuousLlamaDecoder",
           "ContinuousGPTSampler", "ContinuousLlamaSampler"]


class _CapturedDecoderBase:
    """Capture-once / replay-per-token greedy decoder.

    Usage::

        dec = CapturedGPTDecoder(model, max_batch=32, max_seq_len=1024)
        tokens = dec.generate(prompt_ids, max_new_tokens=64)  # [b, new]
    """

    def __init__(self, model, max_batch, max_seq_len):
        self.model = model
        attn = self._first_attn(model)
        from ..utils import distributed as du

        if du.get_dist_util().tensor_parallel_size != 1:
            raise RuntimeError("captured decode supports TP=1 (shard the "
                               "batch over DP ranks for captured serving)")
        head_size, n_kv = self._head_geom(attn)
        if head_size not in (64, 128):
            raise RuntimeError(f"flash_decode needs head_size in (64,128), "
                               f"got {head_size}")
        self.max_batch = max_batch
        self.max_seq_len = max_seq_len
        dev = next(model.parameters()).device
        if dev.type != "cuda":
            raise RuntimeError("captured decode is GPU-only")
        self.device = dev
        b = max_batch
        # ONE shared kv_len tensor (all layers decode in lockstep): its
        # advance is a single captured kernel, not one per layer
        self.kv32 = torch.zeros(b, dtype=torch.int32, device=dev)
        self.caches = [
            (torch.zeros(b, n_kv, max_seq_len, head_size, device=dev,
                         dtype=torch.bfloat16),
             torch.zeros(b, n_kv, max_seq_len, head_size, device=dev,
                         dtype=torch.bfloat16),
             self.kv32)
            for _ in range(self._n_layers(model))
        ]
        self.pos = torch.zeros(1, dtype=torch.int64, device=dev)
        self.step_idx = torch.zeros(1, dtype=torch.int64, device=dev)
        self.static_tok = torch.zeros(b, 1, dtype=torch.int64, device=dev)
        self.out_tokens = None  # sized per generate() call
        self.graph = None

    # -- model adapters (subclass hooks) ----------------------------------
    def _first_attn(self, model):
        raise NotImplementedError

    def _head_geom(self, attn):
        """-> (head_size, n_kv_heads_local)"""
        raise NotImplementedError

    def _n_layers(self, model):
        raise NotImplementedError

    def _logits(self):
        """Run one static step on self.static_tok -> [b, 1, vocab]."""
        raise NotImplementedError

    # -- token selection (overridden by the sampling mixin) ---------------
    def _select(self, logits):
        return logits.argmax(dim=-1, keepdim=True)  # greedy

    # -- the one decode step (the thing that gets captured) ---------------
    def _step(self):
        logits = self._logits()
        nxt = self._select(logits[:, -1, :])  # [b, 1]
        self.out_tokens.index_copy_(1, self.step_idx, nxt)
        self.static_tok.copy_(nxt)
        self.pos.add_(1)
        self.step_idx.add_(1)
        self.kv32.add_(1)

    def _reset_to(self, prompt_len, first_tok):
        self.pos.fill_(prompt_len)
        self.step_idx.fill_(1)
        self.static_tok.copy_(first_tok)
        self.kv32.fill_(prompt_len + 1)  # length AFTER this step's insert

    @torch.no_grad()
    def generate(self, prompt_ids, max_new_tokens):
        """Greedy-decode ``max_new_tokens`` tokens after ``prompt_ids``
        ([b, L] int64, b == max_batch, L + new <= max_seq_len).
        Returns the generated tokens [b, max_new_tokens]."""
        b, L = prompt_ids.shape
        assert b == self.max_batch, "captured graph is shape-static: pad the batch"
        assert L + max_new_tokens <= self.max_seq_len
        self.model.eval()

        # 1) eager prefill (one big flash_fwd pass), fill the static caches
        out = self.model(input_ids=prompt_ids.to(self.device), use_cache=True)
        past = out["past_key_values"]
        logits = out["prediction_scores"]
        for (ck, cv, _), (pk, pv) in zip(self.caches, past):
            ck[:, :, :L].copy_(pk)
            cv[:, :, :L].copy_(pv)
        first = self._select(logits[:, -1, :])  # token at pos L

        if self.out_tokens is None or self.out_tokens.shape[1] < max_new_tokens:
            self.out_tokens = torch.zeros(self.max_batch, max_new_tokens,
                                          dtype=torch.int64, device=self.device)
            self.graph = None  # out buffer is graph-captured state
        self.out_tokens[:, 0:1].copy_(first)

        if max_new_tokens == 1:
            return self.out_tokens[:, :1].clone()

        if self.graph is None:
            # warmup twice on a side stream (allocator + library init),
            # then reset the advance state and capture
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                self._reset_to(L, first)
                self._step()
                self._step()
            torch.cuda.current_stream().wait_stream(s)
            self._reset_to(L, first)
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self._step()
        else:
            self._reset_to(L, first)

        # 2) replay: each replay appends one token, no host work at all
        for _ in range(max_new_tokens - 1):
            self.graph.replay()
        return self.out_tokens[:, :max_new_tokens].clone()


class CapturedGPTDecoder(_CapturedDecoderBase):
    """Captured decode for GPTForPreTraining / GPTModel (learned positions,
    MHA, tied lm head)."""

    def _first_attn(self, model):
        self.gpt = model.GPT_model if hasattr(model, "GPT_model") else model
        return self.gpt.transformer.layers[0].self_attention

    def _head_geom(self, attn):
        return attn.head_size, attn.num_heads_local

    def _n_layers(self, model):
        return len(self.gpt.transformer.layers)

    def _logits(self):
        return self.gpt(self.static_tok, static_caches=self.caches,
                        position=self.pos)


class CapturedLlamaDecoder(_CapturedDecoderBase):
    """Captured decode for LlamaForCausalLM (RoPE at a device position,
    GQA KV caches at num_kv_heads)."""

    def _first_attn(self, model):
        return model.model.layers[0].self_attn

    def _head_geom(self, attn):
        return attn.head_dim, attn.num_kv_local

    def _n_layers(self, model):
        return len(model.model.layers)

    def _logits(self):
        return self.model(self.static_tok, static_caches=self.caches,
                          position=self.pos)


class SamplingMixin:
    """Temperature / top-k sampling INSIDE the captured graph.

    torch's philox generator is hipGraph-aware (the RNG offset lives in
    graph-owned device memory), so `torch.multinomial` replays with fresh
    randomness each replay — the whole sampled decode loop stays on-device.
    Set temperature/top_k before the first generate() (they are captured).
    """

    temperature = 1.0
    top_k = 0  # 0 = no top-k filter

    def _select(self, logits):
        logits = logits.float()
        if self.temperature != 1.0:
            logits = logits / self.temperature
        if self.top_k:
            kth = logits.topk(self.top_k, dim=-1).values[..., -1:]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        probs = torch.softmax(logits, dim=-1)
        return torch.multinomial(probs, 1)


class CapturedGPTSampler(SamplingMixin, CapturedGPTDecoder):
    def __init__(self, model, max_batch, max_seq_len, temperature=1.0,
                 top_k=0):
        super().__init__(model, max_batch, max_seq_len)
        self.temperature = temperature
        self.top_k = top_k


class CapturedLlamaSampler(SamplingMixin, CapturedLlamaDecoder):
    def __init__(self, model, max_batch, max_seq_len, temperature=1.0,
                 top_k=0):
        super().__init__(model, max_batch, max_seq_len)
        self.temperature = temperature
        self.top_k = top_k


class _ContinuousMixin:
    """Continuous batching on the captured loop: each slot decodes at its
    OWN position (per-batch device pos/kv_len; the rope_kv_insert kernel and
    flash_decode both index them per slot), so requests of different lengths
    share one replayed graph and new requests are admitted between replays
    without recapturing.

    Protocol::

        dec = ContinuousGPTDecoder(model, max_batch=8, max_seq_len=1024)
        dec.add_request(0, prompt_a)           # eager prefill into slot 0
        dec.step(16)                           # 16 graph replays
        dec.add_request(1, prompt_b)           # admit mid-flight
        dec.step(16)
        toks_a = dec.tokens(0)                 # generated tokens so far
        dec.release(0)                         # free the slot

    Generated tokens land in a per-slot ring buffer (``ring_cap`` entries):
    harvest at least every ring_cap steps.  EOS/length policy is the
    caller's (check tokens(), then release()).  Idle slots decode a parked
    dummy (pos 0, kv_len 1) whose outputs are ignored.
    """

    ring_cap = 64

    def _init_cont(self):
        B, dev = self.max_batch, self.device
        self.pos = torch.zeros(B, dtype=torch.int64, device=dev)  # per-slot
        self.slot_step = torch.zeros(B, 1, dtype=torch.int64, device=dev)
        self.ring = torch.zeros(B, self.ring_cap, dtype=torch.int64,
                                device=dev)
        self.kv32.fill_(1)
        self.graph = None

    def _step(self):
        logits = self._logits()
        nxt = self._select(logits[:, -1, :])  # [b, 1]
        self.ring.scatter_(1, self.slot_step.remainder(self.ring_cap), nxt)
        self.static_tok.copy_(nxt)
        self.pos.add_(1)
        self.kv32.add_(1)
        self.slot_step.add_(1)

    def _park_all(self):
        self.pos.zero_()
        self.kv32.fill_(1)
        self.slot_step.zero_()
        self.static_tok.zero_()

    def _capture(self):
        # warm up on the LIVE state (admissions may already have happened):
        # the two warmup steps scribble k/v at each slot's current pos and
        # pos+1, which the first two real replays rewrite before kv_len
        # covers them — so only the small advance tensors need restoring
        snap = [t.clone() for t in (self.pos, self.kv32, self.slot_step,
                                    self.static_tok, self.ring)]
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._step()
            self._step()
        torch.cuda.current_stream().wait_stream(s)
        for t, sv in zip((self.pos, self.kv32, self.slot_step,
                          self.static_tok, self.ring), snap):
            t.copy_(sv)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._step()

    @torch.no_grad()
    def add_request(self, slot, prompt_ids):
        """Eager-prefill ``prompt_ids`` (1-D int64) into ``slot``; the
        prompt's first generated token lands in the ring at index 0."""
        self.model.eval()
        prompt_ids = prompt_ids.view(1, -1).to(self.device)
        L = prompt_ids.shape[1]
        assert L + 1 < self.max_seq_len
        out = self.model(input_ids=prompt_ids, use_cache=True)
        for (ck, cv, _), (pk, pv) in zip(self.caches, out["past_key_values"]):
            ck[slot, :, :L].copy_(pk[0])
            cv[slot, :, :L].copy_(pv[0])
        first = self._select(out["prediction_scores"][:, -1, :])
        self.ring[slot, 0] = first[0, 0]
        self.static_tok[slot, 0] = first[0, 0]
        self.pos[slot] = L
        self.kv32[slot] = L + 1
        self.slot_step[slot] = 1

    def release(self, slot):
        self.pos[slot] = 0
        self.kv32[slot] = 1
        self.slot_step[slot] = 0

    @torch.no_grad()
    def step(self, n=1):
        """Replay the captured decode step ``n`` times (every active slot
        gains ``n`` tokens; idle slots burn a parked dummy row)."""
        if self.graph is None:
            self._capture()
        for _ in range(n):
            self.graph.replay()

    def tokens(self, slot):
        """Generated tokens for ``slot`` so far (host sync; ring must not
        have wrapped — harvest at least every ring_cap steps)."""
        n = int(self.slot_step[slot])
        if n > self.ring_cap:
            raise RuntimeError(
                f"slot {slot} generated {n} > ring_cap={self.ring_cap} "
                "tokens since admission; harvest more often")
        return self.ring[slot, :n].clone()

    def generate(self, *a, **k):
        raise RuntimeError("continuous decoder: use add_request/step/tokens")

    @torch.no_grad()
    def serve(self, prompts, max_new_tokens, eos_id=None, chunk=16):
        """Drive a full request list through the slots: admit as slots
        free up, harvest every ``chunk`` replays, trim at ``eos_id``.
        Returns a list of 1-D token tensors aligned with ``prompts``."""
        assert max_new_tokens <= self.ring_cap, "raise ring_cap"
        results = [None] * len(prompts)
        pending = list(range(len(prompts)))[::-1]
        active = {}  # slot -> request index

        def admit():
            for slot in range(self.max_batch):
                if slot not in active and pending:
                    r = pending.pop()
                    self.add_request(slot, prompts[r])
                    active[slot] = r

        admit()
        while active:
            self.step(chunk)
            for slot, r in list(active.items()):
                toks = self.tokens(slot)
                done = toks.numel() >= max_new_tokens
                if eos_id is not None:
                    hit = (toks == eos_id).nonzero()
                    if hit.numel():
                        toks = toks[: int(hit[0, 0]) + 1]
                        done = True
                if done or int(self.pos[slot]) + chunk + 2 >= self.max_seq_len:
                    results[r] = toks[:max_new_tokens]
                    self.release(slot)
                    del active[slot]
            admit()
        return results


class ContinuousGPTDecoder(_ContinuousMixin, CapturedGPTDecoder):
    def __init__(self, model, max_batch, max_seq_len, ring_cap=64):
        super().__init__(model, max_batch, max_seq_len)
        self.ring_cap = ring_cap
        self._init_cont()


class ContinuousLlamaDecoder(_ContinuousMixin, CapturedLlamaDecoder):
    def __init__(self, model, max_batch, max_seq_len, ring_cap=64):
        super().__init__(model, max_batch, max_seq_len)
        self.ring_cap = ring_cap
        self._init_cont()


class ContinuousGPTSampler(SamplingMixin, ContinuousGPTDecoder):
    # continuous batching with in-graph top-k/temperature sampling
    def __init__(self, model, max_batch, max_seq_len, ring_cap=64,
                 temperature=1.0, top_k=0):
        super().__init__(model, max_batch, max_seq_len, ring_cap=ring_cap)
        self.temperature = temperature
        self.top_k = top_k


class ContinuousLlamaSampler(SamplingMixin, ContinuousLlamaDecoder):
    def __init__(self, model, max_batch, max_seq_len, ring_cap=64,
                 temperature=1.0, top_k=0):
        super().__init__(model, max_batch, max_seq_len, ring_cap=ring_cap)
        self.temperature = temperature
        self.top_k = top_k
