"""Host-side scheduling logic of the continuous-batching serve() driver
(admission queueing, chunked harvest, EOS trim, max-token cutoff) — the GPU
decode itself is stubbed so this runs on CPU; the real end-to-end parity
lives in tests/gpu/test_captured_decode_gpu.py."""

import torch

from libai_amd.inference.captured_decode import _ContinuousMixin


class _FakeDecoder(_ContinuousMixin):
    """Deterministic stub: slot s generates tokens 1000*s + step index."""

    def __init__(self, slots, max_seq=100, ring_cap=64):
        self.max_batch = slots
        self.max_seq_len = max_seq
        self.ring_cap = ring_cap
        self.pos = torch.zeros(slots, dtype=torch.int64)
        self._n = [0] * slots
        self._base = [0] * slots
        self._active = set()
        self.admit_log = []

    def add_request(self, slot, prompt):
        assert slot not in self._active
        self._active.add(slot)
        self._n[slot] = 1
        self._base[slot] = 1000 * slot
        self.pos[slot] = len(prompt)
        self.admit_log.append((slot, len(prompt)))

    def step(self, n=1):
        for s in self._active:
            self._n[s] += n
            self.pos[s] += n

    def tokens(self, slot):
        if self._n[slot] > self.ring_cap:
            raise RuntimeError("ring overflow")
        return self._base[slot] + torch.arange(self._n[slot])

    def release(self, slot):
        self._active.discard(slot)
        self.pos[slot] = 0


def test_serve_queues_more_requests_than_slots():
    dec = _FakeDecoder(slots=2)
    prompts = [torch.zeros(5 + i, dtype=torch.int64) for i in range(5)]
    outs = dec.serve(prompts, max_new_tokens=10, chunk=3)
    assert len(outs) == 5
    for i, o in enumerate(outs):
        assert o.numel() == 10
    # the first two admissions are requests 0 and 1; later slots recycle
    assert dec.admit_log[0][0] != dec.admit_log[1][0]
    assert len(dec.admit_log) == 5
    assert not dec._active  # everything released


def test_serve_eos_trims():
    dec = _FakeDecoder(slots=1)
    # slot 0 emits 0,1,2,...; eos=2 must trim to [0,1,2]
    outs = dec.serve([torch.zeros(4, dtype=torch.int64)], max_new_tokens=10,
                     eos_id=2, chunk=4)
    assert outs[0].tolist()[-1] == 2 and outs[0].numel() <= 4


def test_serve_seq_limit_finishes_slot():
    dec = _FakeDecoder(slots=1, max_seq=20)
    # prompt 15 long -> pos hits max_seq before max_new tokens
    outs = dec.serve([torch.zeros(15, dtype=torch.int64)],
                     max_new_tokens=50, chunk=2)
    assert 0 < outs[0].numel() < 50


def test_serve_ring_cap_guard():
    dec = _FakeDecoder(slots=1, ring_cap=8)
    try:
        dec.serve([torch.zeros(3, dtype=torch.int64)], max_new_tokens=16)
        raise AssertionError("expected ring_cap assertion")
    except AssertionError as e:
        assert "ring_cap" in str(e)
