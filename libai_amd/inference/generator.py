"""Autoregressive generation: greedy / sampling / beam search with logits
processors and stopping criteria.

Reference behavior: libai/inference/generator/generation_utils.py:451-800,
logits_process.py, beam_search.py, stopping_criteria.py.  Works with any
model whose forward supports (input_ids, past_key_values, use_cache) and
returns vocab-split logits (gathered over TP here).
"""

import torch
import torch.nn.functional as F

from ..parallel.comm import gather_from_tensor_parallel_region
from ..utils import distributed as du

__all__ = [
    "LogitsProcessorList",
    "TemperatureLogitsWarper",
    "TopKLogitsWarper",
    "TopPLogitsWarper",
    "RepetitionPenaltyLogitsProcessor",
    "NoRepeatNGramLogitsProcessor",
    "MinLengthLogitsProcessor",
    "StoppingCriteriaList",
    "MaxLengthCriteria",
    "EosTokenCriteria",
    "Generator",
]


# ---------------------------------------------------------------------------
# logits processors (reference: generator/logits_process.py)
# ---------------------------------------------------------------------------
class LogitsProcessorList(list):
    def __call__(self, input_ids, scores):
        for proc in self:
            scores = proc(input_ids, scores)
        return scores


class TemperatureLogitsWarper:
    def __init__(self, temperature):
        assert temperature > 0
        self.temperature = temperature

    def __call__(self, input_ids, scores):
        return scores / self.temperature


class TopKLogitsWarper:
    def __init__(self, top_k, filter_value=-float("inf")):
        self.top_k = top_k
        self.filter_value = filter_value

    def __call__(self, input_ids, scores):
        k = min(self.top_k, scores.size(-1))
        if k <= 0:
            return scores
        kth = torch.topk(scores, k)[0][..., -1, None]
        return scores.masked_fill(scores < kth, self.filter_value)


class TopPLogitsWarper:
    def __init__(self, top_p, filter_value=-float("inf"), min_tokens_to_keep=1):
        self.top_p = top_p
        self.filter_value = filter_value
        self.min_keep = min_tokens_to_keep

    def __call__(self, input_ids, scores):
        sorted_logits, sorted_idx = torch.sort(scores, descending=False)
        cum = sorted_logits.softmax(dim=-1).cumsum(dim=-1)
        remove = cum <= (1 - self.top_p)
        remove[..., -self.min_keep :] = False
        mask = remove.scatter(-1, sorted_idx, remove)
        return scores.masked_fill(mask, self.filter_value)


class RepetitionPenaltyLogitsProcessor:
    def __init__(self, penalty):
        self.penalty = penalty

    def __call__(self, input_ids, scores):
        prev = torch.gather(scores, 1, input_ids)
        prev = torch.where(prev < 0, prev * self.penalty, prev / self.penalty)
        return scores.scatter(1, input_ids, prev)


class NoRepeatNGramLogitsProcessor:
    def __init__(self, ngram_size):
        self.n = ngram_size

    def __call__(self, input_ids, scores):
        if self.n <= 0 or input_ids.shape[1] + 1 < self.n:
            return scores
        for b in range(input_ids.shape[0]):
            seq = input_ids[b].tolist()
            prefix = tuple(seq[-(self.n - 1) :]) if self.n > 1 else ()
            banned = set()
            for i in range(len(seq) - self.n + 1):
                if tuple(seq[i : i + self.n - 1]) == prefix:
                    banned.add(seq[i + self.n - 1])
            for tok in banned:
                scores[b, tok] = -float("inf")
        return scores


class MinLengthLogitsProcessor:
    def __init__(self, min_length, eos_token_id):
        self.min_length = min_length
        self.eos = eos_token_id

    def __call__(self, input_ids, scores):
        if input_ids.shape[-1] < self.min_length:
            scores[:, self.eos] = -float("inf")
        return scores


# ---------------------------------------------------------------------------
# stopping criteria (reference: generator/stopping_criteria.py)
# ---------------------------------------------------------------------------
class StoppingCriteriaList(list):
    def __call__(self, input_ids, scores=None):
        return any(c(input_ids, scores) for c in self)


class MaxLengthCriteria:
    def __init__(self, max_length):
        self.max_length = max_length

    def __call__(self, input_ids, scores=None):
        return input_ids.shape[-1] >= self.max_length


class EosTokenCriteria:
    def __init__(self, eos_token_id):
        self.eos = eos_token_id

    def __call__(self, input_ids, scores=None):
        return bool((input_ids[:, -1] == self.eos).all())


# ---------------------------------------------------------------------------
# generation mixin
# ---------------------------------------------------------------------------
class Generator:
    """Mixin-style driver; also usable standalone: Generator(model).generate(...)."""

    def __init__(self, model=None):
        self._gen_model = model

    @property
    def _model(self):
        return self._gen_model if self._gen_model is not None else self

    _enc_input_ids = None  # set by generate() for encoder-decoder models

    @torch.no_grad()
    def _step_logits(self, input_ids, past):
        model = self._model
        if self._enc_input_ids is not None:
            # encoder-decoder (T5-style): recompute the decoder over the
            # generated prefix each step (reference generation_utils.py's
            # enc-dec branch; cache-free v1 at pipeline scale)
            out = model(encoder_input_ids=self._enc_input_ids,
                        decoder_input_ids=input_ids)
            logits = out["prediction_scores"] if isinstance(out, dict) else out
            logits = logits[:, -1, :].float()
            if du.get_dist_util().tensor_parallel_size > 1:
                logits = gather_from_tensor_parallel_region(logits)
            return logits, None
        feed = input_ids if past is None else input_ids[:, -1:]
        out = model(input_ids=feed, past_key_values=past, use_cache=True)
        if isinstance(out, dict):
            logits, past = out["prediction_scores"], out.get("past_key_values")
        else:
            logits, past = out
        logits = logits[:, -1, :].float()
        if du.get_dist_util().tensor_parallel_size > 1:
            logits = gather_from_tensor_parallel_region(logits)
        return logits, past

    @torch.no_grad()
    def greedy_search(self, input_ids, logits_processor=None, stopping_criteria=None,
                      eos_token_id=None, pad_token_id=None):
        logits_processor = logits_processor or LogitsProcessorList()
        stopping_criteria = stopping_criteria or StoppingCriteriaList()
        past = None
        unfinished = torch.ones(input_ids.shape[0], dtype=torch.long,
                                device=input_ids.device)
        while True:
            logits, past = self._step_logits(input_ids, past)
            scores = logits_processor(input_ids, logits)
            next_tokens = scores.argmax(dim=-1)
            if eos_token_id is not None and pad_token_id is not None:
                next_tokens = next_tokens * unfinished + pad_token_id * (1 - unfinished)
            input_ids = torch.cat([input_ids, next_tokens[:, None]], dim=-1)
            if eos_token_id is not None:
                unfinished = unfinished * (next_tokens != eos_token_id).long()
            if stopping_criteria(input_ids) or (
                eos_token_id is not None and unfinished.max() == 0
            ):
                break
        return input_ids

    @torch.no_grad()
    def multinomial_sample(self, input_ids, logits_processor=None,
                           logits_warper=None, stopping_criteria=None,
                           eos_token_id=None, pad_token_id=None):
        logits_processor = logits_processor or LogitsProcessorList()
        logits_warper = logits_warper or LogitsProcessorList()
        stopping_criteria = stopping_criteria or StoppingCriteriaList()
        past = None
        unfinished = torch.ones(input_ids.shape[0], dtype=torch.long,
                                device=input_ids.device)
        while True:
            logits, past = self._step_logits(input_ids, past)
            scores = logits_warper(input_ids, logits_processor(input_ids, logits))
            probs = F.softmax(scores, dim=-1)
            next_tokens = torch.multinomial(probs, num_samples=1).squeeze(1)
            if eos_token_id is not None and pad_token_id is not None:
                next_tokens = next_tokens * unfinished + pad_token_id * (1 - unfinished)
            input_ids = torch.cat([input_ids, next_tokens[:, None]], dim=-1)
            if eos_token_id is not None:
                unfinished = unfinished * (next_tokens != eos_token_id).long()
            if stopping_criteria(input_ids) or (
                eos_token_id is not None and unfinished.max() == 0
            ):
                break
        return input_ids

    @torch.no_grad()
    def beam_search(self, input_ids, num_beams=4, max_length=64,
                    length_penalty=1.0, eos_token_id=None, pad_token_id=0,
                    logits_processor=None):
        """Batched beam search WITHOUT kv-cache re-ordering complexity:
        recomputes the full prefix each step (inference-pipeline scale)."""
        logits_processor = logits_processor or LogitsProcessorList()
        model = self._model
        batch, cur_len = input_ids.shape
        device = input_ids.device
        # expand to beams
        input_ids = input_ids.repeat_interleave(num_beams, dim=0)
        beam_scores = torch.full((batch, num_beams), -1e9, device=device)
        beam_scores[:, 0] = 0.0
        beam_scores = beam_scores.view(-1)
        done = [False] * batch
        finished = [[] for _ in range(batch)]  # (score, seq)

        while input_ids.shape[-1] < max_length and not all(done):
            out = model(input_ids=input_ids)
            logits = out["prediction_scores"] if isinstance(out, dict) else out
            logits = logits[:, -1, :].float()
            if du.get_dist_util().tensor_parallel_size > 1:
                logits = gather_from_tensor_parallel_region(logits)
            logits = logits_processor(input_ids, logits)
            log_probs = F.log_softmax(logits, dim=-1)
            vocab = log_probs.shape[-1]
            next_scores = beam_scores[:, None] + log_probs  # [batch*beams, vocab]
            next_scores = next_scores.view(batch, num_beams * vocab)
            top_scores, top_idx = next_scores.topk(2 * num_beams, dim=-1)

            new_ids, new_scores = [], []
            for b in range(batch):
                if done[b]:
                    new_ids.extend(
                        [input_ids[b * num_beams + i] for i in range(num_beams)]
                    )
                    new_scores.extend([beam_scores[b * num_beams + i]
                                       for i in range(num_beams)])
                    continue
                kept = 0
                for score, idx in zip(top_scores[b], top_idx[b]):
                    beam, tok = int(idx) // vocab, int(idx) % vocab
                    seq = torch.cat(
                        [input_ids[b * num_beams + beam],
                         torch.tensor([tok], device=device)]
                    )
                    if eos_token_id is not None and tok == eos_token_id:
                        finished[b].append(
                            (float(score) / (seq.shape[-1] ** length_penalty), seq)
                        )
                        continue
                    new_ids.append(seq)
                    new_scores.append(score)
                    kept += 1
                    if kept == num_beams:
                        break
                while kept < num_beams:  # degenerate fill
                    new_ids.append(new_ids[-1])
                    new_scores.append(torch.tensor(-1e9, device=device))
                    kept += 1
                if len(finished[b]) >= num_beams:
                    done[b] = True
            input_ids = torch.stack(
                [F.pad(s, (0, 0), value=pad_token_id) for s in new_ids]
            )
            beam_scores = torch.stack(
                [s if torch.is_tensor(s) else torch.tensor(s, device=device)
                 for s in new_scores]
            )

        outs = []
        max_len = 0
        for b in range(batch):
            if finished[b]:
                best = max(finished[b], key=lambda x: x[0])[1]
            else:
                best = input_ids[b * num_beams]
            outs.append(best)
            max_len = max(max_len, best.shape[-1])
        return torch.stack(
            [F.pad(o, (0, max_len - o.shape[-1]), value=pad_token_id) for o in outs]
        )

    @torch.no_grad()
    def generate(self, input_ids, max_length=64, min_length=0, do_sample=False,
                 num_beams=1, temperature=1.0, top_k=0, top_p=1.0,
                 repetition_penalty=1.0, no_repeat_ngram_size=0, eos_token_id=None,
                 pad_token_id=0, length_penalty=1.0, encoder_input_ids=None,
                 decoder_start_token_id=None, **kwargs):
        """Dispatcher (reference: generation_utils.py:787+).

        For encoder-decoder models pass ``encoder_input_ids``; ``input_ids``
        then seeds the decoder (defaults to a [batch, 1] tensor of
        ``decoder_start_token_id``).
        """
        self._enc_input_ids = encoder_input_ids
        if encoder_input_ids is not None and input_ids is None:
            start = decoder_start_token_id if decoder_start_token_id is not None \
                else pad_token_id
            input_ids = torch.full((encoder_input_ids.shape[0], 1), start,
                                   dtype=torch.long,
                                   device=encoder_input_ids.device)
        processors = LogitsProcessorList()
        if repetition_penalty != 1.0:
            processors.append(RepetitionPenaltyLogitsProcessor(repetition_penalty))
        if no_repeat_ngram_size > 0:
            processors.append(NoRepeatNGramLogitsProcessor(no_repeat_ngram_size))
        if min_length > 0 and eos_token_id is not None:
            processors.append(MinLengthLogitsProcessor(min_length, eos_token_id))
        stopping = StoppingCriteriaList([MaxLengthCriteria(max_length)])

        if num_beams > 1:
            return self.beam_search(input_ids, num_beams=num_beams,
                                    max_length=max_length,
                                    length_penalty=length_penalty,
                                    eos_token_id=eos_token_id,
                                    pad_token_id=pad_token_id,
                                    logits_processor=processors)
        if do_sample:
            warpers = LogitsProcessorList()
            if temperature != 1.0:
                warpers.append(TemperatureLogitsWarper(temperature))
            if top_k > 0:
                warpers.append(TopKLogitsWarper(top_k))
            if top_p < 1.0:
                warpers.append(TopPLogitsWarper(top_p))
            return self.multinomial_sample(input_ids, processors, warpers, stopping,
                                           eos_token_id, pad_token_id)
        return self.greedy_search(input_ids, processors, stopping, eos_token_id,
                                  pad_token_id)
