"""lm-evaluation-harness bridge (reference capability:
projects/Eval_LLM/eval_harness.py — adapts the framework's causal LMs to
EleutherAI lm-eval's LM interface).

``LibaiEvalHarnessLM`` implements the three request kinds the harness
issues — loglikelihood, loglikelihood_rolling, generate_until — over any
model emitting ``prediction_scores`` logits (GPT/Llama/BLOOM/...).  When
the ``lm_eval`` package is installed, ``as_lm_eval_model()`` returns a
subclass of its ``LM`` base so it can be passed straight to
``lm_eval.simple_evaluate``; without it, the class still works standalone
(requests as (context, continuation) / (context, gen_kwargs) tuples),
which is how the tests exercise it in this offline image.
"""

import torch
import torch.nn.functional as F

from ..utils import distributed as du

__all__ = ["LibaiEvalHarnessLM", "as_lm_eval_model"]


def _req_args(req):
    """lm-eval>=0.4 passes Instance objects with .args; tuples work too."""
    return req.args if hasattr(req, "args") else req


class LibaiEvalHarnessLM:
    def __init__(self, model, tokenizer, batch_size=8, max_length=2048,
                 device=None):
        self.model = model.eval()
        self.tokenizer = tokenizer
        self.batch_size = batch_size
        self.max_length = max_length
        self.device = device if device is not None else du.get_device()

    # -- tokenization helpers ----------------------------------------------

    def _encode(self, text):
        ids = self.tokenizer.encode(text)
        if hasattr(ids, "ids"):  # tokenizers.Encoding
            ids = ids.ids
        return list(ids)

    def _eot(self):
        for attr in ("eod", "eos_token_id", "eod_id"):
            v = getattr(self.tokenizer, attr, None)
            if v is not None:
                return int(v)
        return 0

    # -- scoring ------------------------------------------------------------

    @torch.no_grad()
    def _score(self, ctx_ids, cont_ids):
        """(sum logprob of continuation, continuation is the greedy argmax)."""
        ids = (ctx_ids + cont_ids)[-(self.max_length + 1):]
        inp = torch.tensor([ids[:-1]], dtype=torch.long, device=self.device)
        out = self.model(input_ids=inp)
        logits = out["prediction_scores"] if isinstance(out, dict) else out
        logits = logits.float()
        if du.get_dist_util().tensor_parallel_size > 1:
            from ..parallel.comm import gather_from_tensor_parallel_region

            logits = gather_from_tensor_parallel_region(logits)
        n = len(cont_ids)
        cont_logits = logits[0, -n:]                     # predicts ids[-n:]
        targets = torch.tensor(ids[-n:], device=self.device)
        lp = F.log_softmax(cont_logits, dim=-1)
        ll = lp[torch.arange(n, device=self.device), targets].sum()
        greedy = bool((cont_logits.argmax(dim=-1) == targets).all())
        return float(ll), greedy

    def loglikelihood(self, requests):
        res = []
        for req in requests:
            context, continuation = _req_args(req)
            ctx_ids = self._encode(context) if context else [self._eot()]
            cont_ids = self._encode(continuation)
            res.append(self._score(ctx_ids, cont_ids))
        return res

    def loglikelihood_rolling(self, requests):
        res = []
        for req in requests:
            (text,) = _req_args(req)
            ids = self._encode(text)
            # windowed full-text loglikelihood, eot as the first context
            total = 0.0
            for start in range(0, len(ids), self.max_length):
                chunk = ids[start : start + self.max_length]
                ctx = [self._eot()] if start == 0 else ids[start - 1 : start]
                ll, _ = self._score(ctx, chunk)
                total += ll
            res.append(total)
        return res

    @torch.no_grad()
    def generate_until(self, requests):
        from .generator import Generator

        gen = Generator(self.model)
        res = []
        for req in requests:
            context, gen_kwargs = _req_args(req)
            gen_kwargs = dict(gen_kwargs or {})
            until = gen_kwargs.pop("until", [])
            if isinstance(until, str):
                until = [until]
            max_new = int(gen_kwargs.pop("max_gen_toks", 64))
            ctx_ids = self._encode(context)[-self.max_length:]
            inp = torch.tensor([ctx_ids], dtype=torch.long, device=self.device)
            out_ids = gen.generate(
                inp, max_length=inp.shape[1] + max_new,
                do_sample=bool(gen_kwargs.pop("do_sample", False)),
                eos_token_id=self._eot(), pad_token_id=self._eot(),
            )
            text = self.tokenizer.decode(out_ids[0][inp.shape[1]:].tolist())
            for stop in until:  # harness contract: cut at the first stop seq
                idx = text.find(stop)
                if idx >= 0:
                    text = text[:idx]
            res.append(text)
        return res


def as_lm_eval_model(model, tokenizer, **kwargs):
    """An ``lm_eval.api.model.LM`` subclass instance wrapping the bridge
    (requires the lm_eval package)."""
    from lm_eval.api.model import LM

    bridge = LibaiEvalHarnessLM(model, tokenizer, **kwargs)

    class _LM(LM):
        def loglikelihood(self, requests):
            return bridge.loglikelihood(requests)

        def loglikelihood_rolling(self, requests):
            return bridge.loglikelihood_rolling(requests)

        def generate_until(self, requests):
            return bridge.generate_until(requests)

    return _LM()
