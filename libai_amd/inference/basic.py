"""Inference pipelines (reference: libai/inference/basic.py:32-208 +
text_generation.py / text_classification.py / image_classification.py).

BasePipeline: config load + dist setup + model build + (optional) sharded
weight load; preprocess/forward/postprocess protocol with rank-0 postprocess.
"""

import logging

import torch

from ..config import LazyConfig, instantiate, try_get_key
from ..utils import distributed as du
from .generator import Generator

__all__ = ["BasePipeline", "TextGenerationPipeline", "TextClassificationPipeline",
           "ImageClassificationPipeline"]

logger = logging.getLogger(__name__)


class BasePipeline:
    def __init__(self, config_file=None, cfg=None, model=None, tokenizer=None,
                 data_parallel=1, tensor_parallel=1, pipeline_parallel=1,
                 model_path=None, dtype=torch.float32, **kwargs):
        assert cfg is not None or config_file is not None or model is not None
        if cfg is None and config_file is not None:
            cfg = LazyConfig.load(config_file)
        self.cfg = cfg
        if cfg is not None:
            dist_cfg = try_get_key(cfg, "train.dist", default={}) or {}
            dist_cfg["tensor_parallel_size"] = tensor_parallel
            dist_cfg["pipeline_parallel_size"] = pipeline_parallel
            du.setup_dist_util(dist_cfg)
        else:
            du.setup_dist_util({"tensor_parallel_size": tensor_parallel,
                                "pipeline_parallel_size": pipeline_parallel})

        self.device = du.get_device()
        self.dtype = dtype
        self.model = model if model is not None else self.load_pretrain_weight(
            cfg, model_path
        )
        self.model = self.model.to(dtype).to(self.device).eval()
        self.tokenizer = tokenizer if tokenizer is not None else self.build_tokenizer(
            cfg
        )

    def load_pretrain_weight(self, cfg, model_path):
        model = instantiate(cfg.model)
        if model_path:
            from ..models.utils.model_loader import ModelLoaderLiBai

            loader = ModelLoaderLiBai(model, cfg, model_path)
            model = loader.load()
        return model

    def build_tokenizer(self, cfg):
        if cfg is not None and try_get_key(cfg, "tokenization", default=None):
            from ..tokenizer import build_tokenizer

            return build_tokenizer(cfg)
        return None

    # -- protocol -----------------------------------------------------------

    def preprocess(self, inputs, **kwargs):
        raise NotImplementedError

    def forward(self, model_inputs, **kwargs):
        raise NotImplementedError

    def postprocess(self, model_outputs, **kwargs):
        raise NotImplementedError

    def __call__(self, inputs, **kwargs):
        model_inputs = self.preprocess(inputs, **kwargs)
        model_outputs = self.forward(model_inputs, **kwargs)
        return self.postprocess(model_outputs, **kwargs)


class TextGenerationPipeline(BasePipeline):
    """Autoregressive text generation (reference: text_generation.py)."""

    def preprocess(self, inputs, **kwargs):
        if isinstance(inputs, str):
            inputs = [inputs]
        ids = [self.tokenizer.encode(t) for t in inputs]
        maxlen = max(len(i) for i in ids)
        pad = self.tokenizer.convert_tokens_to_ids(self.tokenizer.pad_token) \
            if self.tokenizer.pad_token else 0
        batch = torch.full((len(ids), maxlen), pad, dtype=torch.long)
        for r, seq in enumerate(ids):
            batch[r, maxlen - len(seq):] = torch.tensor(seq)
        return {"input_ids": batch.to(self.device)}

    def forward(self, model_inputs, max_length=64, do_sample=False, top_k=0,
                top_p=1.0, temperature=1.0, num_beams=1, captured=False,
                **kwargs):
        if captured:
            # hipGraph-captured serving loop (GPU, TP=1, greedy or top-k
            # sampling; 2x the eager step — profiles/decode_captured.md)
            from .captured_decode import (
                CapturedGPTDecoder,
                CapturedGPTSampler,
                CapturedLlamaDecoder,
                CapturedLlamaSampler,
            )

            ids = model_inputs["input_ids"]
            is_gpt = hasattr(self.model, "GPT_model")
            if do_sample:
                cls = CapturedGPTSampler if is_gpt else CapturedLlamaSampler
                dec = cls(self.model, max_batch=ids.shape[0],
                          max_seq_len=max_length,
                          temperature=temperature, top_k=top_k)
            else:
                cls = CapturedGPTDecoder if is_gpt else CapturedLlamaDecoder
                dec = cls(self.model, max_batch=ids.shape[0],
                          max_seq_len=max_length)
            new_toks = dec.generate(ids, max_length - ids.shape[1])
            return {"sequences": torch.cat([ids, new_toks], dim=1)}
        gen = Generator(self.model)
        eos = (
            self.tokenizer.convert_tokens_to_ids(self.tokenizer.eos_token)
            if self.tokenizer and self.tokenizer.eos_token
            else None
        )
        out = gen.generate(
            model_inputs["input_ids"], max_length=max_length, do_sample=do_sample,
            top_k=top_k, top_p=top_p, temperature=temperature, num_beams=num_beams,
            eos_token_id=eos,
        )
        return {"sequences": out}

    def postprocess(self, model_outputs, **kwargs):
        seqs = model_outputs["sequences"]
        return [
            {"generated_text": self.tokenizer.decode(s.tolist(),
                                                     skip_special_tokens=True)}
            for s in seqs
        ]


class TextClassificationPipeline(BasePipeline):
    def preprocess(self, inputs, **kwargs):
        if isinstance(inputs, str):
            inputs = [inputs]
        encoded = [self.tokenizer.encode(t, add_special_tokens=True) for t in inputs]
        maxlen = max(len(e) for e in encoded)
        pad = self.tokenizer.convert_tokens_to_ids(self.tokenizer.pad_token) or 0
        ids = torch.full((len(encoded), maxlen), pad, dtype=torch.long)
        mask = torch.zeros(len(encoded), maxlen, dtype=torch.uint8)
        for r, seq in enumerate(encoded):
            ids[r, : len(seq)] = torch.tensor(seq)
            mask[r, : len(seq)] = 1
        return {
            "input_ids": ids.to(self.device),
            "attention_mask": mask.to(self.device),
        }

    def forward(self, model_inputs, **kwargs):
        with torch.no_grad():
            return self.model(**model_inputs)

    def postprocess(self, model_outputs, **kwargs):
        logits = model_outputs.get(
            "seq_relationship_scores", model_outputs.get("prediction_scores")
        )
        probs = torch.softmax(logits.float(), dim=-1)
        labels = probs.argmax(-1)
        return [
            {"label": int(l), "score": float(p[l])} for l, p in zip(labels, probs)
        ]


class ImageClassificationPipeline(BasePipeline):
    """Top-k image classification; pass ``class_names`` (list indexed by
    class id, e.g. the ImageNet-1k label table that the reference ships in
    inference/utils/imagenet_class.py) to get readable labels."""

    def __init__(self, *args, class_names=None, **kwargs):
        super().__init__(*args, **kwargs)
        self.class_names = class_names

    def preprocess(self, inputs, **kwargs):
        if torch.is_tensor(inputs):
            images = inputs if inputs.dim() == 4 else inputs.unsqueeze(0)
        else:
            images = torch.stack(list(inputs))
        return {"images": images.to(self.dtype).to(self.device)}

    def forward(self, model_inputs, **kwargs):
        with torch.no_grad():
            return self.model(**model_inputs)

    def postprocess(self, model_outputs, topk=5, **kwargs):
        logits = model_outputs["prediction_scores"].float()
        probs = torch.softmax(logits, dim=-1)
        scores, idx = probs.topk(min(topk, probs.shape[-1]), dim=-1)
        results = []
        for i, s in zip(idx, scores):
            r = {"classes": i.tolist(), "scores": s.tolist()}
            if self.class_names:
                r["labels"] = [self.class_names[c] for c in r["classes"]]
            results.append(r)
        return results
