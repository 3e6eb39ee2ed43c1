"""Contrastive self-supervised models: SimCSE and MoCo v3.

Reference capabilities: projects/SimCSE (unsupervised sentence embeddings —
encode twice under independent dropout, in-batch InfoNCE) and
projects/MOCOV3 (ViT backbone + projection/prediction heads + momentum
encoder, symmetric InfoNCE).
"""

import copy

import torch
import torch.nn.functional as F
from torch import nn

from ..config import configurable
from .bert_model import BertModel
from .vision_transformer import VisionTransformer

__all__ = ["SimCSEModel", "MoCoV3"]


class SimCSEModel(nn.Module):
    """Unsupervised SimCSE: the same batch is encoded twice; the two
    dropout-noised views of a sentence are positives, everything else
    in-batch is a negative (InfoNCE at temperature tau)."""

    @configurable
    def __init__(self, temperature=0.05, pooler="cls", **bert_kwargs):
        super().__init__()
        bert_kwargs.setdefault("add_pooling_layer", False)
        self.encoder = BertModel(**bert_kwargs)
        self.temperature = temperature
        self.pooler = pooler

    @classmethod
    def from_config(cls, cfg):
        kw = dict(BertModel.from_config(cfg))
        kw["temperature"] = cfg.get("temperature", 0.05)
        kw["pooler"] = cfg.get("pooler", "cls")
        return kw

    def embed(self, input_ids, attention_mask=None):
        seq_out, _ = self.encoder(input_ids, attention_mask)
        if self.pooler == "cls":
            return seq_out[:, 0]
        mask = (attention_mask if attention_mask is not None
                else torch.ones_like(input_ids)).unsqueeze(-1).to(seq_out.dtype)
        return (seq_out * mask).sum(1) / mask.sum(1).clamp(min=1.0)

    def forward(self, input_ids, attention_mask=None, labels=None):
        if not self.training:
            return {"embeddings": F.normalize(
                self.embed(input_ids, attention_mask), dim=-1)}
        # two forwards = two independent dropout masks (the SimCSE trick)
        z1 = F.normalize(self.embed(input_ids, attention_mask), dim=-1)
        z2 = F.normalize(self.embed(input_ids, attention_mask), dim=-1)
        logits = z1 @ z2.t() / self.temperature  # [b, b]
        target = torch.arange(z1.shape[0], device=z1.device)
        loss = F.cross_entropy(logits.float(), target)
        return {"contrastive_loss": loss}


class _ProjectionMLP(nn.Module):
    def __init__(self, in_dim, hidden_dim, out_dim, num_layers=3):
        super().__init__()
        layers = []
        d = in_dim
        for i in range(num_layers - 1):
            layers += [nn.Linear(d, hidden_dim), nn.BatchNorm1d(hidden_dim),
                       nn.ReLU(inplace=True)]
            d = hidden_dim
        layers += [nn.Linear(d, out_dim)]
        self.net = nn.Sequential(*layers)

    def forward(self, x):
        return self.net(x)


class MoCoV3(nn.Module):
    """MoCo v3: query encoder (backbone + projector + predictor) vs a
    momentum key encoder (EMA copy, no grads); symmetric InfoNCE over
    in-batch keys."""

    @configurable
    def __init__(self, embed_dim=768, proj_dim=256, proj_hidden=4096,
                 momentum=0.99, temperature=0.2, **vit_kwargs):
        super().__init__()
        vit_kwargs.setdefault("embed_dim", embed_dim)
        vit_kwargs.setdefault("num_classes", 8)  # head unused; keep it tiny
        self.backbone = VisionTransformer(**vit_kwargs)
        self.projector = _ProjectionMLP(embed_dim, proj_hidden, proj_dim)
        self.predictor = _ProjectionMLP(proj_dim, proj_hidden, proj_dim,
                                        num_layers=2)
        self.momentum_backbone = copy.deepcopy(self.backbone)
        self.momentum_projector = copy.deepcopy(self.projector)
        for p in self.momentum_backbone.parameters():
            p.requires_grad_(False)
        for p in self.momentum_projector.parameters():
            p.requires_grad_(False)
        self.momentum = momentum
        self.temperature = temperature

    @classmethod
    def from_config(cls, cfg):
        kw = dict(VisionTransformer.from_config(cfg))
        kw.pop("num_classes", None)
        for k, dflt in (("proj_dim", 256), ("proj_hidden", 4096),
                        ("momentum", 0.99), ("temperature", 0.2)):
            kw[k] = cfg.get(k, dflt)
        return kw

    def _features(self, backbone, images):
        return backbone.forward_features(images)[:, 0]  # CLS token

    @torch.no_grad()
    def update_momentum_encoder(self):
        """EMA step; call once per optimizer step (the trainer's hooks or
        the training script drive it)."""
        for q, k in zip(self.backbone.parameters(),
                        self.momentum_backbone.parameters()):
            k.mul_(self.momentum).add_(q.detach(), alpha=1 - self.momentum)
        for q, k in zip(self.projector.parameters(),
                        self.momentum_projector.parameters()):
            k.mul_(self.momentum).add_(q.detach(), alpha=1 - self.momentum)

    def _ctr(self, q, k):
        q = F.normalize(q, dim=-1)
        k = F.normalize(k, dim=-1)
        logits = q @ k.t() / self.temperature
        target = torch.arange(q.shape[0], device=q.device)
        return F.cross_entropy(logits.float(), target)

    def forward(self, images, images2=None, labels=None):
        """images/images2: the two augmented views ([b, 3, H, W] each); with
        a single view provided the second defaults to it (synthetic data)."""
        v1, v2 = images, images2 if images2 is not None else images
        q1 = self.predictor(self.projector(self._features(self.backbone, v1)))
        q2 = self.predictor(self.projector(self._features(self.backbone, v2)))
        with torch.no_grad():
            k1 = self.momentum_projector(
                self._features(self.momentum_backbone, v1))
            k2 = self.momentum_projector(
                self._features(self.momentum_backbone, v2))
        loss = self._ctr(q1, k2) + self._ctr(q2, k1)
        return {"moco_loss": loss}
