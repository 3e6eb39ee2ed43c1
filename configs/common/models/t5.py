"""T5 model config (reference: configs/common/models/t5.py)."""

from libai_amd.config import ConfigDict, LazyCall
from libai_amd.models import T5ForPreTraining, T5Model

cfg = ConfigDict(
    vocab_size=32128,
    hidden_size=768,
    hidden_layers=12,
    num_attention_heads=12,
    intermediate_size=3072,
    max_position_embeddings=512,
    embedding_dropout_prob=0.1,
    hidden_dropout_prob=0.1,
    attention_dropout_prob=0.1,
    initializer_range=0.02,
)

t5_model = LazyCall(T5Model)(cfg=cfg)
pretrain_model = LazyCall(T5ForPreTraining)(cfg=cfg)
