"""Llama model config (reference: projects/Llama/configs/llama_config.py)."""

from libai_amd.config import ConfigDict, LazyCall
from libai_amd.models import LlamaForCausalLM

cfg = ConfigDict(
    hidden_size=4096,
    intermediate_size=11008,
    hidden_layers=32,
    num_attention_heads=32,
    max_position_embeddings=2048,
    rms_norm_eps=1e-5,
    vocab_size=32000,
    initializer_range=0.02,
    use_scaled_init_for_output_weights=False,
    tie_word_embeddings=False,
    rope_theta=10000.0,
    amp_enabled=True,
)

model = LazyCall(LlamaForCausalLM)(cfg=cfg)
