"""End-to-end generation on GPU (KV-cache decode crosses the odd-SK path)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpt_generate_greedy_gpu():
    from libai_amd.inference.generator import Generator
    from libai_amd.models import GPTForPreTraining
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    gpt = GPTForPreTraining(hidden_layers=2, vocab_size=512, hidden_size=256,
                            ffn_hidden_size=1024, num_attention_heads=4,
                            max_seq_length=64).to(torch.bfloat16).cuda().eval()

    class Wrapper(torch.nn.Module):
        def __init__(self, m):
            super().__init__()
            self.m = m

        def forward(self, input_ids, past_key_values=None, use_cache=False):
            out = self.m.GPT_model(input_ids, past_key_values=past_key_values,
                                   use_cache=use_cache)
            if use_cache:
                return {"prediction_scores": out[0], "past_key_values": out[1]}
            return {"prediction_scores": out}

    gen = Generator(Wrapper(gpt))
    ids = torch.randint(0, 512, (2, 5), device="cuda")
    out = gen.generate(ids, max_length=21)  # crosses odd SK decode widths
    assert out.shape == (2, 21)
    out2 = gen.generate(ids, max_length=21)
    assert torch.equal(out, out2)

    # sampled + beam paths run
    _ = gen.generate(ids, max_length=15, do_sample=True, top_k=5, temperature=0.8)
    _ = gen.generate(ids, max_length=12, num_beams=2)


def test_llama_generate_gpu():
    from libai_amd.inference.generator import Generator
    from libai_amd.models import LlamaForCausalLM
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    m = LlamaForCausalLM(hidden_layers=2, vocab_size=512, hidden_size=256,
                         intermediate_size=512, num_attention_heads=4,
                         max_position_embeddings=64).to(torch.bfloat16).cuda().eval()
    gen = Generator(m)
    ids = torch.randint(0, 512, (1, 4), device="cuda")
    out = gen.generate(ids, max_length=19)
    assert out.shape == (1, 19)
