"""bf16 GPU fwd+bwd smoke across the model zoo + LoRA step."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _setup():
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    yield


def _step(model, batch):
    out = model(**batch)
    loss = sum(v for v in out.values() if torch.is_tensor(v) and v.ndim == 0)
    loss.backward()
    assert torch.isfinite(loss), float(loss)
    return float(loss)


def test_bert_gpu_bf16():
    from libai_amd.models import BertForPreTraining

    torch.manual_seed(0)
    m = BertForPreTraining(
        vocab_size=1024, hidden_size=256, hidden_layers=2, num_attention_heads=4,
        intermediate_size=512, max_position_embeddings=128,
    ).to(torch.bfloat16).cuda()
    b, s = 4, 128
    mask = torch.ones(b, s, dtype=torch.uint8, device="cuda")
    mask[:, -16:] = 0
    _step(m, dict(
        input_ids=torch.randint(0, 1024, (b, s), device="cuda"),
        attention_mask=mask,
        ns_labels=torch.randint(0, 2, (b,), device="cuda"),
        lm_labels=torch.randint(0, 1024, (b, s), device="cuda"),
        loss_mask=(torch.rand(b, s, device="cuda") < 0.15).long(),
    ))


def test_t5_gpu_bf16():
    from libai_amd.models import T5ForPreTraining

    torch.manual_seed(0)
    m = T5ForPreTraining(
        vocab_size=1024, hidden_size=256, hidden_layers=2, num_attention_heads=4,
        intermediate_size=512, max_position_embeddings=128,
    ).to(torch.bfloat16).cuda()
    b = 2
    _step(m, dict(
        encoder_input_ids=torch.randint(0, 1024, (b, 64), device="cuda"),
        decoder_input_ids=torch.randint(0, 1024, (b, 32), device="cuda"),
        encoder_attn_mask=torch.ones(b, 64, dtype=torch.uint8, device="cuda"),
        lm_labels=torch.randint(0, 1024, (b, 32), device="cuda"),
        loss_mask=torch.ones(b, 32, dtype=torch.long, device="cuda"),
    ))


def test_vit_gpu_bf16():
    from libai_amd.models import VisionTransformer

    torch.manual_seed(0)
    m = VisionTransformer(img_size=64, patch_size=8, embed_dim=256, depth=2,
                          num_heads=4, num_classes=100).to(torch.bfloat16).cuda()
    _step(m, dict(
        images=torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16),
        labels=torch.randint(0, 100, (4,), device="cuda"),
    ))


def test_swin_and_resmlp_gpu_bf16():
    from libai_amd.models import ResMLP, SwinTransformer

    torch.manual_seed(0)
    sw = SwinTransformer(img_size=64, patch_size=4, embed_dim=48, depths=(1, 1),
                         num_heads=(2, 4), window_size=4,
                         num_classes=10).to(torch.bfloat16).cuda()
    _step(sw, dict(images=torch.randn(2, 3, 64, 64, device="cuda",
                                      dtype=torch.bfloat16),
                   labels=torch.randint(0, 10, (2,), device="cuda")))
    rm = ResMLP(img_size=64, patch_size=8, embed_dim=64, depth=2,
                num_classes=10).to(torch.bfloat16).cuda()
    _step(rm, dict(images=torch.randn(2, 3, 64, 64, device="cuda",
                                      dtype=torch.bfloat16),
                   labels=torch.randint(0, 10, (2,), device="cuda")))


def test_lora_gpu_step():
    from libai_amd.lora import apply_lora
    from libai_amd.models import LlamaForCausalLM
    from libai_amd.optim import FusedAdamW

    torch.manual_seed(0)
    m = LlamaForCausalLM(hidden_layers=2, vocab_size=512, hidden_size=256,
                         intermediate_size=512, num_attention_heads=4,
                         max_position_embeddings=64).to(torch.bfloat16).cuda()
    apply_lora(m, r=8, alpha=16)
    opt = FusedAdamW([p for p in m.parameters() if p.requires_grad], lr=1e-3)
    ids = torch.randint(0, 512, (2, 33), device="cuda")
    for _ in range(3):
        opt.zero_grad()
        out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
        out["lm_loss"].backward()
        opt.step()
    assert torch.isfinite(out["lm_loss"])


def test_t5_gated_gpu_bf16():
    from libai_amd.models import T5ForPreTraining

    torch.manual_seed(0)
    m = T5ForPreTraining(
        vocab_size=1024, hidden_size=256, hidden_layers=2, num_attention_heads=4,
        intermediate_size=512, max_position_embeddings=128, mlp_type="gated",
        activation="silu",
    ).to(torch.bfloat16).cuda()
    b = 2
    _step(m, dict(
        encoder_input_ids=torch.randint(0, 1024, (b, 64), device="cuda"),
        decoder_input_ids=torch.randint(0, 1024, (b, 32), device="cuda"),
        encoder_attn_mask=torch.ones(b, 64, dtype=torch.uint8, device="cuda"),
        lm_labels=torch.randint(0, 1024, (b, 32), device="cuda"),
        loss_mask=torch.ones(b, 32, dtype=torch.long, device="cuda"),
    ))


def test_bloom_alibi_gpu_bf16():
    from libai_amd.models import BloomForCausalLM

    torch.manual_seed(0)
    m = BloomForCausalLM(vocab_size=1024, hidden_size=256, hidden_layers=2,
                         num_attention_heads=4).to(torch.bfloat16).cuda()
    ids = torch.randint(0, 1024, (2, 65), device="cuda")
    _step(m, dict(input_ids=ids[:, :-1], labels=ids[:, 1:]))


def test_mae_gpu_bf16():
    from libai_amd.models import MAEForPreTraining

    torch.manual_seed(0)
    m = MAEForPreTraining(img_size=64, patch_size=16, embed_dim=256, depth=2,
                          num_heads=4, decoder_embed_dim=128, decoder_depth=1,
                          decoder_num_heads=4).to(torch.bfloat16).cuda()
    _step(m, dict(images=torch.randn(2, 3, 64, 64, device="cuda",
                                     dtype=torch.bfloat16)))


def test_t5_relative_position_bias_gpu_bf16():
    from libai_amd.models import T5ForPreTraining

    torch.manual_seed(0)
    m = T5ForPreTraining(
        vocab_size=1024, hidden_size=256, hidden_layers=2, num_attention_heads=4,
        intermediate_size=512, max_position_embeddings=128,
        relative_attention=True,
    ).to(torch.bfloat16).cuda()
    b = 2
    _step(m, dict(
        encoder_input_ids=torch.randint(0, 1024, (b, 64), device="cuda"),
        decoder_input_ids=torch.randint(0, 1024, (b, 32), device="cuda"),
        encoder_attn_mask=torch.ones(b, 64, dtype=torch.uint8, device="cuda"),
        lm_labels=torch.randint(0, 1024, (b, 32), device="cuda"),
        loss_mask=torch.ones(b, 32, dtype=torch.long, device="cuda"),
    ))
    # relative attention: no absolute positions, bias params got grads
    assert m.t5_model.embedding.position_embeddings is None
    assert m.t5_model.enc_rel_bias.weight.grad is not None
    assert m.t5_model.dec_rel_bias.weight.grad is not None


def test_simcse_gpu_bf16():
    from libai_amd.models import SimCSEModel

    torch.manual_seed(0)
    m = SimCSEModel(vocab_size=1024, hidden_size=256, hidden_layers=2,
                    num_attention_heads=4, intermediate_size=512,
                    max_position_embeddings=128, hidden_dropout_prob=0.1,
                    attention_probs_dropout_prob=0.1).to(torch.bfloat16).cuda()
    ids = torch.randint(0, 1024, (8, 64), device="cuda")
    _step(m, dict(input_ids=ids))


def test_moco_gpu_bf16():
    from libai_amd.models import MoCoV3

    torch.manual_seed(0)
    m = MoCoV3(img_size=64, patch_size=16, embed_dim=256, depth=2, num_heads=4,
               proj_dim=64, proj_hidden=128).to(torch.bfloat16).cuda()
    imgs = torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    _step(m, dict(images=imgs, images2=imgs.flip(-1)))
    m.update_momentum_encoder()


def test_moe_layer_gpu_bf16():
    from libai_amd.layers import TransformerLayer

    torch.manual_seed(0)
    layer = TransformerLayer(256, 512, 4, mlp_type="moe", moe_num_experts=4,
                             moe_top_k=2).to(torch.bfloat16).cuda()
    x = torch.randn(2, 64, 256, device="cuda", dtype=torch.bfloat16)
    y = layer(x)
    (y.float().pow(2).mean() + layer.mlp.last_aux_loss).backward()
    assert torch.isfinite(y.float()).all()
    assert layer.mlp.w1.grad is not None


def test_clip_gpu_bf16():
    from libai_amd.models import CLIPModel

    torch.manual_seed(0)
    m = CLIPModel(embed_dim=64, img_size=64, patch_size=16, vision_width=256,
                  vision_layers=2, vision_heads=4, vocab_size=1024,
                  context_length=32, text_width=256, text_layers=2,
                  text_heads=4).to(torch.bfloat16).cuda()
    imgs = torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    txt = torch.randint(0, 1023, (4, 16), device="cuda")
    txt[:, -1] = 1023
    _step(m, dict(images=imgs, text_ids=txt))


def test_llama_gqa_gpu_bf16():
    from libai_amd.models import LlamaForCausalLM

    torch.manual_seed(0)
    m = LlamaForCausalLM(hidden_layers=2, vocab_size=1024, hidden_size=512,
                         intermediate_size=1024, num_attention_heads=8,
                         num_key_value_heads=2,
                         max_position_embeddings=128).to(torch.bfloat16).cuda()
    ids = torch.randint(0, 1024, (2, 65), device="cuda")
    _step(m, dict(input_ids=ids[:, :-1], labels=ids[:, 1:]))


def test_gpt_moe_gpu_bf16():
    from libai_amd.models import GPTForPreTraining

    torch.manual_seed(0)
    m = GPTForPreTraining(hidden_layers=2, vocab_size=1024, hidden_size=256,
                          ffn_hidden_size=512, num_attention_heads=4,
                          max_seq_length=64, moe_num_experts=4,
                          embedding_dropout_prob=0.0,
                          attention_dropout_prob=0.0,
                          output_dropout_prob=0.0).to(torch.bfloat16).cuda()
    ids = torch.randint(0, 1024, (2, 65), device="cuda")
    _step(m, dict(input_ids=ids[:, :-1], labels=ids[:, 1:]))


def test_palm_gpu_bf16():
    from libai_amd.models import PaLMForCausalLM

    torch.manual_seed(0)
    m = PaLMForCausalLM(hidden_layers=2, vocab_size=1024, hidden_size=512,
                        intermediate_size=1024, num_attention_heads=8,
                        max_position_embeddings=128).to(torch.bfloat16).cuda()
    ids = torch.randint(0, 1024, (2, 65), device="cuda")
    _step(m, dict(input_ids=ids[:, :-1], labels=ids[:, 1:]))


def test_convnext_gpu_bf16():
    from libai_amd.models import ConvNeXt

    torch.manual_seed(0)
    m = ConvNeXt(img_size=64, num_classes=16, depths=(1, 1, 2, 1),
                 dims=(32, 64, 128, 256),
                 drop_path_rate=0.1).to(torch.bfloat16).cuda()
    imgs = torch.randn(2, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    labels = torch.randint(0, 16, (2,), device="cuda")
    _step(m, dict(images=imgs, labels=labels))
