"""CPU smoke tests over the model zoo: fwd/bwd shapes + finite losses."""

import torch

from libai_amd.utils import distributed as du

du.setup_dist_util({})


def test_bert_pretraining_fwd_bwd():
    from libai_amd.models import BertForPreTraining

    torch.manual_seed(0)
    m = BertForPreTraining(
        vocab_size=128, hidden_size=32, hidden_layers=2, num_attention_heads=4,
        intermediate_size=64, max_position_embeddings=32, num_tokentypes=2,
        hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
    )
    b, s = 2, 16
    ids = torch.randint(0, 128, (b, s))
    mask = torch.ones(b, s, dtype=torch.uint8)
    mask[:, -3:] = 0  # padding
    lm_labels = torch.randint(0, 128, (b, s))
    loss_mask = torch.zeros(b, s, dtype=torch.long)
    loss_mask[:, :4] = 1
    out = m(input_ids=ids, attention_mask=mask, ns_labels=torch.randint(0, 2, (b,)),
            lm_labels=lm_labels, loss_mask=loss_mask)
    total = sum(out.values())
    assert torch.isfinite(total)
    total.backward()


def test_bert_padding_mask_blocks_attention():
    from libai_amd.models import BertModel

    torch.manual_seed(0)
    m = BertModel(vocab_size=64, hidden_size=32, hidden_layers=1,
                  num_attention_heads=4, intermediate_size=64,
                  max_position_embeddings=16, hidden_dropout_prob=0.0,
                  attention_probs_dropout_prob=0.0, add_pooling_layer=False)
    m.eval()
    ids = torch.randint(0, 64, (1, 8))
    mask = torch.ones(1, 8, dtype=torch.uint8)
    mask[0, -2:] = 0
    h1, _ = m(ids, attention_mask=mask)
    ids2 = ids.clone()
    ids2[0, -1] = (ids2[0, -1] + 1) % 64  # change a PADDING token
    h2, _ = m(ids2, attention_mask=mask)
    # visible positions must be unaffected by padding-token content
    assert torch.allclose(h1[0, :6], h2[0, :6], atol=1e-5)


def test_llama_fwd_bwd_and_swiglu_path():
    from libai_amd.models import LlamaForCausalLM

    torch.manual_seed(0)
    m = LlamaForCausalLM(hidden_layers=2, vocab_size=128, hidden_size=64,
                         intermediate_size=96, num_attention_heads=4,
                         max_position_embeddings=64)
    ids = torch.randint(0, 128, (2, 17))
    out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
    assert torch.isfinite(out["lm_loss"])
    out["lm_loss"].backward()


def test_vit_fwd_bwd():
    from libai_amd.models import VisionTransformer

    torch.manual_seed(0)
    m = VisionTransformer(img_size=32, patch_size=8, embed_dim=64, depth=2,
                          num_heads=4, num_classes=10)
    out = m(images=torch.randn(2, 3, 32, 32), labels=torch.randint(0, 10, (2,)))
    out["losses"].backward()


def test_rope_reference_properties():
    from libai_amd.ops.rope import apply_rotary_pos_emb

    torch.manual_seed(0)
    x = torch.randn(1, 8, 2, 16)
    y = apply_rotary_pos_emb(x, max_seq=32)
    # rotation preserves pairwise norms
    half = 8
    n_in = x[..., :half] ** 2 + x[..., half:] ** 2
    n_out = y[..., :half] ** 2 + y[..., half:] ** 2
    assert torch.allclose(n_in, n_out, atol=1e-5)
    # position 0 is identity
    assert torch.allclose(y[:, 0], x[:, 0], atol=1e-6)


def test_swiglu_reference():
    from libai_amd.ops.swiglu import swiglu

    x = torch.randn(4, 32, requires_grad=True)
    y = swiglu(x)
    g, u = x.detach().chunk(2, -1)
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(y, ref, atol=1e-6)
    y.sum().backward()
    assert torch.isfinite(x.grad).all()


def test_t5_fwd_bwd_and_masks():
    from libai_amd.models import T5ForPreTraining

    torch.manual_seed(0)
    m = T5ForPreTraining(
        vocab_size=128, hidden_size=32, hidden_layers=2, num_attention_heads=4,
        intermediate_size=64, max_position_embeddings=32,
        hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
        embedding_dropout_prob=0.0,
    )
    b, se, sd = 2, 12, 8
    enc = torch.randint(0, 128, (b, se))
    dec = torch.randint(0, 128, (b, sd))
    enc_mask = torch.ones(b, se, dtype=torch.uint8)
    enc_mask[:, -2:] = 0
    labels = torch.randint(0, 128, (b, sd))
    loss_mask = torch.ones(b, sd, dtype=torch.long)
    out = m(encoder_input_ids=enc, decoder_input_ids=dec, encoder_attn_mask=enc_mask,
            lm_labels=labels, loss_mask=loss_mask)
    assert torch.isfinite(out["masked_lm_loss"])
    out["masked_lm_loss"].backward()


def test_roberta_fwd_bwd():
    from libai_amd.models import RobertaForPreTraining

    torch.manual_seed(0)
    m = RobertaForPreTraining(
        vocab_size=128, hidden_size=32, hidden_layers=2, num_attention_heads=4,
        intermediate_size=64, max_position_embeddings=34, num_tokentypes=0,
        hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
        add_pooling_layer=False,
    )
    ids = torch.randint(2, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    loss_mask = torch.ones(2, 16, dtype=torch.long)
    out = m(input_ids=ids, attention_mask=torch.ones(2, 16, dtype=torch.uint8),
            lm_labels=labels, loss_mask=loss_mask)
    out["lm_loss"].backward()


def test_resmlp_fwd_bwd():
    from libai_amd.models import ResMLP

    m = ResMLP(img_size=32, patch_size=8, embed_dim=32, depth=2, num_classes=10)
    out = m(images=torch.randn(2, 3, 32, 32), labels=torch.randint(0, 10, (2,)))
    out["losses"].backward()


def test_swin_fwd_bwd():
    from libai_amd.models import SwinTransformer

    m = SwinTransformer(img_size=32, patch_size=4, embed_dim=24, depths=(1, 1),
                        num_heads=(2, 4), window_size=4, num_classes=10)
    out = m(images=torch.randn(2, 3, 32, 32), labels=torch.randint(0, 10, (2,)))
    out["losses"].backward()
    assert torch.isfinite(out["losses"])


def test_swin_v2_fwd_bwd():
    from libai_amd.models import SwinTransformerV2

    m = SwinTransformerV2(img_size=32, patch_size=4, embed_dim=24, depths=(1, 1),
                          num_heads=(2, 4), window_size=4, num_classes=10)
    out = m(images=torch.randn(2, 3, 32, 32), labels=torch.randint(0, 10, (2,)))
    out["losses"].backward()
    assert torch.isfinite(out["losses"])


def test_roberta_causal_lm():
    """True CLM head: causal masking verified (future edit leaves prefix logits)."""
    import torch

    from libai_amd.models import RobertaForCausalLM

    torch.manual_seed(0)
    m = RobertaForCausalLM(vocab_size=128, hidden_size=64, hidden_layers=2,
                           num_attention_heads=4, intermediate_size=128,
                           max_position_embeddings=66)
    ids = torch.randint(2, 128, (2, 16))
    out = m(input_ids=ids, labels=ids)
    out["lm_loss"].backward()
    assert torch.isfinite(out["lm_loss"])
    m.eval()
    with torch.no_grad():
        a = m(input_ids=ids)["prediction_scores"]
        ids2 = ids.clone()
        ids2[:, -1] = 3
        b = m(input_ids=ids2)["prediction_scores"]
    assert torch.allclose(a[:, :-1], b[:, :-1], atol=1e-5)


def test_llama_gqa_fwd_bwd_and_decode_parity():
    """GQA (num_key_value_heads < num_heads): train step + incremental
    decode == full forward; the KV cache stores only the KV heads."""
    from libai_amd.models import LlamaForCausalLM

    torch.manual_seed(0)
    m = LlamaForCausalLM(hidden_layers=2, vocab_size=128, hidden_size=64,
                         intermediate_size=128, num_attention_heads=8,
                         num_key_value_heads=2,
                         max_position_embeddings=64).eval()
    ids = torch.randint(0, 128, (2, 18))
    out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
    out["lm_loss"].backward()
    assert torch.isfinite(out["lm_loss"])
    with torch.no_grad():
        full = m(input_ids=ids)["prediction_scores"]
        o = m(input_ids=ids[:, :10], use_cache=True)
        st = m(input_ids=ids[:, 10:11],
               past_key_values=o["past_key_values"], use_cache=True)
    assert (st["prediction_scores"][:, 0] - full[:, 10]).abs().max() < 1e-4
    assert o["past_key_values"][0][0].shape[1] == 2  # kv heads only


def test_palm_parallel_block_and_decode_parity():
    """PaLM: parallel attn+MLP residual, multi-query attention (kv=1),
    tied logits; incremental decode == full forward."""
    from libai_amd.models import PaLMForCausalLM

    torch.manual_seed(0)
    m = PaLMForCausalLM(hidden_layers=2, vocab_size=128, hidden_size=64,
                        intermediate_size=128, num_attention_heads=8,
                        max_position_embeddings=64).eval()
    ids = torch.randint(0, 128, (2, 18))
    out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
    out["lm_loss"].backward()
    assert torch.isfinite(out["lm_loss"])
    # MQA: one kv head
    assert m.model.layers[0].attn.num_kv_heads == 1
    with torch.no_grad():
        full = m(input_ids=ids)["prediction_scores"]
        o = m(input_ids=ids[:, :10], use_cache=True)
        st = m(input_ids=ids[:, 10:11],
               past_key_values=o["past_key_values"], use_cache=True)
    assert (st["prediction_scores"][:, 0] - full[:, 10]).abs().max() < 1e-4
