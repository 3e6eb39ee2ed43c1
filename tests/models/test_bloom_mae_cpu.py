"""BLOOM (ALiBi), MAE, and T5 relative-position-bias correctness on CPU."""

import math

import pytest
import torch

from libai_amd.utils import distributed as du

du.setup_dist_util({})


def test_alibi_slopes_known_values():
    from libai_amd.layers.position_bias import alibi_slopes

    s8 = alibi_slopes(8)
    assert s8[0] == pytest.approx(0.5)
    assert s8[7] == pytest.approx(0.5 ** 8)
    # non-power-of-two head count interleaves the 2x schedule
    s6 = alibi_slopes(6)
    assert len(s6) == 6 and all(s6[i] > s6[i + 1] for i in range(3))


def test_bloom_alibi_matches_manual_reference():
    """BLOOM attention == manual softmax(scale*QK^T + slope*(j-i)) ref."""
    from libai_amd.models import BloomForCausalLM

    torch.manual_seed(0)
    m = BloomForCausalLM(vocab_size=64, hidden_size=32, hidden_layers=1,
                         num_attention_heads=4).eval()
    ids = torch.randint(0, 64, (2, 12))
    with torch.no_grad():
        out = m(input_ids=ids)["prediction_scores"]

    # manual forward through the same weights
    bl = m.bloom
    layer = bl.layers[0]
    with torch.no_grad():
        h = bl.word_embeddings_layernorm(bl.word_embeddings(ids))
        ln1 = layer.input_layernorm(h)
        qkv = layer.self_attention.query_key_value(ln1)
        b, s, _ = qkv.shape
        q, k, v = (qkv.view(b, s, 4, 3, 8).permute(0, 2, 1, 3, 4).unbind(3))
        scores = q @ k.transpose(-1, -2) / math.sqrt(8)
        from libai_amd.layers.position_bias import alibi_slopes

        slopes = alibi_slopes(4)
        i = torch.arange(s)[:, None]
        j = torch.arange(s)[None, :]
        rel = slopes[:, None, None] * (j - i)  # == slope*j up to row consts
        scores = scores + rel[None]
        scores = scores.masked_fill(j > i, float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        ctx = (probs @ v).permute(0, 2, 1, 3).reshape(b, s, 32)
        attn_out, bias = layer.self_attention.dense(ctx)
        attn = h + attn_out + bias
        ln2 = layer.post_attention_layernorm(attn)
        mlp = layer.mlp(ln2, residual=attn)
        ref = bl.lm_head(bl.ln_f(mlp), bl.word_embeddings.weight)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


def test_bloom_incremental_decode_matches_full():
    from libai_amd.models import BloomForCausalLM

    torch.manual_seed(1)
    m = BloomForCausalLM(vocab_size=64, hidden_size=32, hidden_layers=2,
                         num_attention_heads=4).eval()
    ids = torch.randint(0, 64, (1, 10))
    with torch.no_grad():
        full = m(input_ids=ids)["prediction_scores"]
        out = m(input_ids=ids[:, :6], use_cache=True)
        step = m(input_ids=ids[:, 6:7],
                 past_key_values=out["past_key_values"], use_cache=True)
    assert torch.allclose(step["prediction_scores"][:, 0], full[:, 6],
                          atol=1e-4)


def test_mae_masking_and_learning():
    from libai_amd.models import MAEForPreTraining

    torch.manual_seed(0)
    m = MAEForPreTraining(img_size=32, patch_size=8, embed_dim=64, depth=2,
                          num_heads=4, decoder_embed_dim=32, decoder_depth=1,
                          decoder_num_heads=4, mask_ratio=0.75)
    imgs = torch.randn(4, 3, 32, 32)
    # masking stats: 16 patches, keep 4
    x = m.patch_embed(imgs)
    _, mask, restore = m.random_masking(x)
    assert mask.shape == (4, 16)
    assert (mask.sum(dim=1) == 12).all()  # 75% masked
    # restore is a permutation inverse
    assert torch.equal(torch.sort(restore, dim=1).values,
                       torch.arange(16).expand(4, -1))

    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    first = None
    for _ in range(20):
        opt.zero_grad()
        loss = m(images=imgs)["mae_loss"]
        loss.backward()
        opt.step()
        first = first if first is not None else float(loss)
    assert float(loss) < first, (first, float(loss))


def test_t5_relpos_bucket_properties():
    from libai_amd.layers.position_bias import T5RelativePositionBias

    rel = torch.arange(-20, 21)[None, :]
    b_bi = T5RelativePositionBias._bucket(rel, True, 32, 128)
    assert (b_bi >= 0).all() and (b_bi < 32).all()
    assert b_bi[0, 20] == 0  # distance 0
    # bidirectional: past and future land in different halves
    assert b_bi[0, 0] != b_bi[0, 40]
    b_uni = T5RelativePositionBias._bucket(rel, False, 32, 128)
    assert (b_uni[0, rel[0] > 0] == 0).all()  # future clamps to bucket 0


def test_t5_relpos_grads_and_no_abs_positions():
    from libai_amd.models.t5_model import T5ForPreTraining

    torch.manual_seed(0)
    m = T5ForPreTraining(vocab_size=64, hidden_size=32, hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64,
                         relative_attention=True, hidden_dropout_prob=0.0,
                         attention_probs_dropout_prob=0.0,
                         embedding_dropout_prob=0.0)
    assert m.t5_model.embedding.position_embeddings is None
    ids = torch.randint(0, 64, (2, 10))
    out = m(encoder_input_ids=ids, decoder_input_ids=ids[:, :8],
            lm_labels=ids[:, :8], loss_mask=torch.ones(2, 8, dtype=torch.long))
    out["masked_lm_loss"].backward()
    assert m.t5_model.enc_rel_bias.weight.grad is not None
    assert m.t5_model.dec_rel_bias.weight.grad is not None
    assert m.t5_model.enc_rel_bias.weight.grad.abs().sum() > 0
