import math

import pytest
import torch
import torch.nn.functional as F

from libai_amd import layers
from libai_amd.utils import distributed as du

du.setup_dist_util({})


def test_linear_matches_torch():
    torch.manual_seed(0)
    lin = layers.Linear1D(32, 64, parallel="data")
    ref = torch.nn.Linear(32, 64)
    with torch.no_grad():
        ref.weight.copy_(lin.weight)
        ref.bias.copy_(lin.bias)
    x = torch.randn(4, 32)
    assert torch.allclose(lin(x), ref(x), atol=1e-6)


def test_linear_skip_bias_add():
    lin = layers.Linear1D(8, 8, skip_bias_add=True)
    out, bias = lin(torch.randn(2, 8))
    assert out.shape == (2, 8) and bias.shape == (8,)


def test_attention_causal_masks_future():
    torch.manual_seed(0)
    attn = layers.MultiheadAttention(32, 4, attn_mask_type="causal")
    attn.eval()
    x = torch.randn(1, 8, 32)
    y1 = attn(x)
    # changing a FUTURE token must not change past outputs
    x2 = x.clone()
    x2[0, -1] += 10.0
    y2 = attn(x2)
    assert torch.allclose(y1[0, :-1], y2[0, :-1], atol=1e-5)
    assert not torch.allclose(y1[0, -1], y2[0, -1], atol=1e-3)


def test_attention_kv_cache_matches_full():
    torch.manual_seed(0)
    attn = layers.MultiheadAttention(32, 4, attn_mask_type="causal")
    attn.eval()
    x = torch.randn(1, 6, 32)
    full = attn(x)
    out1, kv = attn(x[:, :5], use_cache=True)
    out2, _ = attn(x[:, 5:6], past_key_value=kv, use_cache=True)
    assert torch.allclose(full[:, :5], out1, atol=1e-5)
    assert torch.allclose(full[:, 5:6], out2, atol=1e-5)


def test_layernorm_matches_torch():
    torch.manual_seed(0)
    ln = layers.LayerNorm(64)
    with torch.no_grad():
        ln.weight.normal_()
        ln.bias.normal_()
    x = torch.randn(3, 5, 64)
    ref = F.layer_norm(x, (64,), ln.weight, ln.bias, 1e-5)
    assert torch.allclose(ln(x), ref, atol=1e-6)


def test_rmsnorm_math():
    torch.manual_seed(0)
    rms = layers.RMSLayerNorm(16)
    x = torch.randn(4, 16)
    y = rms(x)
    expected = x / torch.sqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6)
    assert torch.allclose(y, expected, atol=1e-5)


def test_vocab_embedding_single_rank():
    emb = layers.VocabEmbedding(100, 32)
    ids = torch.randint(0, 100, (2, 7))
    out = emb(ids)
    assert out.shape == (2, 7, 32)
    assert torch.allclose(out, F.embedding(ids, emb.weight))


def test_parallel_ce_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(4, 9, 50, requires_grad=True)
    target = torch.randint(0, 50, (4, 9))
    ce = layers.ParallelCrossEntropyLoss()
    loss = ce(logits, target)
    ref = F.cross_entropy(
        logits.reshape(-1, 50), target.reshape(-1), reduction="none"
    ).view(4, 9)
    assert torch.allclose(loss, ref, atol=1e-5)
    loss.mean().backward()
    assert logits.grad is not None and torch.isfinite(logits.grad).all()


def test_mlp_shapes_and_grad():
    mlp = layers.MLP(32, 128, output_dropout_prob=0.0)
    x = torch.randn(2, 5, 32, requires_grad=True)
    y = mlp(x, residual=x)
    assert y.shape == x.shape
    y.sum().backward()
    assert x.grad is not None


def test_transformer_layer_decoder_cross_attention():
    layer = layers.TransformerLayer(32, 128, 4, is_decoder=True)
    x = torch.randn(2, 5, 32)
    enc = torch.randn(2, 7, 32)
    y = layer(x, encoder_states=enc)
    assert y.shape == x.shape


def test_activation_registry():
    for name in ["gelu", "tanh", "relu", "quick_gelu", "squared_relu"]:
        act = layers.build_activation(name)
        out = act(torch.randn(4))
        assert out.shape == (4,)
    with pytest.raises(KeyError):
        layers.build_activation("nope")


def test_droppath_eval_identity():
    dp = layers.DropPath(0.5)
    dp.eval()
    x = torch.randn(3, 4)
    assert torch.equal(dp(x), x)


def test_conv1d_matches_linear():
    torch.manual_seed(0)
    conv = layers.Conv1D(16, 24)
    x = torch.randn(2, 16)
    ref = x @ conv.weight + conv.bias
    assert torch.allclose(conv(x), ref, atol=1e-6)


def test_sine_positional_embedding():
    pe = layers.SinePositionalEmbedding(64, 32)
    pos = torch.arange(10).unsqueeze(0)
    out = pe(pos)
    assert out.shape == (1, 10, 32)
    # first position: sin(0)=0, cos(0)=1
    assert torch.allclose(out[0, 0, 0::2], torch.zeros(16), atol=1e-6)
    assert torch.allclose(out[0, 0, 1::2], torch.ones(16), atol=1e-6)


def test_patch_embedding():
    pe = layers.PatchEmbedding(img_size=32, patch_size=8, in_chans=3, embed_dim=24)
    x = torch.randn(2, 3, 32, 32)
    out = pe(x)
    assert out.shape == (2, 16, 24)
