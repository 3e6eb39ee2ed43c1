"""SimCSE + MoCo v3 contrastive models (reference: projects/SimCSE,
projects/MOCOV3)."""

import torch

from libai_amd.models import MoCoV3, SimCSEModel
from libai_amd.utils import distributed as du

du.setup_dist_util({})

BERT_KW = dict(vocab_size=128, hidden_size=32, hidden_layers=2,
               num_attention_heads=4, intermediate_size=64,
               max_position_embeddings=64, hidden_dropout_prob=0.1,
               attention_probs_dropout_prob=0.1)


def test_simcse_learns_and_normalizes():
    torch.manual_seed(0)
    m = SimCSEModel(**BERT_KW)
    ids = torch.randint(0, 128, (8, 16))
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    first = None
    for _ in range(15):
        opt.zero_grad()
        loss = m(input_ids=ids)["contrastive_loss"]
        loss.backward()
        opt.step()
        first = first if first is not None else float(loss)
    assert float(loss) < first
    m.eval()
    with torch.no_grad():
        e = m(input_ids=ids)["embeddings"]
    assert torch.allclose(e.norm(dim=-1), torch.ones(8), atol=1e-4)
    # eval embeddings are deterministic (dropout off)
    with torch.no_grad():
        e2 = m(input_ids=ids)["embeddings"]
    assert torch.equal(e, e2)


def test_moco_momentum_and_loss():
    torch.manual_seed(0)
    m = MoCoV3(img_size=32, patch_size=8, embed_dim=32, depth=2, num_heads=4,
               proj_dim=16, proj_hidden=32, momentum=0.9)
    imgs = torch.randn(4, 3, 32, 32)
    out = m(images=imgs, images2=imgs.flip(-1))
    out["moco_loss"].backward()
    # momentum encoder holds no grads
    assert all(p.grad is None for p in m.momentum_backbone.parameters())
    # EMA math: k' = m*k + (1-m)*q
    q0 = next(m.backbone.parameters()).detach().clone()
    k0 = next(m.momentum_backbone.parameters()).detach().clone()
    with torch.no_grad():
        next(m.backbone.parameters()).add_(1.0)
    m.update_momentum_encoder()
    k1 = next(m.momentum_backbone.parameters()).detach()
    want = 0.9 * k0 + 0.1 * (q0 + 1.0)
    assert torch.allclose(k1, want, atol=1e-6)


def test_trainer_calls_momentum_update():
    from libai_amd.engine.trainer import EagerTrainer
    from libai_amd.optim import FusedAdamW

    torch.manual_seed(0)
    m = MoCoV3(img_size=32, patch_size=8, embed_dim=32, depth=1, num_heads=4,
               proj_dim=8, proj_hidden=16, momentum=0.5)
    calls = []
    orig = m.update_momentum_encoder
    m.update_momentum_encoder = lambda: calls.append(1) or orig()

    data = [{"images": torch.randn(2, 3, 32, 32)} for _ in range(4)]
    opt = FusedAdamW([p for p in m.parameters() if p.requires_grad], lr=1e-3)
    tr = EagerTrainer(m, data, opt, grad_acc_steps=1)
    tr.train(0, 2)
    assert len(calls) == 2


def test_clip_learns_and_inference_shapes():
    from libai_amd.models import CLIPModel

    torch.manual_seed(0)
    m = CLIPModel(embed_dim=32, img_size=32, patch_size=8, vision_width=32,
                  vision_layers=2, vision_heads=4, vocab_size=128,
                  context_length=16, text_width=32, text_layers=2, text_heads=4)
    imgs = torch.randn(4, 3, 32, 32)
    txt = torch.randint(0, 127, (4, 12))
    txt[:, -1] = 127  # eot = max id
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    first = None
    for _ in range(12):
        opt.zero_grad()
        loss = m(images=imgs, text_ids=txt)["clip_loss"]
        loss.backward()
        opt.step()
        first = first if first is not None else float(loss)
    assert float(loss) < first, (first, float(loss))
    m.eval()
    with torch.no_grad():
        out = m(images=imgs, text_ids=txt)
        assert out["logits_per_image"].shape == (4, 4)
        zi = m(images=imgs)["image_embeds"]
        zt = m(text_ids=txt)["text_embeds"]
    assert torch.allclose(zi.norm(dim=-1), torch.ones(4), atol=1e-4)
    assert torch.allclose(zt.norm(dim=-1), torch.ones(4), atol=1e-4)
