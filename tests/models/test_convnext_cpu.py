"""ConvNeXt: forward/backward shapes, LayerScale wiring, config instantiation
(reference capability: projects/ConvNeXT)."""

import torch

from libai_amd.utils import distributed as du

du.setup_dist_util({})


def _tiny():
    from libai_amd.models import ConvNeXt

    torch.manual_seed(0)
    return ConvNeXt(img_size=32, num_classes=10, depths=(1, 1, 2, 1),
                    dims=(16, 32, 64, 128), drop_path_rate=0.1)


def test_convnext_forward_backward():
    m = _tiny()
    imgs = torch.randn(2, 3, 32, 32)
    labels = torch.randint(0, 10, (2,))
    out = m(images=imgs, labels=labels)
    assert out["losses"].ndim == 0
    out["losses"].backward()
    assert m.head.weight.grad is not None
    assert m.stages[0][0].dwconv.weight.grad is not None
    assert m.stages[0][0].gamma.grad is not None  # LayerScale learns


def test_convnext_eval_deterministic():
    m = _tiny().eval()
    imgs = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        a = m(images=imgs)["prediction_scores"]
        b = m(images=imgs)["prediction_scores"]
    assert torch.equal(a, b)  # drop_path disabled in eval
    assert a.shape == (2, 10)


def test_convnext_config_instantiates():
    from libai_amd.config import LazyCall, instantiate
    from libai_amd.models import ConvNeXt

    cfg = LazyCall(ConvNeXt)(img_size=32, num_classes=7, depths=(1, 1, 1, 1),
                             dims=(8, 16, 32, 64))
    m = instantiate(cfg)
    assert m.head.out_features == 7
