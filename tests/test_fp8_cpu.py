"""fp8 GEMM option: gating logic on CPU (the compute path is GPU-only)."""

import torch

from libai_amd.utils import distributed as du

du.setup_dist_util({})


def test_fp8_gating_and_fallback():
    from libai_amd.layers import Linear1D
    from libai_amd.ops import fp8

    assert not fp8.fp8_gemms_enabled()
    fp8.set_fp8_gemms(True)
    try:
        x = torch.randn(4, 32)
        w = torch.randn(16, 32)
        # CPU tensors are never eligible -> F.linear fallback, same result
        assert not fp8.fp8_eligible(x, w)
        lin = Linear1D(32, 16, parallel="data")
        y = lin(x)
        assert y.shape == (4, 16)
        # shape gating: non-multiple-of-16 dims are ineligible even on GPU
        assert not fp8.fp8_eligible(torch.randn(4, 30), torch.randn(17, 30))
    finally:
        fp8.set_fp8_gemms(False)


def test_fp8_config_wiring():
    from libai_amd.config import ConfigDict
    from libai_amd.engine.default import default_setup
    from libai_amd.ops import fp8
    import tempfile

    with tempfile.TemporaryDirectory() as d:
        cfg = ConfigDict(train=ConfigDict(
            output_dir=d, fp8=dict(enabled=True),
            train_micro_batch_size=1, dist=dict(),
        ))
        try:
            default_setup(cfg)
            assert fp8.fp8_gemms_enabled()
        finally:
            fp8.set_fp8_gemms(False)
