"""Expert-parallel MoE: ep2 (all-to-all dispatch) == single-process with all
experts local — the sharded-vs-replicated oracle for EP (beyond the
reference's feature set; EP named in the MI355X RCCL axis list)."""

import pytest
import torch

from tests.dist_helper import run_dist

H, FFN, E, TOPK, N = 32, 64, 4, 2, 10


def _tokens():
    torch.manual_seed(9)
    return torch.randn(2, 5, H)


def _build(seed=123):
    from libai_amd.layers.moe import MoELayer

    torch.manual_seed(seed)
    return MoELayer(H, FFN, num_experts=E, top_k=TOPK)


def _reference():
    from libai_amd.utils import distributed as du

    du._DIST_UTIL = None
    du.setup_dist_util({})
    layer = _build()
    x = _tokens().requires_grad_(True)
    y = layer(x)
    (y.pow(2).mean() + layer.last_aux_loss).backward()
    grads = {n: p.grad.clone() for n, p in layer.named_parameters()}
    return y.detach(), grads, x.grad.clone()


def _ep2_worker(rank, world):
    import torch

    from libai_amd.optim import FusedAdamW
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})  # dp2 == ep2
    layer = _build()
    # experts are a SLICE of the same full init
    assert layer.w1.shape[0] == E // 2
    opt = FusedAdamW(layer.parameters(), lr=0.0)
    opt.zero_grad()
    x = _tokens().requires_grad_(True)
    y = layer(x)
    (y.pow(2).mean() + layer.last_aux_loss).backward()
    # grad_sync applies the DP average to dense grads and the matching 1/dp
    # normalization (no collective) to expert grads: with both ranks running
    # the SAME batch, everything must equal the single-process reference
    opt.grad_sync()
    grads = {n: p.grad.clone() for n, p in layer.named_parameters()}
    return y.detach(), grads, x.grad.clone()


def test_moe_ep2_matches_single_process():
    ref_y, ref_g, ref_xg = _reference()
    results = run_dist(_ep2_worker, 2)
    for rank, (y, grads, xg) in enumerate(results):
        assert torch.allclose(y, ref_y, atol=1e-5), (y - ref_y).abs().max()
        assert torch.allclose(xg, ref_xg, atol=1e-5), "dx mismatch"
        lo, hi = rank * E // 2, (rank + 1) * E // 2
        for name in ("w1", "b1", "w2", "b2"):
            want = ref_g[name][lo:hi]
            got = grads[name]
            assert torch.allclose(got, want, atol=1e-5), (
                f"{name} grad mismatch (rank {rank}): "
                f"{(got - want).abs().max()}"
            )
        assert torch.allclose(grads["router.weight"], ref_g["router.weight"],
                              atol=1e-5), "router grad mismatch"


def _ep2_train_worker(rank, world):
    import torch
    import torch.distributed as dist

    from libai_amd.layers import TransformerLayer
    from libai_amd.optim import FusedAdamW
    from libai_amd.utils import distributed as du

    du.setup_dist_util({})
    torch.manual_seed(0)
    layer = TransformerLayer(H, FFN, 4, mlp_type="moe", moe_num_experts=E,
                             moe_top_k=TOPK)
    opt = FusedAdamW(layer.parameters(), lr=1e-3, clip_grad=1.0)
    torch.manual_seed(50 + rank)  # different data per DP rank
    for _ in range(3):
        opt.zero_grad()
        x = torch.randn(2, 8, H)
        y = layer(x)
        (y.pow(2).mean() + layer.mlp.last_aux_loss).backward()
        opt.grad_sync()
        opt.step()
    # NON-expert params stay DP-identical; expert params may differ
    flats = torch.cat([
        p.detach().reshape(-1) for n, p in layer.named_parameters()
        if not getattr(p, "expert_parallel", False)
    ])
    gathered = [torch.empty_like(flats) for _ in range(world)]
    dist.all_gather(gathered, flats)
    assert torch.allclose(gathered[0], gathered[1], atol=1e-6), \
        "non-expert DP divergence"
    ew = layer.mlp.w1.detach().reshape(-1)
    eg = [torch.empty_like(ew) for _ in range(world)]
    dist.all_gather(eg, ew)
    assert not torch.allclose(eg[0], eg[1]), "experts should differ per rank"
    return True


def test_moe_ep2_training_with_fused_adamw():
    """End-to-end: expert grads skip the DP all-reduce, the rest stays
    DP-synced, grad-norm clip runs without divergent collectives."""
    run_dist(_ep2_train_worker, 2)


def _ep2_ckpt_worker(rank, world, tmpdir):
    import os

    import torch

    from libai_amd.layers import TransformerLayer
    from libai_amd.optim import FusedAdamW
    from libai_amd.utils import distributed as du
    from libai_amd.utils.checkpoint import Checkpointer

    du.setup_dist_util({})
    torch.manual_seed(0)
    layer = TransformerLayer(H, FFN, 4, mlp_type="moe", moe_num_experts=E,
                             moe_top_k=TOPK)
    opt = FusedAdamW(layer.parameters(), lr=1e-3)
    opt.set_param_names(layer.named_parameters())
    torch.manual_seed(50 + rank)
    for _ in range(2):
        opt.zero_grad()
        y = layer(torch.randn(2, 8, H))
        (y.pow(2).mean() + layer.mlp.last_aux_loss).backward()
        opt.grad_sync()
        opt.step()
    ck = Checkpointer(layer, tmpdir, optimizer=opt)
    ck.save("moe")
    my_w1 = layer.mlp.w1.detach().clone()

    # resume into a FRESH ep2 replica: each rank's experts must come back
    torch.manual_seed(777 + rank)
    layer2 = TransformerLayer(H, FFN, 4, mlp_type="moe", moe_num_experts=E,
                              moe_top_k=TOPK)
    opt2 = FusedAdamW(layer2.parameters(), lr=1e-3)
    opt2.set_param_names(layer2.named_parameters())
    ck2 = Checkpointer(layer2, tmpdir, optimizer=opt2)
    ck2.load(os.path.join(tmpdir, "moe"))
    assert torch.allclose(layer2.mlp.w1.detach(), my_w1, atol=1e-6), \
        "expert weights did not round-trip per EP rank"
    # the saved model.pt holds the FULL expert dim (EP-independent)
    full = torch.load(os.path.join(tmpdir, "moe", "model.pt"),
                      map_location="cpu", weights_only=False)
    assert full["mlp.w1"].shape[0] == E
    return True


@pytest.mark.timeout(300)
def test_moe_checkpoint_roundtrip_ep2(tmp_path):
    run_dist(_ep2_ckpt_worker, 2, args=(str(tmp_path),))
