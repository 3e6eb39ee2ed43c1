"""LoRA under TP=2 (gloo): sharded forward/backward == replicated reference.

Regression for ADVICE round 1: the row-parallel A-projection is a partial sum
that must be all-reduced in forward; the col-parallel bottleneck needs a
backward all-reduce so the replicated lora_A gets coherent gradients.
"""

import pytest
import torch

from tests.dist_helper import run_dist


def _worker(rank, world, parallel):
    import torch

    from libai_amd import layers
    from libai_amd.lora import LoRALinear
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    in_f, out_f, r, bsz = 16, 24, 4, 5
    torch.manual_seed(42)
    base = layers.Linear1D(in_f, out_f, parallel=parallel)
    torch.manual_seed(42)
    base_full = layers.Linear1D(in_f, out_f, parallel="data")

    torch.manual_seed(7)
    lora = LoRALinear(base, r=r, alpha=8, dropout=0.0)
    # nonzero factors, sharded consistently with a full reference
    torch.manual_seed(11)
    A_full = torch.randn(r, in_f) * 0.1
    B_full = torch.randn(out_f, r) * 0.1
    with torch.no_grad():
        if parallel == "col":
            lora.lora_A.copy_(A_full)
            lora.lora_B.copy_(B_full.chunk(2, dim=0)[rank])
        else:  # row: A sharded on input dim
            lora.lora_A.copy_(A_full.chunk(2, dim=1)[rank])
            lora.lora_B.copy_(B_full)

    torch.manual_seed(3)
    x_full = torch.randn(bsz, in_f, requires_grad=True)
    scaling = lora.scaling
    ref = base_full(x_full) + (x_full @ A_full.t() @ B_full.t()) * scaling
    ref.pow(2).mean().backward()

    if parallel == "row":
        x_local = x_full.detach().chunk(2, dim=-1)[rank].clone().requires_grad_(True)
    else:
        x_local = x_full.detach().clone().requires_grad_(True)
    out = lora(x_local)
    if parallel == "col":
        assert torch.allclose(out, ref.detach().chunk(2, dim=-1)[rank], atol=1e-5), \
            "col fwd mismatch"
        # grads: reproduce ref loss = mean over the FULL output
        loss = out.pow(2).sum() / ref.numel()
        # the other shard's contribution is on the other rank; per-rank partial
        # losses sum to the full loss, and cross terms vanish for pow(2)
        loss.backward()
        gA_ref = _ref_grad_A(x_full, A_full, B_full, base_full, scaling)
        assert torch.allclose(lora.lora_A.grad, gA_ref, atol=1e-5), \
            f"col lora_A grad mismatch {(lora.lora_A.grad - gA_ref).abs().max()}"
        gB_ref = _ref_grad_B(x_full, A_full, B_full, base_full, scaling)
        assert torch.allclose(lora.lora_B.grad, gB_ref.chunk(2, dim=0)[rank],
                              atol=1e-5), "col lora_B grad mismatch"
        assert torch.allclose(x_local.grad, x_full.grad, atol=1e-5), \
            "col x grad mismatch"
    else:
        assert torch.allclose(out, ref.detach(), atol=1e-5), "row fwd mismatch"
        loss = out.pow(2).mean()
        loss.backward()
        gA_ref = _ref_grad_A(x_full, A_full, B_full, base_full, scaling)
        assert torch.allclose(lora.lora_A.grad, gA_ref.chunk(2, dim=1)[rank],
                              atol=1e-5), "row lora_A grad mismatch"
        gB_ref = _ref_grad_B(x_full, A_full, B_full, base_full, scaling)
        assert torch.allclose(lora.lora_B.grad, gB_ref, atol=1e-5), \
            f"row lora_B grad mismatch {(lora.lora_B.grad - gB_ref).abs().max()}"
        assert torch.allclose(x_local.grad, x_full.grad.chunk(2, dim=-1)[rank],
                              atol=1e-5), "row x grad mismatch"
    return True


def _ref_grad_A(x_full, A_full, B_full, base_full, scaling):
    A = A_full.clone().requires_grad_(True)
    B = B_full.clone().requires_grad_(True)
    x = x_full.detach().clone().requires_grad_(True)
    ref = base_full(x) + (x @ A.t() @ B.t()) * scaling
    ref.pow(2).mean().backward()
    return A.grad


def _ref_grad_B(x_full, A_full, B_full, base_full, scaling):
    A = A_full.clone().requires_grad_(True)
    B = B_full.clone().requires_grad_(True)
    x = x_full.detach().clone().requires_grad_(True)
    ref = base_full(x) + (x @ A.t() @ B.t()) * scaling
    ref.pow(2).mean().backward()
    return B.grad


@pytest.mark.parametrize("parallel", ["col", "row"])
def test_lora_tp2_matches_replicated(parallel):
    run_dist(_worker, 2, args=(parallel,))
