"""TP correctness-by-equivalence: sharded == replicated (the reference's main
oracle, tests/layers/test_linear.py:59-114), run as 2-process gloo on CPU."""

import pytest
import torch

from tests.dist_helper import run_dist


def _tp2_linear_worker(rank, world, parallel):
    import torch

    from libai_amd import layers
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    torch.manual_seed(42)  # identical init across TP ranks
    lin = layers.Linear1D(16, 24, parallel=parallel)
    torch.manual_seed(42)
    full = layers.Linear1D(16, 24, parallel="data")

    torch.manual_seed(7)
    x_full = torch.randn(4, 16, requires_grad=True)
    if parallel == "row":
        # row-parallel input is split along the last dim
        x_local = x_full.detach().chunk(2, dim=-1)[rank].clone().requires_grad_(True)
        out = lin(x_local)
    else:
        x_local = x_full.detach().clone().requires_grad_(True)
        out = lin(x_local)

    ref = full(x_full)
    if parallel == "col":
        ref_shard = ref.chunk(2, dim=-1)[rank]
        assert torch.allclose(out, ref_shard, atol=1e-5), "col fwd mismatch"
    else:
        assert torch.allclose(out, ref, atol=1e-5), "row fwd mismatch"

    # backward equivalence
    g = torch.ones_like(ref)
    ref.backward(g)
    if parallel == "col":
        out.backward(torch.ones_like(out))
        assert torch.allclose(x_local.grad, x_full.grad, atol=1e-5), "col dx mismatch"
        wg_ref = full.weight.grad.chunk(2, dim=0)[rank]
        assert torch.allclose(lin.weight.grad, wg_ref, atol=1e-5), "col dw mismatch"
    else:
        out.backward(torch.ones_like(out))
        dx_ref = x_full.grad.chunk(2, dim=-1)[rank]
        assert torch.allclose(x_local.grad, dx_ref, atol=1e-5), "row dx mismatch"
        wg_ref = full.weight.grad.chunk(2, dim=1)[rank]
        assert torch.allclose(lin.weight.grad, wg_ref, atol=1e-5), "row dw mismatch"
    return True


@pytest.mark.parametrize("parallel", ["col", "row"])
def test_tp2_linear_equivalence(parallel):
    assert all(run_dist(_tp2_linear_worker, 2, args=(parallel,)))


def _tp2_vocab_embedding_worker(rank, world):
    import torch
    import torch.nn.functional as F

    from libai_amd import layers
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    torch.manual_seed(3)
    emb = layers.VocabEmbedding(64, 16)
    torch.manual_seed(3)
    full = layers.VocabEmbedding.__new__(layers.VocabEmbedding)
    # reconstruct the full table the same way init_tp_shard_ drew it
    g = torch.empty(64, 16)
    torch.manual_seed(3)
    torch.nn.init.xavier_normal_(g)
    ids = torch.randint(0, 64, (2, 9))
    out = emb(ids)
    ref = F.embedding(ids, g)
    assert torch.allclose(out, ref, atol=1e-5), "vocab embedding mismatch"
    return True


def test_tp2_vocab_embedding():
    assert all(run_dist(_tp2_vocab_embedding_worker, 2))


def _tp2_ce_worker(rank, world):
    import torch
    import torch.nn.functional as F

    from libai_amd import layers
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    torch.manual_seed(11)
    logits_full = torch.randn(6, 40)
    target = torch.randint(0, 40, (6,))
    local = logits_full.chunk(2, dim=-1)[rank].clone().requires_grad_(True)
    ce = layers.ParallelCrossEntropyLoss()
    loss = ce(local, target)
    ref = F.cross_entropy(logits_full, target, reduction="none")
    assert torch.allclose(loss, ref, atol=1e-5), "vocab-parallel CE fwd mismatch"

    loss.mean().backward()
    lf = logits_full.clone().requires_grad_(True)
    F.cross_entropy(lf, target).backward()
    dref = lf.grad.chunk(2, dim=-1)[rank]
    assert torch.allclose(local.grad, dref, atol=1e-5), "vocab-parallel CE bwd mismatch"
    return True


def test_tp2_parallel_cross_entropy():
    assert all(run_dist(_tp2_ce_worker, 2))


def _tp2_transformer_worker(rank, world):
    """Full TransformerLayer sharded-vs-replicated equivalence (eval mode)."""
    import torch

    from libai_amd import layers
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    torch.manual_seed(5)
    tp_layer = layers.TransformerLayer(32, 128, 4, attn_mask_type="causal")
    tp_layer.eval()
    x = torch.randn(2, 8, 32)
    out_tp = tp_layer(x)

    # single-process replicated reference built with the same RNG stream
    import os

    del os.environ["WORLD_SIZE"]
    import torch.distributed as dist

    du._DIST_UTIL = None
    saved = dist.is_initialized()
    # emulate tp=1 by building on a fresh dist util without groups
    du._DIST_UTIL = du._DistributeUtil.__new__(du._DistributeUtil)
    d = du._DIST_UTIL
    d._world_size, d._rank, d._local_rank = 1, 0, 0
    d._dp_size, d._tp_size, d._pp_size = 1, 1, 1
    d._dp_rank, d._tp_rank, d._pp_rank = 0, 0, 0
    d._pipeline_num_layers, d._custom_stage_id = None, None
    d._tp_group = d._dp_group = d._pp_group = d._dp_tp_group = None
    torch.manual_seed(5)
    ref_layer = layers.TransformerLayer(32, 128, 4, attn_mask_type="causal")
    ref_layer.eval()
    out_ref = ref_layer(x)
    assert torch.allclose(out_tp, out_ref, atol=1e-4), (
        f"transformer TP mismatch: {(out_tp - out_ref).abs().max()}"
    )
    return True


def test_tp2_transformer_layer_equivalence():
    assert all(run_dist(_tp2_transformer_worker, 2))


def _tp2_gated_mlp_worker(rank, world):
    import torch
    import torch.distributed as dist

    from libai_amd.layers.linear import tp_merge
    from libai_amd.layers.mlp import GatedMLP
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    dutil = du.get_dist_util()
    torch.manual_seed(42)
    sharded = GatedMLP(16, 32, activation="silu")

    # reconstruct the CANONICAL full weights from the shards and compute the
    # gated MLP in plain torch: the [gate|up] projection must shard as PAIRED
    # slices ([gate_r | up_r]) for the local swiglu halves to mean the same
    # thing the canonical [gate | up] layout does
    def gather_full(p, fused):
        shards = [torch.empty_like(p.data) for _ in range(2)]
        dist.all_gather(shards, p.data.contiguous(),
                        group=dutil.tensor_parallel_group)
        return tp_merge(shards, p.tp_shard_dim, fused)

    gu_full = gather_full(sharded.gate_up_proj.weight,
                          getattr(sharded.gate_up_proj.weight,
                                  "tp_fused_chunks", 1))
    w2_full = gather_full(sharded.down_proj.weight, 1)

    torch.manual_seed(7)
    x = torch.randn(4, 16)
    out_s = sharded(x)
    g, u = gu_full.chunk(2, dim=0)
    ref = (torch.nn.functional.silu(x @ g.t()) * (x @ u.t())) @ w2_full.t()
    assert torch.allclose(out_s, ref, atol=1e-5), (
        f"gated MLP sharded != canonical: {(out_s - ref).abs().max()}"
    )


def test_tp2_gated_mlp_equivalence():
    run_dist(_tp2_gated_mlp_worker, world_size=2)


def _tp2_optimizer_state_canonical_worker(rank, world):
    import torch

    from libai_amd.layers import Linear1D
    from libai_amd.optim import FusedAdamW
    from libai_amd.utils import distributed as du

    du.setup_dist_util({"tensor_parallel_size": 2})
    torch.manual_seed(42)
    lin = Linear1D(16, 24, parallel="col")
    opt = FusedAdamW(lin.parameters(), lr=1e-2)
    out = lin(torch.randn(4, 16))
    (out[0] if isinstance(out, tuple) else out).pow(2).mean().backward()
    opt.grad_sync()
    opt.step()

    sd = opt.state_dict()
    # TP-sharded params save CANONICAL (full) tensors -> topology-independent
    w_entry = sd["per_param"][0]
    assert w_entry["shape"] == [24, 16], w_entry["shape"]
    assert w_entry["master"].numel() == 24 * 16

    # roundtrip: loading the canonical state back reproduces the local shard
    before = [b.flat_master.clone() for _, b in opt.buckets]
    torch.manual_seed(42)
    lin2 = Linear1D(16, 24, parallel="col")
    opt2 = FusedAdamW(lin2.parameters(), lr=1e-2)
    opt2.load_state_dict(sd)
    for (_, b1), bm in zip(opt2.buckets, before):
        assert torch.equal(b1.flat_master, bm)


def test_tp2_optimizer_state_canonical():
    run_dist(_tp2_optimizer_state_canonical_worker, world_size=2)
