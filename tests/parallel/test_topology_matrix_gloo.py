"""End-to-end topology matrix on gloo CPU — the reference's model-test grid
(reference tests/models/test_gpt.py:122-199: eager dp2xtp2, pp4, dp4xtp2,
dp2xpp2+grad-acc, ZeRO composition) at 4 and 8 ranks, plus the full 3D
dp2xtp2xpp2 compose.  Smoke criterion matches the reference: N iterations
complete, losses finite, DP replicas stay bit-consistent.
"""

import pytest
import torch

from tests.dist_helper import run_dist

MODEL_KW = dict(
    hidden_layers=4,
    vocab_size=128,
    hidden_size=32,
    ffn_hidden_size=128,
    num_attention_heads=4,
    max_seq_length=32,
    embedding_dropout_prob=0.0,
    attention_dropout_prob=0.0,
    output_dropout_prob=0.0,
)
STEPS = 3


def _matrix_worker(rank, world, tp, pp, zero, acc, moe=0):
    import torch
    import torch.distributed as dist

    from libai_amd.models import GPTForPreTraining
    from libai_amd.optim import FusedAdamW
    from libai_amd.parallel.pipeline import PipelineScheduler
    from libai_amd.utils import distributed as du

    du.setup_dist_util({
        "tensor_parallel_size": tp,
        "pipeline_parallel_size": pp,
        "pipeline_num_layers": MODEL_KW["hidden_layers"],
    })
    dutil = du.get_dist_util()
    assert dutil.data_parallel_size == world // (tp * pp)

    torch.manual_seed(123)  # same init everywhere; TP shards slice it
    model = GPTForPreTraining(**MODEL_KW, moe_num_experts=moe)
    model.hidden_size = MODEL_KW["hidden_size"]
    sched = PipelineScheduler(model, dtype=torch.float32) if pp > 1 else None
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.01,
                     clip_grad=1.0, zero_stage=zero if zero < 3 else 0)
    if zero == 3:
        from libai_amd.parallel.zero import setup_zero3

        setup_zero3(model, opt)

    torch.manual_seed(500 + dutil.data_parallel_rank)  # data differs per DP
    losses = []
    for _ in range(STEPS):
        opt.zero_grad()
        batches = []
        for _ in range(acc):
            ids = torch.randint(0, 128, (2, 33))
            batches.append({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        if sched is not None:
            loss_dict = sched.run_1f1b(batches)
            if loss_dict:
                losses.append(float(loss_dict["lm_loss"]))
        else:
            for b in batches:
                out = model(**b)
                (out["lm_loss"] / acc).backward()
            losses.append(float(out["lm_loss"]))
        opt.grad_sync()
        opt.step()

    assert all(l == l and l < 100 for l in losses), f"bad losses {losses}"

    # DP replicas must hold identical NON-expert parameters after the steps
    # (expert-parallel params are intentionally different per EP(=DP) rank)
    if dutil.data_parallel_size > 1:
        if zero == 3:  # ZeRO-3 params are released at rest; gather them back
            opt.materialize_all_params()
        flats = torch.cat([
            p.detach().reshape(-1) for p in model.parameters()
            if not getattr(p, "expert_parallel", False)
        ])
        gathered = [torch.empty_like(flats)
                    for _ in range(dutil.data_parallel_size)]
        dist.all_gather(gathered, flats, group=dutil.data_parallel_group)
        for g in gathered[1:]:
            assert torch.allclose(gathered[0], g, atol=1e-6), "DP divergence"
    return losses


# (world, tp, pp, zero, acc) — mirrors the reference grid + ZeRO/3D composes
MATRIX_4 = [
    pytest.param(4, 2, 1, 0, 1, id="dp2xtp2"),
    pytest.param(4, 1, 4, 0, 4, id="pp4_acc4"),
    pytest.param(4, 1, 2, 0, 2, id="dp2xpp2_acc2"),
    pytest.param(4, 1, 2, 1, 2, id="dp2xpp2_zero1"),
    pytest.param(4, 1, 1, 2, 2, id="dp4_zero2_acc2"),
    pytest.param(4, 1, 1, 3, 1, id="dp4_zero3"),
]
MATRIX_8 = [
    pytest.param(8, 2, 1, 0, 1, id="dp4xtp2"),
    pytest.param(8, 2, 2, 0, 2, id="dp2xtp2xpp2_acc2"),
    pytest.param(8, 1, 4, 1, 4, id="dp2xpp4_zero1_acc4"),
    pytest.param(8, 4, 1, 2, 1, id="dp2xtp4_zero2"),
]


@pytest.mark.timeout(600)
@pytest.mark.parametrize("world,tp,pp,zero,acc", MATRIX_4)
def test_topology_matrix_4rank(world, tp, pp, zero, acc):
    run_dist(_matrix_worker, world, args=(tp, pp, zero, acc))


@pytest.mark.timeout(600)
def test_topology_moe_dp2xpp2():
    """Expert parallelism composed with PP: EP == the stage-local DP
    group; experts dispatch within each stage, dense grads DP-sync."""
    run_dist(_matrix_worker, 4, args=(1, 2, 0, 2, 4))


@pytest.mark.timeout(900)
@pytest.mark.parametrize("world,tp,pp,zero,acc", MATRIX_8)
def test_topology_matrix_8rank(world, tp, pp, zero, acc):
    run_dist(_matrix_worker, world, args=(tp, pp, zero, acc))
