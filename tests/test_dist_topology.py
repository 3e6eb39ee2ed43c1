"""Device-mesh topology unit tests (rank layout + layer->stage auto-balance)."""

import pytest

from libai_amd.utils.distributed import _DistributeUtil


def _mk(world, tp=1, pp=1, nlayers=None, custom=None, rank=0):
    du = _DistributeUtil.__new__(_DistributeUtil)
    du._world_size = world
    du._rank = rank
    du._local_rank = rank
    du._tp_size, du._pp_size = tp, pp
    du._dp_size = world // (tp * pp)
    dp, tpn = du._dp_size, tp
    du._pp_rank = rank // (dp * tpn)
    du._dp_rank = (rank % (dp * tpn)) // tpn
    du._tp_rank = rank % tpn
    du._pipeline_num_layers = nlayers
    du._custom_stage_id = custom
    du._tp_group = du._dp_group = du._pp_group = du._dp_tp_group = None
    return du


def test_rank_coordinates_stage_major():
    # world 8 = pp2 x dp2 x tp2; rank = pp*4 + dp*2 + tp
    du = _mk(8, tp=2, pp=2, rank=5)  # 5 = 1*4 + 0*2 + 1
    assert (du.pipeline_parallel_rank, du.data_parallel_rank,
            du.tensor_parallel_rank) == (1, 0, 1)
    assert du.prev_pipeline_rank() == 1
    du0 = _mk(8, tp=2, pp=2, rank=1)
    assert du0.next_pipeline_rank() == 5


def test_layer_stage_auto_balance_even():
    du = _mk(4, pp=4, nlayers=8)
    assert [du.layer_stage_id(i) for i in range(8)] == [0, 0, 1, 1, 2, 2, 3, 3]
    assert du.layer_stage_id(-1) == 3


def test_layer_stage_remainder_to_later_stages():
    # 10 layers over 4 stages: remainder 2 goes to the LAST stages
    du = _mk(4, pp=4, nlayers=10)
    stages = [du.layer_stage_id(i) for i in range(10)]
    assert stages == [0, 0, 1, 1, 2, 2, 2, 3, 3, 3]
    for s in range(4):
        lo, hi = du.stage_layer_range(s)
        assert [du.layer_stage_id(i) for i in range(lo, hi)] == [s] * (hi - lo)


def test_custom_stage_map():
    du = _mk(2, pp=2, nlayers=4, custom=[0, 0, 0, 1])
    assert [du.layer_stage_id(i) for i in range(4)] == [0, 0, 0, 1]
    assert du.stage_layer_range(1) == (3, 4)


def test_world_size_validation():
    import os

    os.environ.pop("WORLD_SIZE", None)
    with pytest.raises(ValueError):
        _DistributeUtil({"tensor_parallel_size": 2})
