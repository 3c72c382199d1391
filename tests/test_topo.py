"""Topology registry tests (CPU, gloo).

Mirrors the reference's tpc self-test (process_topo.py:267-316) plus unit
coverage of the axis-enumeration math without any process group.
"""

import pytest

from torchdistpackage_amd.dist.topo import gen_axis_groups
from tests.dist_helpers import run_distributed


def test_gen_axis_groups_inner():
    # innermost axis (stride 1): adjacent ranks
    assert gen_axis_groups(8, 2, 1) == [[0, 1], [2, 3], [4, 5], [6, 7]]


def test_gen_axis_groups_outer():
    # outermost axis of [('data',2),('pipe',2),('tensor',2)]: stride 4
    assert gen_axis_groups(8, 2, 4) == [[0, 4], [1, 5], [2, 6], [3, 7]]


def test_gen_axis_groups_middle():
    assert gen_axis_groups(8, 2, 2) == [[0, 2], [1, 3], [4, 6], [5, 7]]


def test_gen_axis_groups_bad():
    with pytest.raises(ValueError):
        gen_axis_groups(8, 3, 1)


def _topo_check(rank, world_size):
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc, is_using_pp, test_comm

    tpc.setup_process_groups([("data", 2), ("tensor", 2)])
    assert tpc.get_dp_size() == 2
    assert tpc.get_tp_size() == 2
    assert tpc.get_pp_size() == 1
    assert not is_using_pp()
    # ranks 0,1 share a tensor group (innermost); 0,2 share data
    tp_ranks = tpc.get_ranks_in_group("tensor")
    dp_ranks = tpc.get_ranks_in_group("data")
    if rank in (0, 1):
        assert tp_ranks == [0, 1]
    else:
        assert tp_ranks == [2, 3]
    assert dp_ranks in ([0, 2], [1, 3])
    # derived model axis = transpose of data groups
    assert tpc.is_mode_inited("model")
    assert tpc.get_mp_size() == 2
    test_comm()
    # moe groups on top of the dp axis
    tpc.build_moe_groups(moe_dp_size=1, moe_ep_size=2)
    assert tpc.get_group_size("moe_ep") == 2
    assert tpc.get_group_size("moe_dp") == 1
    return (rank, tp_ranks, dp_ranks)


def test_topology_world4():
    run_distributed(_topo_check, world_size=4)


def _ring_check(rank, world_size):
    from torchdistpackage_amd.dist.topo import tpc

    tpc.setup_process_groups([("pipe", world_size)])
    nxt = tpc.get_next_global_rank("pipe")
    prv = tpc.get_prev_global_rank("pipe")
    assert nxt == (rank + 1) % world_size
    assert prv == (rank - 1) % world_size
    assert tpc.is_first_in_pipeline_group() == (rank == 0)
    assert tpc.is_last_in_pipeline_group() == (rank == world_size - 1)
    return rank


def test_pipe_ring_world2():
    run_distributed(_ring_check, world_size=2)


def _first_last_helpers(rank, world_size):
    from torchdistpackage_amd.dist.topo import tpc
    tpc.setup_process_groups([("data", 2), ("tensor", 2)])
    # first group of 'tensor' = ranks [0, 1]; of 'data' = [0, 2]
    assert tpc.is_first_group("tensor") == (rank in (0, 1))
    assert tpc.is_first_group("data") == (rank in (0, 2))
    assert tpc.is_first_in_group("tensor") == (rank % 2 == 0)
    assert tpc.is_last_in_group("tensor") == (rank % 2 == 1)
    assert tpc.is_last_in_data_group() == (rank >= 2)
    assert tpc.is_last_in_tensor_group() == (rank % 2 == 1)
    assert tpc.all_ranks() == [0, 1, 2, 3]
    assert tpc.all_ranks("tensor") == [[0, 1], [2, 3]]
    assert tpc.all_dp_ranks() == [[0, 2], [1, 3]]
    # model = transpose of data
    assert tpc.is_first_in_model_group() == (tpc.get_mp_rank() == 0)
    return True


def test_first_last_helpers():
    run_distributed(_first_last_helpers, world_size=4)
