"""ShardedEMA vs dense EMA lockstep test (reference: examples/test_shard_ema.py)."""

import copy

import torch
import torch.nn as nn

from tests.dist_helpers import run_distributed
from torchdistpackage_amd.dist.sharded_ema import partition_by_numel


def test_partition_by_numel_balance():
    params = [torch.empty(n) for n in (100, 90, 10, 10, 5, 5)]
    parts = partition_by_numel(params, 2)
    loads = [sum(params[i].numel() for i in p) for p in parts]
    assert abs(loads[0] - loads[1]) <= 10
    assert sorted(i for p in parts for i in p) == list(range(6))


def _ema_check(rank, world_size):
    from torchdistpackage_amd.dist import ShardedEMA

    torch.manual_seed(42)
    model = nn.Sequential(nn.Linear(32, 64), nn.Linear(64, 16))
    dense = {n: p.detach().clone().float()
             for n, p in model.named_parameters()}
    ema = ShardedEMA(model, decay=0.9)

    for it in range(20):
        with torch.no_grad():
            torch.manual_seed(it)  # same "training" on all ranks
            for p in model.parameters():
                p.add_(torch.randn_like(p) * 0.01)
        ema.update()
        for n, p in model.named_parameters():
            dense[n].mul_(0.9).add_(p.float(), alpha=0.1)

    assert ema.verify_with_gt(dense, rtol=1e-5, atol=1e-6)
    full = ema.state_dict_cpu()
    if rank == 0:
        assert full is not None
        for n in dense:
            assert torch.allclose(full[n], dense[n], rtol=1e-5, atol=1e-6), n
    else:
        assert full is None
    return True


def test_sharded_ema_world2():
    run_distributed(_ema_check, world_size=2)


def test_sharded_ema_world1():
    run_distributed(_ema_check, world_size=1)
