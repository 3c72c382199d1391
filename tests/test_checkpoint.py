"""Checkpoint save/resume round trip (single process + gloo world2)."""

import os
import tempfile

import torch
import torch.nn as nn

from tests.dist_helpers import run_distributed


def test_ckpt_roundtrip_single():
    from torchdistpackage_amd.dist.checkpoint import (save_checkpoint,
                                                      load_checkpoint,
                                                      latest_step)
    from torchdistpackage_amd.ops.optim import FusedAdamW

    with tempfile.TemporaryDirectory() as d:
        torch.manual_seed(0)
        model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 4))
        opt = FusedAdamW(model.parameters(), lr=1e-2)
        for it in range(3):
            model(torch.randn(4, 8)).sum().backward()
            opt.step()
            opt.zero_grad()
        save_checkpoint(d, 3, model, optimizer=opt, extra={"note": "hi"})
        assert latest_step(d) == 3

        torch.manual_seed(99)
        model2 = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 4))
        opt2 = FusedAdamW(model2.parameters(), lr=1e-2)
        payload = load_checkpoint(d, model2, optimizer=opt2)
        assert payload["extra"]["note"] == "hi"
        for p1, p2 in zip(model.parameters(), model2.parameters()):
            assert torch.equal(p1, p2)
        # continue training both; must stay identical
        x = torch.randn(4, 8)
        model(x).sum().backward()
        model2(x).sum().backward()
        opt.step()
        opt2.step()
        for p1, p2 in zip(model.parameters(), model2.parameters()):
            assert torch.allclose(p1, p2, atol=1e-7)


def _ckpt_dp2(rank, world_size, tmpdir):
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.dist.checkpoint import (save_checkpoint,
                                                      load_checkpoint)
    from torchdistpackage_amd.dist import ShardedEMA

    tpc.setup_process_groups([("data", world_size)])
    torch.manual_seed(0)
    model = nn.Linear(8, 8)
    ema = ShardedEMA(model, decay=0.9)
    ema.update()
    save_checkpoint(tmpdir, 1, model, ema=ema)
    # only one ckpt file (dp-replicated -> dp-rank-0 writes)
    files = sorted(os.listdir(tmpdir))
    assert "ckpt_step1.pth" in files and "latest" in files
    model2 = nn.Linear(8, 8)
    ema2 = ShardedEMA(model2, decay=0.9)
    load_checkpoint(tmpdir, model2, ema=ema2)
    assert torch.equal(model2.weight, model.weight)
    # EMA shard round-trips through the dense ema file
    for i in ema._my_idx:
        assert torch.allclose(ema2._views[i], ema._views[i], atol=1e-7)
    return True


def test_ckpt_dp2():
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_ckpt_dp2, world_size=2, args=(d,))


def _zero_ckpt_resume(rank, world_size, tmpdir):
    """Save mid-run with ZeRO (sharded optimizer state), reload into a fresh
    model+optimizer, and verify training continues identically to the
    uninterrupted run."""
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer
    from torchdistpackage_amd.dist.checkpoint import (save_checkpoint,
                                                      load_checkpoint)

    def make():
        torch.manual_seed(7)
        return nn.Sequential(nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 16))

    def batch(it):
        torch.manual_seed(1000 + 10 * it + rank)
        return torch.randn(4, 16)

    model = make()
    opt = Bf16ZeroOptimizer(torch.optim.Adam(model.parameters(), lr=1e-2))
    for it in range(3):
        model(batch(it)).pow(2).mean().backward()
        opt.step()
        opt.zero_grad()
    save_checkpoint(tmpdir, 3, model, optimizer=opt)
    # one model file + one optim shard file per dp rank
    files = sorted(os.listdir(tmpdir))
    assert "ckpt_step3.pth" in files
    for r in range(world_size):
        assert f"optim_step3_dp{r}.pth" in files, files

    # uninterrupted continuation
    for it in range(3, 6):
        model(batch(it)).pow(2).mean().backward()
        opt.step()
        opt.zero_grad()

    # resumed continuation from the checkpoint
    model2 = make()
    with torch.no_grad():  # perturb so a failed load is caught
        for p in model2.parameters():
            p.add_(1.0)
    opt2 = Bf16ZeroOptimizer(torch.optim.Adam(model2.parameters(), lr=1e-2))
    load_checkpoint(tmpdir, model2, optimizer=opt2)
    for it in range(3, 6):
        model2(batch(it)).pow(2).mean().backward()
        opt2.step()
        opt2.zero_grad()

    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  model2.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-7), \
            f"{n1} diverged after resume: {(p1 - p2).abs().max().item()}"
    return True


def test_zero_ckpt_resume_world2():
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_zero_ckpt_resume, world_size=2, args=(d,))


def test_zero_ckpt_resume_world1():
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_zero_ckpt_resume, world_size=1, args=(d,))
