"""FusedAdamW CPU-path semantics tests (the GPU multi-tensor kernel is
covered by test_ops_gpu.py; these check stock-torch-optimizer parity that
both paths must share)."""

import torch
import torch.nn as nn

from torchdistpackage_amd.ops.optim import FusedAdamW


def test_grad_none_params_untouched():
    """params whose .grad is None must be SKIPPED — no weight
    decay, no moment update (stock torch.optim semantics)."""
    torch.manual_seed(0)
    p1 = nn.Parameter(torch.randn(16))
    p2 = nn.Parameter(torch.randn(16))
    before = p2.detach().clone()
    opt = FusedAdamW([p1, p2], lr=0.1, weight_decay=0.5)
    for _ in range(3):
        p1.grad = torch.randn(16)
        opt.step()          # p2.grad stays None
        opt.zero_grad()
    assert torch.equal(p2.detach(), before), "grad-None param was modified"
    assert not torch.allclose(p1.detach(),
                              torch.zeros_like(p1)), "p1 should have moved"
    # moments of the skipped param must remain zero
    fg = opt.param_groups[0]["_flats"][0]
    i2 = next(i for i, q in enumerate(fg.params) if q is p2)
    o, n = fg.offs[i2], p2.numel()
    assert torch.all(fg.exp_avg[o:o + n] == 0)
    assert torch.all(fg.exp_avg_sq[o:o + n] == 0)


def test_matches_torch_adamw():
    torch.manual_seed(1)
    model_a = nn.Sequential(nn.Linear(8, 8), nn.Tanh(), nn.Linear(8, 4))
    import copy
    model_b = copy.deepcopy(model_a)
    oa = FusedAdamW(model_a.parameters(), lr=1e-2, betas=(0.9, 0.999),
                    eps=1e-8, weight_decay=0.01)
    ob = torch.optim.AdamW(model_b.parameters(), lr=1e-2,
                           betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01)
    for it in range(5):
        x = torch.randn(4, 8)
        model_a(x).pow(2).mean().backward()
        model_b(x).pow(2).mean().backward()
        oa.step()
        ob.step()
        oa.zero_grad()
        ob.zero_grad()
        for pa, pb in zip(model_a.parameters(), model_b.parameters()):
            assert torch.allclose(pa, pb, atol=1e-6), it
