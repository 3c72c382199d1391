import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.ops import ext

def run(S, causal=False):
    B, H, D = 1, 1, 128
    q = torch.zeros(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.zeros(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    # V[key][d] = key*0.01 + d  (distinct along both axes)
    v = (torch.arange(S).view(1, 1, S, 1) * 0.01 +
         torch.arange(D).view(1, 1, 1, D) * 1.0).bfloat16().cuda()
    o = torch.empty_like(q)
    _, lse = ext("x").attn_fwd(q, k, v, o, causal, 0.088)
    # uniform P -> o[q] = mean over allowed keys of V
    if causal:
        ref = torch.stack([v[0, 0, :i + 1].float().mean(0)
                           for i in range(S)])
    else:
        ref = v[0, 0].float().mean(0).expand(S, D)
    err = (o[0, 0].float() - ref)
    print(f"S={S} causal={causal} maxerr={err.abs().max().item():.4f}")
    if err.abs().max() > 0.05:
        bad = (err.abs() > 0.05)
        qs, ds = bad.nonzero(as_tuple=True)
        print("  bad q rows:", sorted(set(qs.tolist()))[:10], "...")
        print("  bad d cols:", sorted(set(ds.tolist()))[:20], "...")
        print("  sample o[0,0:8]:", o[0,0,0,:8].float().tolist())
        print("  sample ref[0,0:8]:", ref[0,:8].tolist())
        print("  sample o[0,64:72]:", o[0,0,0,64:72].float().tolist())
        print("  sample ref[0,64:72]:", ref[0,64:72].tolist())

run(64); run(128); run(256); run(256, causal=True)
