import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.moe_model import MoEModel, mixtral_style_8x
from torchdistpackage_amd import register_profile_hooks, report_prof
dev = torch.device("cuda")
torch.manual_seed(0)
m = MoEModel(mixtral_style_8x(), device=dev, dtype=torch.bfloat16)
x = torch.randint(0, 50304, (16, 1024), device=dev)
with torch.no_grad():
    m(x)
register_profile_hooks(m, use_roctx=False)
with torch.no_grad():
    m(x)
rows = report_prof(top=12, sort_by="time")
