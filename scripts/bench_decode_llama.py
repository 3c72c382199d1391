"""Llama-8B serving decode: graphed single-token step at B=1 and B=8."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchdistpackage_amd.inference.generate import GraphedLlamaDecoder
from torchdistpackage_amd.models.llama import LlamaModel, llama3_8b

PROMPT, NEW = 128, 64
dev = torch.device("cuda")
torch.manual_seed(0)
m = LlamaModel(llama3_8b(), device=dev, dtype=torch.bfloat16).eval()

for B in (1, 8):
    idx = torch.randint(0, 128256, (B, PROMPT), device=dev)
    dec = GraphedLlamaDecoder(m, batch=B, max_seq=PROMPT + NEW)
    dec.prefill(idx)
    for _ in range(8):
        dec.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n = NEW - 8
    for _ in range(n):
        dec.step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"llama-8b graphed decode B={B}: {dt/n*1e3:.2f} ms/step, "
          f"{B*n/dt:.0f} tok/s")
print("LLAMA DECODE OK")
