"""Serving-side decode throughput: GPT-2 1.3B KV-cache generation.

Prints prefill time and steady-state decode tokens/s (whole batch) —
single GPU, bf16, random weights + synthetic prompt.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchdistpackage_amd.inference.generate import (_alloc_caches,
                                                     _gpt2_decode_forward)
from torchdistpackage_amd.models.gpt2 import GPT2Model, gpt2_xl_1p3b

B = int(os.environ.get("DEC_BATCH", "16"))
PROMPT = int(os.environ.get("DEC_PROMPT", "128"))
NEW = int(os.environ.get("DEC_NEW", "128"))
B1 = os.environ.get("DEC_B1", "1") == "1"   # also run the batch-1 latency case

dev = torch.device("cuda")
torch.manual_seed(0)
cfg = gpt2_xl_1p3b()
m = GPT2Model(cfg, device=dev, dtype=torch.bfloat16).eval()
idx = torch.randint(0, cfg.vocab_size, (B, PROMPT), device=dev)
hd = cfg.dim // cfg.n_head
caches = _alloc_caches(cfg.n_layer, B, cfg.n_head, PROMPT + NEW, hd, dev,
                       torch.bfloat16)

with torch.no_grad():
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    logits = _gpt2_decode_forward(m, idx, caches, 0)
    torch.cuda.synchronize()
    t_prefill = time.perf_counter() - t0

    nxt = logits.argmax(-1)[:, None]
    pos0 = PROMPT
    # warmup decode steps
    for _ in range(8):
        logits = _gpt2_decode_forward(m, nxt, caches, pos0)
        nxt = logits.argmax(-1)[:, None]
        pos0 += 1
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n_timed = NEW - 8
    for _ in range(n_timed):
        logits = _gpt2_decode_forward(m, nxt, caches, pos0)
        nxt = logits.argmax(-1)[:, None]
        pos0 += 1
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

print(f"prefill {B}x{PROMPT}: {t_prefill*1e3:.1f} ms "
      f"({B*PROMPT/t_prefill:.0f} tok/s)")
print(f"eager decode: {dt/n_timed*1e3:.2f} ms/step, "
      f"{B*n_timed/dt:.0f} tok/s (batch {B})")

from torchdistpackage_amd.inference.generate import GraphedGPT2Decoder

dec = GraphedGPT2Decoder(m, batch=B, max_seq=PROMPT + NEW)
dec.prefill(idx)
for _ in range(8):
    dec.step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(n_timed):
    dec.step()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"graphed decode: {dt/n_timed*1e3:.2f} ms/step, "
      f"{B*n_timed/dt:.0f} tok/s (batch {B})")

if B1:
    idx1 = idx[:1]
    dec1 = GraphedGPT2Decoder(m, batch=1, max_seq=PROMPT + NEW)
    dec1.prefill(idx1)
    for _ in range(8):
        dec1.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n_timed):
        dec1.step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"graphed decode B=1: {dt/n_timed*1e3:.2f} ms/step, "
          f"{n_timed/dt:.0f} tok/s (single stream)")
print("DECODE BENCH OK")
