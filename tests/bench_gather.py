"""Gather micro-benchmark: random-permutation gather of 268M int64
rows (the build's bucket-gather shape), V4-ILP vs the 1-elem
grid-stride reference (HS_GATHER_V1=1)."""
import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from hyperspace_amd.ops import native

ext = native.ext()
n = 268_000_000
dev = torch.device("cuda:0")
vals = torch.arange(n, dtype=torch.int64, device=dev)
idx = torch.randperm(n, device=dev)
for _ in range(2):
    out = ext.gather_rows(vals, idx)
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 5
for _ in range(iters):
    out = ext.gather_rows(vals, idx)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
bw = n * (8 + 8 + 8) / dt / 1e9
print(f"gather {n/1e6:.0f}M rows: {dt*1000:.2f} ms = {n/dt/1e9:.2f} "
      f"Grows/s ({bw:.0f} GB/s moved)")
assert bool((out[:5] == idx[:5]).all())
