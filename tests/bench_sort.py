"""Radix sort micro-benchmark (GPU box)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import hyperspace_amd.ops as ops
from hyperspace_amd.ops import cpu_ref

def bench(n, key_hi, label):
    rng = np.random.default_rng(1)
    keys = cpu_ref.normalize_key(torch.from_numpy(
        rng.integers(0, key_hi, n, dtype=np.int64))).cuda()
    payload = torch.arange(n, dtype=torch.int64, device="cuda")
    # correctness
    ck, cp = cpu_ref.stable_sort_u64(keys[:100000].cpu(), payload[:100000].cpu())
    gk, gp = ops.sort_pairs(keys[:100000], payload[:100000])
    assert torch.equal(ck, gk.cpu()) and torch.equal(cp, gp.cpu()), label
    for _ in range(2):
        ops.sort_pairs(keys, payload)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        ops.sort_pairs(keys, payload)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{label}: n={n} {dt*1000:.1f} ms = {n/dt/1e9:.2f} Grows/s "
          f"({n*16/dt/2**30:.0f} GiB/s of key+payload)")

if __name__ == "__main__":
    bench(268_000_000, 33_500_000, "bench-shaped (26-bit keys)")
    bench(268_000_000, 2**62, "full-range 64-bit keys")
    bench(10_000_000, 33_500_000, "10M")
