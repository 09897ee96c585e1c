"""Debug harness for the multi-tile radix/select path (GPU box only)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

import hyperspace_amd._hip as ext
from hyperspace_amd.ops import cpu_ref

def check(n, key_hi, seed):
    rng = np.random.default_rng(seed)
    keys = cpu_ref.normalize_key(
        torch.from_numpy(rng.integers(0, key_hi, n, dtype=np.int64)))
    payload = torch.arange(n, dtype=torch.int64)
    ck, cp = cpu_ref.stable_sort_u64(keys, payload)
    gk, gp = ext.radix_sort_pairs(keys.cuda(), payload.cuda())
    gk, gp = gk.cpu(), gp.cpu()
    ok_k = torch.equal(ck, gk)
    ok_p = torch.equal(cp, gp)
    print(f"n={n} key_hi={key_hi}: keys_ok={ok_k} payload_ok={ok_p}")
    if not ok_k or not ok_p:
        # where does it first diverge?
        dk = (ck != gk).nonzero().flatten()
        dp = (cp != gp).nonzero().flatten()
        print("  first key mismatch:", dk[:5].tolist(), "of", dk.numel())
        print("  first payload mismatch:", dp[:5].tolist(), "of", dp.numel())
        # is the gpu output a permutation at all?
        print("  gpu payload unique:", gp.unique().numel(), "of", n)
        print("  gpu keys sorted:",
              bool((gk[1:] >= gk[:-1]).all().item()) if n > 1 else True)
        i = int(dk[0]) if dk.numel() else int(dp[0])
        print("  around first mismatch: cpu",
              ck[max(0,i-2):i+3].tolist(), cp[max(0,i-2):i+3].tolist())
        print("                         gpu",
              gk[max(0,i-2):i+3].tolist(), gp[max(0,i-2):i+3].tolist())
    return ok_k and ok_p

if __name__ == "__main__":
    # single-tile-per-block regime (tiles < 1024)
    check(100_000, 50, 1)
    check(262_143, 50, 2)   # just under 1024 tiles
    check(262_145, 50, 3)   # just over: blocks 0 gets 2 tiles
    check(300_000, 50, 4)
    check(500_000, 50, 5)
    check(500_000, 2**62, 6)
    # select_range
    vals = torch.from_numpy(
        np.random.default_rng(8).integers(-1000, 1000, 300_000,
                                          dtype=np.int64))
    keys = cpu_ref.normalize_key(vals)
    lo = int(cpu_ref.normalize_key(torch.tensor([-500]))[0])
    hi = int(cpu_ref.normalize_key(torch.tensor([500]))[0])
    cpu = cpu_ref.select_range_u64(keys, lo, hi, True, True)
    gpu = ext.select_range_u64(keys.cuda(), lo, hi, True, True).cpu()
    print("select 300k:", torch.equal(cpu, gpu), cpu.numel(), gpu.numel())
