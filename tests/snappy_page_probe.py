"""Per-page snappy kernel timing for the refresh-shaped file: which
page costs 140 ms and what does its op stream look like?"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
from hyperspace_amd.ops import native
from hyperspace_amd.sources.native_parquet import read_native_layout

d = "/dev/shm/pp"; os.system(f"rm -rf {d}"); os.makedirs(d)
rng = np.random.default_rng(9)
n = 2_000_000  # 2 row groups worth
pq.write_table(pa.table({"key": rng.integers(0, 1 << 25, n),
                         "val": rng.random(n)}), f"{d}/a.parquet")
p = f"{d}/a.parquet"
data, chunks = read_native_layout(p)
ext = native.ext()
dev = torch.device("cuda:0")
raw = torch.from_numpy(np.frombuffer(data, dtype=np.uint8).copy()).to(dev)

def snappy_host_opstats(src):
    # host parse of the op stream: count ops / literals / copies
    s = 0; n_lit = n_cp = lit_bytes = cp_bytes = 0
    # skip varint
    while src[s] & 0x80: s += 1
    s += 1
    while s < len(src):
        tag = src[s]; k = tag & 3
        if k == 0:
            l = tag >> 2
            if l < 60: ln = l + 1; s += 1
            else:
                nb = l - 59
                ln = int.from_bytes(src[s+1:s+1+nb], "little") + 1
                s += 1 + nb
            s += ln; n_lit += 1; lit_bytes += ln
        elif k == 1:
            ln = ((tag >> 2) & 7) + 4; s += 2; n_cp += 1; cp_bytes += ln
        elif k == 2:
            ln = (tag >> 2) + 1; s += 3; n_cp += 1; cp_bytes += ln
        else:
            ln = (tag >> 2) + 1; s += 5; n_cp += 1; cp_bytes += ln
    return n_lit, lit_bytes, n_cp, cp_bytes

for c in chunks[:1]:  # key chunk of rg0
    segs = []
    if c.encoding == "dict_z" and len(c.dict_page) == 5:
        _, a, b, dn, unc = c.dict_page
        segs.append(("dict", a, b, unc))
    for pg in c.pages[:6]:
        segs.append((pg[0], pg[1], pg[2], pg[4]))
    for name, a, b, unc in segs:
        comp = b - a
        scratch = torch.empty(unc + 4, dtype=torch.uint8, device=dev)
        st = ext.snappy_decompress(
            raw, torch.tensor([a]), torch.tensor([b]), scratch,
            torch.tensor([0]), torch.tensor([unc]))
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(3):
            ext.snappy_decompress(
                raw, torch.tensor([a]), torch.tensor([b]), scratch,
                torch.tensor([0]), torch.tensor([unc]))
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 3
        nl, lb, nc, cb = snappy_host_opstats(bytes(data[a:b]))
        print(f"{name}: comp={comp} unc={unc} {dt*1000:.2f} ms "
              f"lit={nl}({lb}B) copies={nc}({cb}B) status={int(st[0])}")
