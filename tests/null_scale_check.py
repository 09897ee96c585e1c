import os, sys, time
sys.path.insert(0, "/root/repo")
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
import hyperspace_amd as hs

work = "/dev/shm/nullscale"
os.system(f"rm -rf {work}"); os.makedirs(work + "/data")
os.environ["HYPERSPACE_SYSTEM_PATH"] = work + "/idx"
rng = np.random.default_rng(1)
N = 134_000_000  # ~2 GiB of (key,val)
CH = N // 8
t0 = time.time()
null_total = 0
eq_total = 0
for i in range(8):
    key = rng.integers(0, 100_000, CH)
    mask = rng.random(CH) > 0.07
    key[::173] = 777
    val = rng.random(CH)
    null_total += int((~mask).sum())
    eq_total += int((mask & (key == 777)).sum())
    pq.write_table(pa.table({"key": pa.array(key, mask=~mask),
                             "val": val}),
                   f"{work}/data/part-{i}.parquet", compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
print(f"gen {N/1e6:.0f}M rows ({null_total/1e6:.1f}M nulls) in {time.time()-t0:.1f}s")

session = hs.HyperspaceSession(device="cuda:0")
session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 64)
h = hs.Hyperspace(session)
df = session.read_parquet(work + "/data")
t0 = time.time()
h.create_index(df, hs.CoveringIndexConfig("nsx", ["key"], ["val"]))
torch.cuda.synchronize()
print(f"nullable build: {time.time()-t0:.2f}s")
session.enable_hyperspace()
from hyperspace_amd.plan.expr import col
for q, want in [("eq", eq_total), ("isnull", null_total),
                ("notnull", N - null_total)]:
    t0 = time.time()
    if q == "eq":
        out = df.filter("key = 777").select("key", "val").collect()
    elif q == "isnull":
        out = df.filter(col("key").is_null()).collect()
    else:
        out = df.filter(col("key").is_not_null()).collect()
    torch.cuda.synchronize()
    ok = out.num_rows == want
    print(f"{q}: rows={out.num_rows} want={want} {'OK' if ok else 'MISMATCH'} ({time.time()-t0:.2f}s)")
    assert ok
print("nullable-at-volume OK")
