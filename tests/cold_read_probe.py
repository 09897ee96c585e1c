"""Breakdown of the cold index read (the 313 ms join_query_cold term).

Builds a bench-shaped covering index (2 GB source, 200 buckets), then
times read_files_batch_device over the written index files in three
modes: full path, read-only (file -> pinned), and H2D-only, to locate
the bound.  Not a pytest test: run under gpurun.
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import hyperspace_amd as hs
from hyperspace_amd.sources import parquet_io as pio

GB = 2
root = "/dev/shm/coldprobe"
os.makedirs(root + "/data", exist_ok=True)
os.environ["HYPERSPACE_SYSTEM_PATH"] = root + "/idx"

rows = (GB << 30) // 16
import pyarrow as pa
import pyarrow.parquet as pq
rng = np.random.default_rng(7)
per = rows // 8
for i in range(8):
    pq.write_table(pa.table({
        "key": rng.integers(0, 10_000_000, per),
        "val": rng.random(per)}),
        f"{root}/data/part-{i}.parquet",
        compression="NONE", use_dictionary=False,
        data_page_version="1.0")

session = hs.HyperspaceSession(device="cuda")
session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 200)
h = hs.Hyperspace(session)
df = session.read_parquet(root + "/data")
t0 = time.time()
h.create_index(df, hs.CoveringIndexConfig("cp", ["key"], ["val"]))
torch.cuda.synchronize()
print(f"build: {time.time()-t0:.3f}s")

entry = session.index_manager().get_index("cp")
files = [f for f in entry.content.os_files() if f.endswith(".parquet")]
total = sum(os.path.getsize(f) for f in files)
print(f"{len(files)} index files, {total/2**30:.2f} GiB")

# drop page cache? can't (no root sysctl guarantee) — /dev/shm IS the
# page cache, matching the bench's cold read exactly.

dev = torch.device("cuda")

# 1) full path, first touch (cold: layout cache present in this proc)
torch.cuda.synchronize(); t0 = time.time()
batch, rc = pio.read_files_batch_device(files, dev)
torch.cuda.synchronize(); t1 = time.time()
print(f"full path cold: {t1-t0:.3f}s  ({total/2**30/(t1-t0):.1f} GB/s)")

# 2) again (pinned pool warm)
t0 = time.time()
batch, rc = pio.read_files_batch_device(files, dev)
torch.cuda.synchronize(); t1 = time.time()
print(f"full path warm: {t1-t0:.3f}s  ({total/2**30/(t1-t0):.1f} GB/s)")

# 3) read-only: file -> pinned, 16 threads (no GPU work)
from concurrent.futures import ThreadPoolExecutor
bufs = {}
def rd(p):
    sz = os.path.getsize(p)
    b = pio._pinned_get(sz + 4)
    with open(p, "rb", buffering=0) as f:
        f.readinto(memoryview(b.numpy())[:sz])
    bufs[p] = (b, sz)
t0 = time.time()
with ThreadPoolExecutor(max_workers=16) as pool:
    list(pool.map(rd, files))
t1 = time.time()
print(f"read->pinned only: {t1-t0:.3f}s  ({total/2**30/(t1-t0):.1f} GB/s)")

# 4) H2D only from those pinned buffers, 8 streams
streams = [torch.cuda.Stream() for _ in range(8)]
t0 = time.time()
for i, p in enumerate(files):
    b, sz = bufs[p]
    with torch.cuda.stream(streams[i % 8]):
        b[:sz].to(dev, non_blocking=True)
torch.cuda.synchronize()
t1 = time.time()
print(f"H2D only: {t1-t0:.3f}s  ({total/2**30/(t1-t0):.1f} GB/s)")

# 5) single big pinned H2D for reference peak
big = pio._pinned_get(1 << 30)
t0 = time.time()
big.to(dev, non_blocking=True)
torch.cuda.synchronize()
t1 = time.time()
print(f"1GiB single H2D: {1/(t1-t0):.1f} GB/s")
