"""Cold-load phase breakdown: where does read_files_batch_device time go
at bench shape (8 GiB, 200 native files)?  Phases measured separately:
disk->pinned readinto, +H2D upload, full decode pipeline (cold + pool-
warm)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from hyperspace_amd import bench_utils
from hyperspace_amd.sources import parquet_io
from hyperspace_amd.sources.parquet_io import (_pinned_get, _pinned_put,
                                               read_files_batch_device)

GB = 8
work = "/dev/shm/coldprof"
os.system(f"rm -rf {work}")
os.makedirs(work)
t0 = time.time()
bench_utils.generate_fact_parquet(work, GB << 30, seed=0, key_hi=1 << 22)
paths = sorted(os.path.join(work, f) for f in os.listdir(work)
               if f.endswith(".parquet"))
total = sum(os.path.getsize(p) for p in paths)
print(f"gen {len(paths)} files {total/2**30:.1f}G in {time.time()-t0:.1f}s")

# phase A: readinto only (16 threads, pooled pinned)
from concurrent.futures import ThreadPoolExecutor


def read_one(p):
    size = os.path.getsize(p)
    buf = _pinned_get(size + 4)
    view = memoryview(buf.numpy())
    with open(p, "rb", buffering=0) as f:
        f.readinto(view[:size])
    return buf


for trial in range(2):
    t0 = time.time()
    with ThreadPoolExecutor(max_workers=16) as pool:
        bufs = list(pool.map(read_one, paths))
    dt = time.time() - t0
    print(f"A readinto (pool {'cold' if trial==0 else 'warm'}): "
          f"{dt:.3f}s = {total/2**30/dt:.1f} GB/s")
    for b in bufs:
        _pinned_put(b)

# phase B: readinto + H2D (no decode)
dev = torch.device("cuda:0")
torch.cuda.init()
streams = [torch.cuda.Stream(device=dev) for _ in range(8)]


def read_up(i):
    p = paths[i]
    size = os.path.getsize(p)
    buf = _pinned_get(size + 4)
    view = memoryview(buf.numpy())
    with open(p, "rb", buffering=0) as f:
        f.readinto(view[:size])
    with torch.cuda.stream(streams[i % 8]):
        d = buf[:size].to(dev, non_blocking=True)
    return buf, d


t0 = time.time()
with ThreadPoolExecutor(max_workers=16) as pool:
    res = list(pool.map(read_up, range(len(paths))))
cur = torch.cuda.current_stream()
for s in streams:
    cur.wait_stream(s)
cur.synchronize()
dt = time.time() - t0
print(f"B readinto+H2D: {dt:.3f}s = {total/2**30/dt:.1f} GB/s")
for b, d in res:
    _pinned_put(b)
del res
torch.cuda.empty_cache()

# phase C: full device decode (cold pool entries already warm)
for trial in range(2):
    t0 = time.time()
    batch, counts = read_files_batch_device(paths, dev)
    dt = time.time() - t0
    print(f"C full decode trial {trial}: {dt:.3f}s = "
          f"{total/2**30/dt:.1f} GB/s rows={batch.num_rows/1e6:.0f}M")
    del batch
    torch.cuda.empty_cache()
