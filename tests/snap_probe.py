import os, sys, time
sys.path.insert(0, "/root/repo")
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
from hyperspace_amd.sources import parquet_io
from hyperspace_amd.sources.native_parquet import read_native_layout

d = "/dev/shm/snapb2"; os.system(f"rm -rf {d}"); os.makedirs(d)
rng = np.random.default_rng(1)
N = 16_750_000
pq.write_table(pa.table({"key": rng.integers(0, 5000, N),
                         "val": rng.random(N)}),
               f"{d}/p0.parquet", compression="SNAPPY",
               use_dictionary=True, data_page_version="1.0")
p = f"{d}/p0.parquet"
t0 = time.time()
lay = read_native_layout(p)
print(f"layout: {time.time()-t0:.2f}s ->", "OK" if lay else "None")
if lay:
    for c in lay[1]:
        kinds = {}
        for pg in c.pages: kinds[pg[0]] = kinds.get(pg[0], 0) + 1
        print(f"  {c.name}: enc={c.encoding} pages={kinds}")
orig = parquet_io.read_files_batch
def spy(*a, **k):
    print("!! HOST FALLBACK"); return orig(*a, **k)
parquet_io.read_files_batch = spy
t0 = time.time()
b, _ = parquet_io.read_files_batch_device([p], torch.device("cuda:0"))
torch.cuda.synchronize()
print(f"device read 1 file: {time.time()-t0:.2f}s")
