import os, sys, time, tempfile
sys.path.insert(0, "/root/repo")
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
from hyperspace_amd.sources.parquet_io import read_files_batch_device

d = "/dev/shm/snapb"; os.system(f"rm -rf {d}"); os.makedirs(d)
rng = np.random.default_rng(1)
N = 16_750_000
for i in range(8):
    pq.write_table(pa.table({"key": rng.integers(0, 5000, N),
                             "val": rng.random(N)}),
                   f"{d}/p{i}.parquet", compression="SNAPPY",
                   use_dictionary=True, data_page_version="1.0")
paths = sorted(f"{d}/{f}" for f in os.listdir(d))
comp = sum(os.path.getsize(p) for p in paths)
unc = 8 * N * 16
for t in range(2):
    t0 = time.time()
    b, _ = read_files_batch_device(paths, torch.device("cuda:0"))
    torch.cuda.synchronize()
    dt = time.time() - t0
    print(f"trial {t}: {dt:.3f}s  comp {comp/2**30:.2f}G -> unc "
          f"{unc/2**30:.2f}G = {unc/2**30/dt:.1f} GB/s decompressed")
    del b; torch.cuda.empty_cache()
