"""Decode probe shaped exactly like the incremental-refresh appended
file (27M rows, key_hi 1<<25, pyarrow defaults: snappy + dict)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
from hyperspace_amd.sources import parquet_io

d = "/dev/shm/refdec"; os.system(f"rm -rf {d}"); os.makedirs(d)
rng = np.random.default_rng(9)
n = 27_000_000
pq.write_table(pa.table({"key": rng.integers(0, 1 << 25, n),
                         "val": rng.random(n)}),
               f"{d}/append.parquet")
p = f"{d}/append.parquet"
md = pq.ParquetFile(p).metadata
print("rgs:", md.num_row_groups, "col0 encodings:",
      md.row_group(0).column(0).encodings,
      "col1:", md.row_group(0).column(1).encodings)
for t in range(2):
    t0 = time.time()
    b, _ = parquet_io.read_files_batch_device([p], torch.device("cuda:0"))
    torch.cuda.synchronize()
    print(f"trial {t}: {time.time()-t0:.3f}s")
    k = b.tensor("key")
    del b; torch.cuda.empty_cache()
