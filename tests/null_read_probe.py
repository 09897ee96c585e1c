import os, sys, time
sys.path.insert(0, "/root/repo")
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
work = "/dev/shm/nullprobe"
os.system(f"rm -rf {work}"); os.makedirs(work)
rng = np.random.default_rng(1)
CH = 16_750_000
for i in range(4):
    key = rng.integers(0, 100_000, CH)
    mask = rng.random(CH) > 0.07
    pq.write_table(pa.table({"key": pa.array(key, mask=~mask),
                             "val": rng.random(CH)}),
                   f"{work}/part-{i}.parquet", compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
paths = sorted(f"{work}/{f}" for f in os.listdir(work))
from hyperspace_amd.sources import parquet_io
from hyperspace_amd.sources.native_parquet import read_native_layout, _decode_defs
# probe fallback: monkeypatch read_files_batch to detect host-path use
orig = parquet_io.read_files_batch
def spy(*a, **k):
    print("!! HOST FALLBACK TAKEN")
    return orig(*a, **k)
parquet_io.read_files_batch = spy
t0 = time.time()
lay = read_native_layout(paths[0])
print(f"layout parse 1 file: {time.time()-t0:.2f}s -> {'ok' if lay else 'None'}")
t0 = time.time()
b, rc = parquet_io.read_files_batch_device(paths, torch.device("cuda:0"))
torch.cuda.synchronize()
print(f"device read 4 files: {time.time()-t0:.2f}s rows={b.num_rows/1e6:.0f}M mask={'yes' if b.mask('key') is not None else 'no'}")
