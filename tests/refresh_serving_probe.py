"""Post-refresh serving probe: build 8 GiB index, append ~5% source,
refresh incrementally, then time the cold + warm indexed join (the
multi-file buckets exercise the K4b run merge at scan)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, pyarrow as pa, pyarrow.parquet as pq, torch
import hyperspace_amd as hs
from hyperspace_amd import bench_utils
from hyperspace_amd.execution.executor import Executor

work = "/dev/shm/refprobe"
os.system(f"rm -rf {work}"); os.makedirs(work)
os.environ["HYPERSPACE_SYSTEM_PATH"] = work + "/idx"
bench_utils.generate_fact_parquet(work + "/fact", 8 << 30, seed=0,
                                  key_hi=1 << 25)
bench_utils.generate_dim_parquet(work + "/dim", n_rows=5_000_000, seed=0)
session = hs.HyperspaceSession(device="cuda:0")
session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 200)
session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
h = hs.Hyperspace(session)
fact = session.read_parquet(work + "/fact")
dim = session.read_parquet(work + "/dim")
h.create_index(dim, hs.CoveringIndexConfig("d", ["key"], ["status"]))
t0 = time.time()
h.create_index(fact, hs.CoveringIndexConfig("f", ["key"], ["val"]))
torch.cuda.synchronize(); print(f"build: {time.time()-t0:.2f}s")
# append ~5%
rng = np.random.default_rng(9)
n = 27_000_000
pq.write_table(pa.table({"key": rng.integers(0, 1 << 25, n),
                         "val": rng.random(n)}),
               work + "/fact/part-append.parquet")
import cProfile, pstats, io as _io
pr = cProfile.Profile()
t0 = time.time()
pr.enable()
h.refresh_index("f", mode="incremental")
pr.disable()
torch.cuda.synchronize(); print(f"refresh incremental (+5%): {time.time()-t0:.2f}s")
sbuf = _io.StringIO()
pstats.Stats(pr, stream=sbuf).sort_stats("cumulative").print_stats(30)
print("\n".join(sbuf.getvalue().splitlines()[4:40]))
session.enable_hyperspace()
q = fact.select("key", "val").join(dim.select("key", "status"), on="key")
plan = q.optimized_plan()
for label in ("cold", "warm", "warm2"):
    t0 = time.time()
    ex = Executor(session)
    out = ex.execute(plan); torch.cuda.synchronize()
    print(f"join {label}: {time.time()-t0:.3f}s rows={out.num_rows/1e6:.1f}M "
          f"shuffles={ex.stats.shuffles}")
