"""Query-path phase timing (GPU box): warm filter + join breakdown."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import hyperspace_amd as hs
from hyperspace_amd import bench_utils, ops
from hyperspace_amd.execution.executor import Executor

GB = 1 << 30


def t_sync():
    torch.cuda.synchronize()
    return time.perf_counter()


def main():
    work = "/tmp/profile_q"
    os.environ["HYPERSPACE_SYSTEM_PATH"] = os.path.join(work, "indexes")
    total = int(float(os.environ.get("PROF_GB", 4.0)) * GB)
    bench_utils.generate_fact_parquet(os.path.join(work, "fact"), total,
                                      seed=0,
                                      key_hi=max(1000, total // 16 // 8))
    bench_utils.generate_dim_parquet(os.path.join(work, "dim"), 5_000_000)

    session = hs.HyperspaceSession(device="cuda")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 200)
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     True)
    h = hs.Hyperspace(session)
    fact = session.read_parquet(os.path.join(work, "fact"))
    dim = session.read_parquet(os.path.join(work, "dim"))
    t0 = time.perf_counter()
    h.create_index(fact, hs.CoveringIndexConfig("fix", ["key"], ["val"]))
    h.create_index(dim, hs.CoveringIndexConfig("dix", ["key"], ["status"]))
    t1 = t_sync()
    print(f"build both: {t1-t0:.2f}s")

    session.enable_hyperspace()
    jq = fact.select("key", "val").join(dim.select("key", "status"),
                                        on="key").optimized_plan()
    # warm the cache
    ex = Executor(session)
    out = ex.execute(jq)
    t2 = t_sync()
    print(f"join cold: {t2-t1:.2f}s rows={out.num_rows}")
    for i in range(3):
        ta = t_sync()
        ex = Executor(session)
        out = ex.execute(jq)
        tb = t_sync()
        print(f"join warm {i}: {tb-ta:.3f}s rows={out.num_rows}")

    # inner breakdown of the warm join
    from hyperspace_amd.plan.nodes import IndexScan, Project
    lplan, rplan = jq.children if hasattr(jq, 'children') else (None, None)
    ex = Executor(session)
    ta = t_sync()
    lbatch, lseg = ex._exec(jq.left)
    rbatch, rseg = ex._exec(jq.right)
    tb = t_sync()
    lk = ops.normalize_key(lbatch.tensor("key"))
    rk = ops.normalize_key(rbatch.tensor("key"))
    tc = t_sync()
    lidx, ridx = ops.merge_join(lk, rk, lseg, rseg)
    td = t_sync()
    lout = lbatch.gather(lidx)
    rout = rbatch.gather(ridx)
    te = t_sync()
    print(f"  load(cached): {tb-ta:.3f}s  normalize: {tc-tb:.3f}s  "
          f"merge_join: {td-tc:.3f}s ({lidx.numel()} pairs)  "
          f"gather: {te-td:.3f}s")

    fq = fact.filter("key = 4242").select("key", "val").optimized_plan()
    ex = Executor(session)
    ex.execute(fq)
    ta = t_sync()
    for _ in range(10):
        ex = Executor(session)
        out = ex.execute(fq)
    tb = t_sync()
    print(f"filter warm x10: {(tb-ta)*100:.1f}ms/query rows={out.num_rows}")


if __name__ == "__main__":
    main()
