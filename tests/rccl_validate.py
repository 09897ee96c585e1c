"""RCCL wiring validation: N ranks sharing the available GPU(s).

Run under torchrun on a GPU box:

  python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nproc-per-node 2 tests/rccl_validate.py

Each rank maps to device (local_rank % device_count) — on a 1-GPU box
both ranks share cuda:0, which RCCL supports — and exercises the REAL
multi-GPU code path end to end: packed device all-to-all (strings +
masks), chunked pipelined exchange, and a distributed covering-index
build + bucket-pruned filter query over RCCL collectives.

Prints RCCL_VALIDATE_OK on rank 0 when every assertion holds.
"""

import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))  # repo root (torchrun puts tests/ first)

import numpy as np
import torch
import torch.distributed as dist


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    assert torch.cuda.is_available(), "needs a GPU"
    dev_idx = local_rank % torch.cuda.device_count()
    torch.cuda.set_device(dev_idx)
    # RCCL refuses >1 rank per device; with more ranks than GPUs the
    # exchange stages via gloo/host while all compute stays on device
    backend = ("nccl" if world <= torch.cuda.device_count() else "gloo")
    dist.init_process_group(backend=backend)
    device = torch.device(f"cuda:{dev_idx}")
    if rank == 0:
        print(f"backend={backend} world={world} "
              f"gpus={torch.cuda.device_count()}", flush=True)

    from hyperspace_amd.execution.columnar import ColumnBatch, StringColumn
    from hyperspace_amd.parallel.exchange import BucketExchange
    from hyperspace_amd import ops

    # -- 1. packed device all-to-all with strings + masks ------------------
    n = 2_000_000
    rng = np.random.default_rng(7 + rank)
    vocab = [f"r{rank}-{i}" for i in range(64)]
    batch = ColumnBatch(
        {"key": torch.from_numpy(rng.integers(0, 1 << 30, n)).to(device),
         "val": torch.from_numpy(rng.random(n)).to(device),
         "tag": StringColumn(
             torch.from_numpy(rng.integers(0, 64, n).astype(np.int32))
             .to(device), sorted(vocab))},
        {"val": torch.from_numpy(rng.random(n) > 0.5).to(device)})
    bucket_ids = ops.murmur3_bucket([batch.tensor("key")], 200)
    ex = BucketExchange(200)
    out, ob = ex.finish(ex.start(batch, bucket_ids))
    assert ((ob.long() % world) == rank).all(), "bucket ownership"
    assert out.device.type == "cuda", "exchange left the device"
    # row_bytes = 8 + 8 + 4 + 1 + 4 = 25
    assert ex.bytes_sent == n * 25, ex.bytes_sent
    # global multiset: all-reduce row count + key checksum
    t = torch.tensor([out.num_rows, int(out.tensor("key").sum())],
                     device=device)
    t0 = torch.tensor([n, int(batch.tensor("key").sum())], device=device)
    dist.all_reduce(t)
    dist.all_reduce(t0)
    assert torch.equal(t.cpu(), t0.cpu()), (t, t0)
    assert len(out.column("tag").values) == world * 64, "merged dict"
    mask = out.mask("val")
    assert mask is not None and mask.device.type == "cuda"
    if rank == 0:
        print(f"a2a ok: {ex.bytes_sent/1e6:.0f} MB sent/rank, "
              f"{out.num_rows} rows received", flush=True)

    # -- 2. distributed covering-index build + query (chunked) -------------
    tmp = None
    if rank == 0:
        tmp = tempfile.mkdtemp(prefix="rcclval")
    obj = [tmp]
    dist.broadcast_object_list(obj, src=0)
    tmp = obj[0]
    os.environ["HYPERSPACE_SYSTEM_PATH"] = os.path.join(tmp, "indexes")
    src = os.path.join(tmp, "src")
    if rank == 0:
        import pyarrow as pa
        import pyarrow.parquet as pq
        os.makedirs(src, exist_ok=True)
        g = np.random.default_rng(3)
        for i in range(4):
            key = g.integers(0, 5000, 500_000)
            pq.write_table(
                pa.table({"key": key, "val": g.random(500_000)}),
                os.path.join(src, f"part-{i}.parquet"))
    dist.barrier()

    import hyperspace_amd as hs
    from hyperspace_amd.execution.executor import Executor
    session = hs.HyperspaceSession(device=str(device))
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 16)
    # force several pipelined chunks per group
    session.conf.set("spark.hyperspace.exchange.chunkBytes", 4 << 20)
    h = hs.Hyperspace(session)
    df = session.read_parquet(src)
    h.create_index(df, hs.CoveringIndexConfig("rv", ["key"], ["val"]))
    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     True)
    out = Executor(session).execute(
        df.filter("key = 777").select("key", "val").optimized_plan())
    t = torch.tensor([out.num_rows], device=device)
    dist.all_reduce(t)
    import pyarrow.parquet as pq
    expected = int((pq.read_table(src, columns=["key"])
                    .column("key").to_numpy() == 777).sum())
    assert int(t[0]) == expected, (int(t[0]), expected)
    dist.barrier()
    if rank == 0:
        print(f"distributed build ok: filter total {int(t[0])} == "
              f"{expected} expected", flush=True)
        print("RCCL_VALIDATE_OK", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(main())
