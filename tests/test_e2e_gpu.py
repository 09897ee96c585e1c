"""GPU end-to-end: covering index build + indexed filter + co-bucketed
join on device, validated against the CPU engine on the same data."""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.ops import native
from hyperspace_amd.plan.nodes import IndexScan

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require(tmp_path_factory):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert native.available()


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(21)
    data = tmp_path / "data"
    data.mkdir()
    n = 500_000
    for i in range(2):
        t = pa.table({
            "key": rng.integers(0, 50_000, n),
            "val": rng.random(n),
        })
        pq.write_table(t, str(data / f"part-{i}.parquet"))
    rdir = tmp_path / "right"
    rdir.mkdir()
    t = pa.table({"key": np.arange(50_000, dtype=np.int64),
                  "status": rng.integers(0, 5, 50_000)})
    pq.write_table(t, str(rdir / "part-0.parquet"))
    return data, rdir


def _rows(batch, cols):
    arrs = batch.to_numpy()
    return sorted(zip(*[arrs[c].tolist() for c in cols]))


def test_gpu_build_and_filter(env, tmp_path):
    data, _ = env
    gpu = hs.HyperspaceSession(device="cuda")
    gpu.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 64)
    h = hs.Hyperspace(gpu)
    df = gpu.read_parquet(str(data))
    h.create_index(df, hs.CoveringIndexConfig("gix", ["key"], ["val"]))

    gpu.enable_hyperspace()
    gpu.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC, True)
    q = df.filter("key = 4242").select("key", "val")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    ex = Executor(gpu)
    out = ex.execute(plan)
    assert ex.stats.bucket_pruned_files > 0
    gpu.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["key", "val"]) == _rows(base, ["key", "val"])


def test_gpu_cobucketed_join(env, tmp_path):
    data, rdir = env
    gpu = hs.HyperspaceSession(device="cuda")
    gpu.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 64)
    h = hs.Hyperspace(gpu)
    left = gpu.read_parquet(str(data))
    right = gpu.read_parquet(str(rdir))
    h.create_index(left, hs.CoveringIndexConfig("glix", ["key"], ["val"]))
    h.create_index(right, hs.CoveringIndexConfig("grix", ["key"],
                                                 ["status"]))
    gpu.enable_hyperspace()
    q = left.select("key", "val").join(right.select("key", "status"),
                                       on="key")
    plan = q.optimized_plan()
    ex = Executor(gpu)
    out = ex.execute(plan)
    assert ex.stats.merge_joins == 1
    assert ex.stats.shuffles == 0
    gpu.disable_hyperspace()
    base = q.collect()
    # full (key, val, status) multiset equality, not just row count
    assert _rows(out, ["key", "val", "status"]) == \
        _rows(base, ["key", "val", "status"])


def test_gpu_string_native_decode_and_join(tmp_path, monkeypatch):
    """Device string path: dictionary BYTE_ARRAY files
    decode on device (codes stay in HBM), string-keyed index build +
    co-bucketed join return content equal to the CPU engine."""
    from hyperspace_amd.execution.columnar import (ColumnBatch,
                                                   StringColumn)
    from hyperspace_amd.sources.parquet_io import (
        read_files_batch_device, write_batch_parquet)
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "sidx"))
    rng = np.random.default_rng(33)
    src = tmp_path / "ssrc"
    src.mkdir()
    vocab = sorted(f"sku-{i:05d}" for i in range(5000))
    paths = []
    for i in range(3):
        codes = rng.integers(0, 5000, 300_000).astype(np.int32)
        mask = rng.random(300_000) > 0.05
        b = ColumnBatch(
            {"sku": StringColumn(torch.from_numpy(codes), list(vocab)),
             "v": torch.from_numpy(rng.random(300_000))},
            masks={"sku": torch.from_numpy(mask)})
        p = str(src / f"part-{i}.parquet")
        write_batch_parquet(b, p)
        paths.append(p)

    # 1. device decode == host decode, codes resident on device
    from hyperspace_amd.sources.native_parquet import _LAYOUT_CACHE
    _LAYOUT_CACHE.clear()  # force the footer/page parse path
    dev_batch, rc_dev = read_files_batch_device(
        paths, torch.device("cuda"))
    from hyperspace_amd.sources.parquet_io import read_files_batch
    host_batch, rc_host = read_files_batch(paths)
    assert rc_dev == rc_host
    s_dev = dev_batch.column("sku")
    assert isinstance(s_dev, StringColumn)
    assert s_dev.codes.device.type == "cuda"
    s_host = host_batch.column("sku")
    assert s_dev.values == s_host.values
    assert torch.equal(s_dev.codes.cpu(), s_host.codes)
    assert torch.equal(dev_batch.mask("sku").cpu(), host_batch.mask("sku"))

    # 2. string-keyed index + join: content equality vs CPU baseline
    rdir = tmp_path / "sdim"
    rdir.mkdir()
    write_batch_parquet(ColumnBatch(
        {"sku": StringColumn(
            torch.arange(5000, dtype=torch.int32), list(vocab)),
         "status": torch.from_numpy(rng.integers(0, 5, 5000))}),
        str(rdir / "part-0.parquet"))
    gpu = hs.HyperspaceSession(device="cuda")
    gpu.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 32)
    h = hs.Hyperspace(gpu)
    left = gpu.read_parquet(str(src))
    right = gpu.read_parquet(str(rdir))
    h.create_index(left, hs.CoveringIndexConfig("gsl", ["sku"], ["v"]))
    h.create_index(right, hs.CoveringIndexConfig("gsr", ["sku"],
                                                 ["status"]))
    gpu.enable_hyperspace()
    q = left.select("sku", "v").join(right.select("sku", "status"),
                                     on="sku")
    ex = Executor(gpu)
    out = ex.execute(q.optimized_plan())
    assert ex.stats.merge_joins == 1 and ex.stats.shuffles == 0
    gpu.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["sku", "v", "status"]) == \
        _rows(base, ["sku", "v", "status"])


def test_gpu_sorted_within_buckets(env, tmp_path):
    # verify the on-disk contract: each bucket file is sorted by key and
    # all rows hash to that bucket
    data, _ = env
    gpu = hs.HyperspaceSession(device="cuda")
    gpu.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(gpu)
    df = gpu.read_parquet(str(data))
    h.create_index(df, hs.CoveringIndexConfig("six", ["key"], ["val"]))
    entry = gpu.index_manager().get_index("six")
    from hyperspace_amd.sources.parquet_io import bucket_id_of_file
    from hyperspace_amd.ops import cpu_ref
    for f in entry.content.os_files():
        b = bucket_id_of_file(f)
        t = pq.read_table(f)
        keys = torch.from_numpy(t.column("key").to_numpy())
        assert torch.all(keys[1:] >= keys[:-1]), f
        assert (cpu_ref.murmur3_bucket([keys], 8) == b).all(), f


def test_gpu_hybrid_scan(env, tmp_path):
    """Hybrid scan on device: appended delta merged via Union, deleted
    file rows excluded via the lineage kernel."""
    import pyarrow as pa
    data, _ = env
    gpu = hs.HyperspaceSession(device="cuda")
    gpu.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 16)
    gpu.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(gpu)
    df = gpu.read_parquet(str(data))
    h.create_index(df, hs.CoveringIndexConfig("hgx", ["key"], ["val"]))
    # append a small delta
    rng = np.random.default_rng(77)
    t = pa.table({"key": rng.integers(0, 50_000, 20_000),
                  "val": rng.random(20_000)})
    pq.write_table(t, str(data / "part-delta.parquet"))
    gpu.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    gpu.enable_hyperspace()
    q = df.filter("key = 4242").select("key", "val")
    plan = q.optimized_plan()
    from hyperspace_amd.plan.nodes import UnionNode, BucketUnionNode
    found = []

    def walk(n):
        if isinstance(n, (UnionNode, BucketUnionNode)):
            found.append(n)
        for c in n.children:
            walk(c)
    walk(plan)
    assert found, plan.pretty()
    out = q.collect()
    gpu.disable_hyperspace()
    base = q.collect()
    assert out.num_rows == base.num_rows


def test_gpu_index_build_from_spark_shaped_source(tmp_path, monkeypatch):
    """Full create_index + filter + join over a Spark-default-shaped
    source: SNAPPY + dictionary encoding, string dimension column,
    nullable measure — the device read path (dict_z/splain_z/host-codec
    pages) feeds the build end to end, results equal the disabled-path
    scan."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH",
                       str(tmp_path / "indexes"))
    rng = np.random.default_rng(71)
    data = tmp_path / "src"
    data.mkdir()
    n = 300_000
    skus = sorted(f"sku-{i:05d}" for i in range(3000))
    for i in range(4):
        t = pa.table({
            "key": rng.integers(0, 40_000, n),
            "sku": [skus[j] for j in rng.integers(0, 3000, n)],
            "price": pa.array(
                [None if j % 17 == 0 else float(j % 997)
                 for j in range(n)], type=pa.float64()),
        })
        pq.write_table(t, str(data / f"part-{i}.parquet"),
                       compression="SNAPPY", use_dictionary=True,
                       data_page_version="1.0")
    gpu = hs.HyperspaceSession(device="cuda")
    gpu.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 16)
    h = hs.Hyperspace(gpu)
    df = gpu.read_parquet(str(data))
    h.create_index(df, hs.CoveringIndexConfig(
        "zsrc", ["key"], ["sku", "price"]))
    gpu.enable_hyperspace()
    q = df.filter("key = 777").select("key", "sku", "price")
    out = q.collect()
    plan = q.optimized_plan().pretty()
    assert "IndexScan(zsrc" in plan, plan
    gpu.disable_hyperspace()
    base = q.collect()
    assert out.num_rows == base.num_rows
    # content equality including the string column and null mask
    def key_of(b):
        s = b.column("sku")
        m = b.mask("price")
        pv = b.tensor("price").cpu().numpy()
        mv = m.cpu().numpy() if m is not None else np.ones(b.num_rows,
                                                          bool)
        return sorted((s.values[c], float(p) if ok else None)
                      for c, p, ok in zip(s.codes.cpu().numpy(), pv, mv))
    assert key_of(out) == key_of(base)
