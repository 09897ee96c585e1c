"""Hybrid Scan tests (the reference's HybridScanSuite behaviors):
appended files merged at query time via Union/BucketUnion, deleted files
excluded via the lineage column filter."""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.plan.nodes import (BucketUnionNode, IndexScan,
                                       UnionNode)

N = 10000


def _write(rng, path, n=N, key_hi=1000):
    t = pa.table({
        "key": rng.integers(0, key_hi, n),
        "val": rng.random(n),
    })
    pq.write_table(t, str(path))


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(11)
    data = tmp_path / "data"
    data.mkdir()
    for i in range(6):
        _write(rng, data / f"part-{i}.parquet")
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    return session, h, df, data, rng


def _rows(batch, cols):
    arrs = batch.to_numpy()
    return sorted(zip(*[arrs[c].tolist() for c in cols]))


def test_hybrid_scan_appended_filter(env):
    session, h, df, data, rng = env
    h.create_index(df, hs.CoveringIndexConfig("hix", ["key"], ["val"]))
    # append a small delta (< 0.3 appended ratio)
    _write(rng, data / "part-new.parquet", n=500)
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    session.enable_hyperspace()
    q = df.filter("key = 77").select("key", "val")
    plan = q.optimized_plan()
    # rewrite = Union(IndexScan, appended scan)
    has_union = False

    def walk(n):
        nonlocal has_union
        if isinstance(n, (UnionNode, BucketUnionNode)):
            has_union = True
        for c in n.children:
            walk(c)
    walk(plan)
    assert has_union, plan.pretty()

    accel = q.collect()
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(accel, ["key", "val"]) == _rows(base, ["key", "val"])


def test_hybrid_scan_deleted_filter(env):
    session, h, df, data, rng = env
    h.create_index(df, hs.CoveringIndexConfig("hix", ["key"], ["val"]))
    # one of six files deleted: ratio 1/6 < the 0.2 threshold
    os.unlink(str(data / "part-5.parquet"))
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    session.enable_hyperspace()
    q = df.filter("key = 77").select("key", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, IndexScan)]
    assert scans and scans[0].excluded_source_file_ids, plan.pretty()
    accel = q.collect()
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(accel, ["key", "val"]) == _rows(base, ["key", "val"])


def test_hybrid_scan_join_bucket_union(env, tmp_path):
    session, h, df, data, rng = env
    rdir = tmp_path / "right"
    rdir.mkdir()
    t = pa.table({"key": np.arange(1000, dtype=np.int64),
                  "status": rng.integers(0, 5, 1000)})
    pq.write_table(t, str(rdir / "part-0.parquet"))
    right = session.read_parquet(str(rdir))

    h.create_index(df, hs.CoveringIndexConfig("lix", ["key"], ["val"]))
    h.create_index(right, hs.CoveringIndexConfig("rix", ["key"], ["status"]))
    _write(rng, data / "part-new.parquet", n=500)
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    session.enable_hyperspace()

    q = df.select("key", "val").join(right.select("key", "status"),
                                     on="key")
    plan = q.optimized_plan()
    has_bucket_union = False

    def walk(n):
        nonlocal has_bucket_union
        if isinstance(n, BucketUnionNode):
            has_bucket_union = True
        for c in n.children:
            walk(c)
    walk(plan)
    assert has_bucket_union, plan.pretty()

    ex = Executor(session)
    accel = ex.execute(plan)
    assert ex.stats.merge_joins == 1
    session.disable_hyperspace()
    base = q.collect()
    assert accel.num_rows == base.num_rows
    assert _rows(accel, ["key", "val", "status"]) == \
        _rows(base, ["key", "val", "status"])


def test_too_much_appended_blocks_hybrid(env):
    session, h, df, data, rng = env
    h.create_index(df, hs.CoveringIndexConfig("hix", ["key"], ["val"]))
    # append 4 more files = 40% appended ratio > 0.3
    for i in range(4):
        _write(rng, data / f"part-big-{i}.parquet")
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    session.enable_hyperspace()
    plan = df.filter("key = 77").select("key", "val").optimized_plan()
    assert not any(isinstance(l, IndexScan)
                   for l in plan.collect_leaves())


# ---------------------------------------------------------------------------
# Hybrid Scan over transactional table formats (reference:
# HybridScanForDeltaLakeTest / HybridScanForIcebergTest — append and
# delete through the TABLE LOG, then assert plan shape and results)
# ---------------------------------------------------------------------------

def _tbatch(rng, k_hi=1000, n=N):
    import torch
    return __import__("hyperspace_amd").execution.columnar.ColumnBatch({
        "key": torch.from_numpy(rng.integers(0, k_hi, n)),
        "val": torch.from_numpy(rng.random(n))})


def test_hybrid_scan_delta_append_and_delete(tmp_path, monkeypatch):
    from hyperspace_amd.execution.columnar import ColumnBatch
    import torch
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    from hyperspace_amd.sources.delta_source import DeltaTable
    rng = np.random.default_rng(41)
    t = DeltaTable.create(str(tmp_path / "t"))
    # 8 base files so deleting one stays under the 0.2 deleted-ratio
    # ceiling (reference maxDeletedRatio semantics)
    for _ in range(8):
        t.append_batch(ColumnBatch({
            "key": torch.from_numpy(rng.integers(0, 1000, N)),
            "val": torch.from_numpy(rng.random(N))}))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(session)
    df = session.read_delta(str(tmp_path / "t"))
    h.create_index(df, hs.CoveringIndexConfig("hdx", ["key"], ["val"]))

    # append one more file + delete one original THROUGH THE DELTA LOG
    t.append_batch(ColumnBatch({
        "key": torch.from_numpy(rng.integers(0, 1000, 1000)),
        "val": torch.from_numpy(rng.random(1000))}))
    victim = t.files_at()[0].name
    t.remove_files([victim])

    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    live = session.read_delta(str(tmp_path / "t"))
    q = live.filter("key = 77").select("key", "val")
    plan = q.optimized_plan()
    # hybrid shape: index scan + appended-file scan under a Union
    assert any(isinstance(n, UnionNode)
               for n in _walk(plan)), plan.pretty()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = q.collect()
    session.disable_hyperspace()
    assert _rows(out, ["key", "val"]) == _rows(q.collect(),
                                               ["key", "val"])


def test_hybrid_scan_iceberg_append(tmp_path, monkeypatch):
    from hyperspace_amd.execution.columnar import ColumnBatch
    import torch
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    from hyperspace_amd.sources.iceberg_source import IcebergTable
    rng = np.random.default_rng(43)
    t = IcebergTable.create(str(tmp_path / "it"))
    for _ in range(4):
        t.append_batch(ColumnBatch({
            "key": torch.from_numpy(rng.integers(0, 1000, N)),
            "val": torch.from_numpy(rng.random(N))}))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(session)
    df = session.read_iceberg(str(tmp_path / "it"))
    h.create_index(df, hs.CoveringIndexConfig("hix", ["key"], ["val"]))

    t.append_batch(ColumnBatch({
        "key": torch.from_numpy(rng.integers(0, 1000, 1000)),
        "val": torch.from_numpy(rng.random(1000))}))
    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    live = session.read_iceberg(str(tmp_path / "it"))
    q = live.filter("key = 77").select("key", "val")
    plan = q.optimized_plan()
    assert any(isinstance(n, UnionNode)
               for n in _walk(plan)), plan.pretty()
    out = q.collect()
    session.disable_hyperspace()
    assert _rows(out, ["key", "val"]) == _rows(q.collect(),
                                               ["key", "val"])


def _walk(plan):
    out = [plan]
    for c in getattr(plan, "children", []):
        out.extend(_walk(c))
    return out


def test_hybrid_scan_partitioned_source(tmp_path, monkeypatch):
    """Hybrid scan over hive-partitioned data (reference
    HybridScanForPartitionedDataTest): the appended file lives in a
    partition directory; the appended-scan keeps partition columns."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(53)
    root = tmp_path / "pt"
    for day in (1, 2):
        d = root / f"day={day}"
        d.mkdir(parents=True)
        for i in range(3):
            _write(rng, d / f"part-{i}.parquet", n=4000)
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(root))
    h.create_index(df, hs.CoveringIndexConfig(
        "hpx", ["key"], ["val", "day"]))

    # append INTO a partition after the build
    _write(rng, root / "day=2" / "part-new.parquet", n=1500)
    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    live = session.read_parquet(str(root))
    q = live.filter("key = 77").select("key", "val", "day")
    plan = q.optimized_plan()
    assert any(isinstance(n, UnionNode) for n in _walk(plan)), \
        plan.pretty()
    out = q.collect()
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["key", "val", "day"]) == \
        _rows(base, ["key", "val", "day"])
