"""Hive-style partitioned sources: key=value directories become table
columns, partition-only predicates prune files on metadata, covering
indexes can cover partition columns, and the data-skipping config
auto-adds PartitionSketch (reference
default/DefaultFileBasedRelation.scala:75-89,
DataSkippingIndexConfig.scala:56-84)."""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.index.dataskipping.sketches import PartitionSketch
from hyperspace_amd.plan.expr import col
from hyperspace_amd.plan.nodes import IndexScan

N_PER = 4000


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(55)
    root = tmp_path / "ptab"
    frames = {}
    for day in (1, 2, 3):
        for region in ("eu", "us"):
            d = root / f"day={day}" / f"region={region}"
            d.mkdir(parents=True)
            key = rng.integers(0, 300, N_PER)
            val = rng.random(N_PER)
            pq.write_table(pa.table({"key": key, "val": val}),
                           str(d / "part-0.parquet"))
            frames[(day, region)] = (key, val)
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    return session, hs.Hyperspace(session), str(root), frames


def test_partition_schema_and_read(env):
    session, h, root, frames = env
    df = session.read_parquet(root)
    names = df.plan.collect_leaves()[0].relation.schema.field_names()
    assert "day" in names and "region" in names
    out = df.collect()
    assert out.num_rows == 6 * N_PER
    day = out.tensor("day")
    assert day.dtype.is_floating_point is False
    assert sorted(set(day.tolist())) == [1, 2, 3]
    # string partition column materializes as a dictionary column
    regions = set(out.column("region").to_numpy().tolist())
    assert regions == {"eu", "us"}


def test_partition_pruning(env):
    session, h, root, frames = env
    df = session.read_parquet(root)
    q = df.filter("day = 2").select("key", "val", "day")
    ex = Executor(session)
    out = ex.execute(q.optimized_plan())
    assert out.num_rows == 2 * N_PER
    assert ex.stats.scanned_files == 2  # 2 of 6 files read

    # string partition + data conjunct: prune on the partition part,
    # evaluate the rest on the rows
    q2 = (df.filter((col("region") == "eu") & (col("key") < 100))
          .select("key", "region"))
    ex2 = Executor(session)
    out2 = ex2.execute(q2.optimized_plan())
    assert ex2.stats.scanned_files == 3
    expected = sum(int((frames[(d, "eu")][0] < 100).sum())
                   for d in (1, 2, 3))
    assert out2.num_rows == expected


def test_partition_only_projection(env):
    session, h, root, frames = env
    df = session.read_parquet(root)
    out = df.select("day", "region").collect()
    assert out.num_rows == 6 * N_PER


def test_covering_index_covers_partition_column(env):
    session, h, root, frames = env
    df = session.read_parquet(root)
    h.create_index(df, hs.CoveringIndexConfig(
        "pix", ["key"], ["val", "day"]))
    session.enable_hyperspace()
    q = df.filter("key = 7").select("key", "val", "day")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = q.collect()
    expected = sum(int((k == 7).sum()) for k, _ in frames.values())
    assert out.num_rows == expected
    # day values correct per partition
    day_counts = {}
    for (d, r), (k, _) in frames.items():
        day_counts[d] = day_counts.get(d, 0) + int((k == 7).sum())
    got = out.tensor("day").tolist()
    for d, c in day_counts.items():
        assert got.count(d) == c


def test_auto_partition_sketch(env):
    session, h, root, frames = env
    df = session.read_parquet(root)
    h.create_index(df, hs.DataSkippingIndexConfig(
        "psk", hs.MinMaxSketch("key")))
    entry = session.index_manager().get_index("psk")
    kinds = [type(s).__name__ for s in entry.derivedDataset.sketches]
    assert "PartitionSketch" in kinds  # auto-added for "day"
    session.enable_hyperspace()
    # disjunction stays convertible thanks to the partition sketch
    q = df.filter((col("key") == 99999) | (col("day") == 1))
    plan = q.optimized_plan()
    leaf = plan.collect_leaves()[0]
    assert leaf.file_subset is not None
    assert len(leaf.file_subset) == 2  # only day=1 files can match


def test_covering_auto_includes_partition_columns(env):
    """Reference CreateActionBase: partition columns missing from the
    config join the covering slice, so partition projections stay
    covered."""
    session, h, root, frames = env
    df = session.read_parquet(root)
    h.create_index(df, hs.CoveringIndexConfig("pauto", ["key"], ["val"]))
    entry = session.index_manager().get_index("pauto")
    inc = [c.lower() for c in entry.derivedDataset.included_columns]
    assert "day" in inc and "region" in inc
    session.enable_hyperspace()
    q = df.filter("key = 3").select("key", "val", "day", "region")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = q.collect()
    expected = sum(int((k == 3).sum()) for k, _ in frames.values())
    assert out.num_rows == expected
