"""Z-order covering index tests (reference:
index/zordercovering/E2EHyperspaceZOrderIndexTest, ZOrderFieldTest)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.config import IndexConstants
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.plan.nodes import IndexScan


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    data = tmp_path / "data"
    data.mkdir()
    rng = np.random.default_rng(61)
    n = 50_000
    t = pa.table({
        "x": rng.integers(0, 10_000, n),
        "y": rng.integers(0, 10_000, n),
        "val": rng.random(n),
    })
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    # small target file size so the z-sorted data splits into many files
    session.conf.set(
        IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION, 64 * 1024)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    return session, h, df, data, rng


def _rows(batch, cols):
    arrs = batch.to_numpy()
    return sorted(zip(*[arrs[c].tolist() for c in cols]))


def test_zorder_build_and_filter_any_indexed_col(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zix", ["x", "y"], ["val"]))
    entry = session.index_manager().get_index("zix")
    assert entry.derivedDataset.kind == "ZOrderCoveringIndex"
    assert len(entry.content.os_files()) > 4  # split into chunks

    session.enable_hyperspace()
    # filter on y (NOT the first indexed column) still uses the index —
    # the z-order rule admits any indexed column
    q = df.filter("y = 5000").select("y", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, IndexScan)]
    assert scans and scans[0].entry.name == "zix", plan.pretty()
    out = q.collect()
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["y", "val"]) == _rows(base, ["y", "val"])


def test_zorder_stats_pruning(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zix", ["x", "y"], ["val"]))
    session.enable_hyperspace()
    # a tight 2-d box query: z-ordering clusters it into few files
    q = df.filter("x <= 300 AND y <= 300").select("x", "y", "val")
    ex = Executor(session)
    out = ex.execute(q.optimized_plan())
    assert ex.stats.bucket_pruned_files > 0
    entry = session.index_manager().get_index("zix")
    n_files = len(entry.content.os_files())
    assert ex.stats.scanned_files < n_files
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["x", "y", "val"]) == _rows(base, ["x", "y", "val"])


def test_zorder_clustering_effective(env):
    """Z-ordering must cluster 2-d boxes better than linear order: the
    box query should scan well under half the files."""
    session, h, df, _, _ = env
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zix", ["x", "y"], ["val"]))
    entry = session.index_manager().get_index("zix")
    n_files = len(entry.content.os_files())
    session.enable_hyperspace()
    q = df.filter("x <= 600 AND y <= 600").select("x", "y")
    ex = Executor(session)
    ex.execute(q.optimized_plan())
    assert ex.stats.scanned_files <= max(2, n_files // 2), (
        ex.stats.scanned_files, n_files)


def test_covering_score_beats_zorder_for_first_col(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zix", ["x", "y"], ["val"]))
    h.create_index(df, hs.CoveringIndexConfig("cix", ["x"], ["val"]))
    session.enable_hyperspace()
    # z-order (60) outranks plain covering filter (50)
    q = df.filter("x = 77").select("x", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, IndexScan)]
    assert scans[0].entry.name == "zix"


def test_zorder_json_roundtrip(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zr", ["x", "y"], ["val"]))
    entry = session.index_manager().get_index("zr")
    from hyperspace_amd.log.entry import IndexLogEntry
    back = IndexLogEntry.from_json(entry.to_json())
    assert back.derivedDataset.indexed_columns == ["x", "y"]
    assert back.derivedDataset.kind == "ZOrderCoveringIndex"


def test_zorder_quantile_mode_on_skewed_data(tmp_path, monkeypatch):
    """Quantile scaling spreads a heavily skewed column across z-cells:
    a hot-range box query scans fewer files than min/max scaling."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(71)
    data = tmp_path / "d"
    data.mkdir()
    n = 50_000
    # x: 99% of mass in [0,100], tail to 10^9 (minmax scaling squashes
    # the hot range into one cell)
    x = np.where(rng.random(n) < 0.99,
                 rng.integers(0, 100, n),
                 rng.integers(0, 10**9, n))
    t = pa.table({"x": x, "y": rng.integers(0, 100, n),
                  "val": rng.random(n)})
    pq.write_table(t, str(data / "part-0.parquet"))

    def scanned(quantile):
        session = hs.HyperspaceSession(device="cpu")
        session.conf.set(
            IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION,
            64 * 1024)
        session.conf.set(IndexConstants.ZORDER_QUANTILE_ENABLED, quantile)
        h = hs.Hyperspace(session)
        df = session.read_parquet(str(data))
        name = f"zq_{int(quantile)}"
        h.create_index(df, hs.ZOrderCoveringIndexConfig(
            name, ["x", "y"], ["val"]))
        session.enable_hyperspace()
        q = df.filter("x <= 50 AND y <= 50").select("x", "y", "val")
        ex = Executor(session)
        out = ex.execute(q.optimized_plan())
        session.disable_hyperspace()
        assert out.num_rows == q.collect().num_rows
        return ex.stats.scanned_files

    assert scanned(True) <= scanned(False)


def test_zorder_with_string_column(tmp_path, monkeypatch):
    """String z-order columns: dictionary codes are order-preserving, so
    the normalize-then-interleave path covers them (reference
    ZOrderField string encoding); file pruning stays numeric-only."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(13)
    d = tmp_path / "zs"
    d.mkdir()
    cats = np.array(["aa", "bb", "cc", "dd"])[rng.integers(0, 4, 8000)]
    key = rng.integers(0, 1000, 8000)
    pq.write_table(pa.table({"cat": cats.tolist(), "key": key,
                             "val": rng.random(8000)}),
                   str(d / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zs", ["cat", "key"], ["val"]))
    session.enable_hyperspace()
    out = df.filter("cat = 'bb'").select("cat", "key", "val").collect()
    assert out.num_rows == int((cats == "bb").sum())
    out2 = df.filter("key >= 900").select("key", "val").collect()
    assert out2.num_rows == int((key >= 900).sum())
