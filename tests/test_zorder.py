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


# ---------------------------------------------------------------------------
# Per-type encodings: decimal / date / timestamp (reference
# ZOrderField.scala:26-569 covers int/long/double/decimal/string/
# date/timestamp, plus percentile buckets for skew)
# ---------------------------------------------------------------------------

def test_zorder_decimal_and_timestamp(tmp_path, monkeypatch):
    """Decimal(12,2) and timestamp columns z-order and filter exactly:
    decimals ingest as unscaled int64 with literal scaling at bind."""
    import decimal
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(77)
    n = 20_000
    cents = rng.integers(0, 1_000_000, n)  # 0.00 .. 9999.99
    prices = [decimal.Decimal(int(c)) / 100 for c in cents]
    ts = rng.integers(1_600_000_000, 1_700_000_000, n) * 1_000_000
    t = pa.table({
        "price": pa.array(prices, type=pa.decimal128(12, 2)),
        "ts": pa.array(ts, type=pa.timestamp("us")),
        "val": rng.random(n),
    })
    data = tmp_path / "d"
    data.mkdir()
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(
        IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION,
        64 * 1024)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    # schema carries the decimal type
    from hyperspace_amd.plan.nodes import Scan
    scan = df.plan.collect_leaves()[0]
    assert scan.relation.schema.field_type("price") == "decimal(12,2)"

    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zdec", ["price", "ts"], ["val"]))
    session.enable_hyperspace()
    q = df.filter("price <= 100.50").select("price", "val")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = Executor(session).execute(plan)
    session.disable_hyperspace()
    expected = int((cents <= 10050).sum())
    assert out.num_rows == expected
    assert q.collect().num_rows == expected
    # decimal equality with an exact literal
    session.enable_hyperspace()
    target = int(cents[0])
    eq = df.filter(f"price = {decimal.Decimal(target) / 100}")
    assert eq.collect().num_rows == int((cents == target).sum())


def test_zorder_date_column(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(78)
    n = 10_000
    days = rng.integers(18_000, 19_000, n)  # ~2019-2021
    t = pa.table({
        "d": pa.array(days.astype("datetime64[D]")),
        "v": rng.random(n),
    })
    data = tmp_path / "dd"
    data.mkdir()
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(
        IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION,
        32 * 1024)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    h.create_index(df, hs.ZOrderCoveringIndexConfig("zd", ["d"], ["v"]))
    session.enable_hyperspace()
    out = df.filter("d >= 18500").select("d", "v").collect()
    session.disable_hyperspace()
    assert out.num_rows == int((days >= 18500).sum())


def test_zorder_decimal_quantile_skew(tmp_path, monkeypatch):
    """Percentile-bucketed z-cells on a heavily skewed decimal column
    (reference ZOrderField percentile variant): quantile mode must
    cluster better than linear min/max scaling."""
    import decimal
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(79)
    n = 40_000
    # 99% of mass near zero, tail to 10^8 cents
    cents = (rng.pareto(1.5, n) * 1000).astype(np.int64).clip(0, 10**8)
    prices = [decimal.Decimal(int(c)) / 100 for c in cents]
    other = rng.integers(0, 1000, n)
    t = pa.table({
        "price": pa.array(prices, type=pa.decimal128(14, 2)),
        "o": other, "v": rng.random(n)})
    data = tmp_path / "sk"
    data.mkdir()
    pq.write_table(t, str(data / "part-0.parquet"))

    def files_scanned(quantile):
        root = tmp_path / f"q{quantile}"
        monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(root))
        session = hs.HyperspaceSession(device="cpu")
        session.conf.set(
            IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION,
            64 * 1024)
        session.conf.set(IndexConstants.ZORDER_QUANTILE_ENABLED,
                         quantile)
        h = hs.Hyperspace(session)
        df = session.read_parquet(str(data))
        h.create_index(df, hs.ZOrderCoveringIndexConfig(
            "zq", ["price", "o"], ["v"]))
        session.enable_hyperspace()
        q = df.filter("price <= 5.00").select("price", "v")
        ex = Executor(session)
        out = ex.execute(q.optimized_plan())
        assert out.num_rows == int((cents <= 500).sum())
        return ex.stats.scanned_files

    scanned_lin = files_scanned(False)
    scanned_q = files_scanned(True)
    # on this skew the linear scaling collapses most rows into one
    # z-cell; quantile buckets spread them so the stats pruning bites
    assert scanned_q <= scanned_lin, (scanned_q, scanned_lin)
