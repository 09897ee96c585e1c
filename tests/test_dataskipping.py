"""DataSkipping index tests (reference: index/dataskipping/* suites)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.exceptions import HyperspaceException
from hyperspace_amd.plan.nodes import Scan


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    data = tmp_path / "data"
    data.mkdir()
    rng = np.random.default_rng(51)
    # files with DISJOINT key ranges so min/max skipping bites
    for i in range(8):
        t = pa.table({
            "key": rng.integers(i * 1000, (i + 1) * 1000, 5000),
            "cat": rng.integers(0, 10, 5000),
            "val": rng.random(5000),
        })
        pq.write_table(t, str(data / f"part-{i}.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    return session, h, df, data, rng


def _rows(batch, cols):
    arrs = batch.to_numpy()
    return sorted(zip(*[arrs[c].tolist() for c in cols]))


def test_config_validation():
    with pytest.raises(HyperspaceException):
        hs.DataSkippingIndexConfig("x")  # no sketches
    with pytest.raises(HyperspaceException):
        hs.DataSkippingIndexConfig(
            "x", hs.MinMaxSketch("a"), hs.MinMaxSketch("a"))  # dup


def test_minmax_skips_files(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "ds1", hs.MinMaxSketch("key")))
    session.enable_hyperspace()
    q = df.filter("key = 2500").select("key", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, Scan)]
    assert scans and scans[0].file_subset is not None, plan.pretty()
    assert scans[0].skipped_files == 7  # only file 2 can contain 2500
    ex = Executor(session)
    out = ex.execute(plan)
    assert ex.stats.scanned_files == 1
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["key", "val"]) == _rows(base, ["key", "val"])


def test_minmax_range_predicates(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "ds1", hs.MinMaxSketch("key")))
    session.enable_hyperspace()
    for pred, max_files in [("key < 1500", 2), ("key >= 6500", 2),
                            ("key <= 999", 1)]:
        q = df.filter(pred).select("key")
        ex = Executor(session)
        out = ex.execute(q.optimized_plan())
        assert ex.stats.scanned_files <= max_files, pred
        session.disable_hyperspace()
        assert out.num_rows == q.collect().num_rows, pred
        session.enable_hyperspace()


def test_bloom_skips_files(env):
    session, h, df, data, rng = env
    # add a column where equality has no range structure: use cat values
    # that only exist in one file
    t = pa.table({"key": rng.integers(0, 8000, 1000),
                  "cat": np.full(1000, 777, dtype=np.int64),
                  "val": rng.random(1000)})
    pq.write_table(t, str(data / "part-special.parquet"))
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsb", hs.BloomFilterSketch("cat", 0.01, 1000)))
    session.enable_hyperspace()
    q = df.filter("cat = 777").select("cat", "val")
    ex = Executor(session)
    out = ex.execute(q.optimized_plan())
    assert ex.stats.scanned_files <= 2  # the special file (+ fp allowance)
    session.disable_hyperspace()
    assert out.num_rows == q.collect().num_rows == 1000


def test_combined_and_or_translation(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "ds2", hs.MinMaxSketch("key"), hs.MinMaxSketch("cat")))
    session.enable_hyperspace()
    # AND: either side prunes; OR: both must convert
    q = df.filter("key <= 999 AND cat <= 5").select("key", "cat")
    ex = Executor(session)
    out = ex.execute(q.optimized_plan())
    assert ex.stats.scanned_files == 1
    session.disable_hyperspace()
    assert out.num_rows == q.collect().num_rows


def test_covering_index_beats_dataskipping(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "ds1", hs.MinMaxSketch("key")))
    h.create_index(df, hs.CoveringIndexConfig("ci1", ["key"], ["val"]))
    session.enable_hyperspace()
    q = df.filter("key = 2500").select("key", "val")
    plan = q.optimized_plan()
    from hyperspace_amd.plan.nodes import IndexScan
    # covering index (score 50) wins over data skipping (score 1)
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())


def test_dataskipping_json_roundtrip(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsr", hs.MinMaxSketch("key"),
        hs.BloomFilterSketch("cat", 0.05, 500)))
    entry = session.index_manager().get_index("dsr")
    d = entry.to_json()
    assert d["derivedDataset"]["type"].endswith("DataSkippingIndex")
    from hyperspace_amd.log.entry import IndexLogEntry
    back = IndexLogEntry.from_json(d)
    kinds = [s.kind for s in back.derivedDataset.sketches]
    assert kinds == ["MinMax", "BloomFilter"]
    assert back.derivedDataset.sketches[1].fpp == 0.05


def test_dataskipping_refresh_incremental(env, tmp_path):
    session, h, df, data, rng = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsr2", hs.MinMaxSketch("key")))
    # append a file with a disjoint key range
    t = pa.table({"key": rng.integers(50_000, 51_000, 2000),
                  "cat": rng.integers(0, 10, 2000),
                  "val": rng.random(2000)})
    pq.write_table(t, str(data / "part-new.parquet"))
    h.refresh_index("dsr2", "incremental")
    entry = session.index_manager().get_index("dsr2")
    assert len(entry.source_file_infos()) == 9
    session.enable_hyperspace()
    # a query on the NEW range must skip the 8 old files
    q = df.filter("key >= 50000").select("key", "val")
    ex = Executor(session)
    out = ex.execute(q.optimized_plan())
    assert ex.stats.scanned_files == 1
    session.disable_hyperspace()
    assert out.num_rows == q.collect().num_rows == 2000


def test_dataskipping_refresh_incremental_deletes(env, tmp_path):
    import os as _os
    session, h, df, data, rng = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsr3", hs.MinMaxSketch("key")))
    _os.unlink(str(data / "part-7.parquet"))
    h.refresh_index("dsr3", "incremental")
    entry = session.index_manager().get_index("dsr3")
    assert len(entry.source_file_infos()) == 7
    session.enable_hyperspace()
    q = df.filter("key = 2500").select("key", "val")
    accel = q.collect()
    session.disable_hyperspace()
    assert accel.num_rows == q.collect().num_rows


def test_dataskipping_refresh_full(env, tmp_path):
    session, h, df, data, rng = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsr4", hs.MinMaxSketch("key")))
    t = pa.table({"key": rng.integers(60_000, 61_000, 1000),
                  "cat": rng.integers(0, 10, 1000),
                  "val": rng.random(1000)})
    pq.write_table(t, str(data / "part-new2.parquet"))
    h.refresh_index("dsr4", "full")
    entry = session.index_manager().get_index("dsr4")
    assert entry.derivedDataset.kind == "DataSkippingIndex"
    assert len(entry.source_file_infos()) == 9


def test_zorder_refresh_full(env, tmp_path):
    session, h, df, data, rng = env
    h.create_index(df, hs.ZOrderCoveringIndexConfig(
        "zr4", ["key", "cat"], ["val"]))
    t = pa.table({"key": rng.integers(0, 8000, 1000),
                  "cat": rng.integers(0, 10, 1000),
                  "val": rng.random(1000)})
    pq.write_table(t, str(data / "part-newz.parquet"))
    h.refresh_index("zr4", "full")
    entry = session.index_manager().get_index("zr4")
    assert entry.derivedDataset.kind == "ZOrderCoveringIndex"
    assert len(entry.source_file_infos()) == 9


def test_sketch_over_arithmetic_expression(tmp_path, monkeypatch):
    """Sketches over scalar expressions (reference ExpressionUtils
    accepts deterministic scalar exprs): MinMaxSketch('key % 100')
    converts predicates whose LHS is structurally the same expression."""
    from hyperspace_amd.plan.expr import col
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    d = tmp_path / "data"
    d.mkdir()
    rng = np.random.default_rng(91)
    # file i holds keys whose (key % 100) lies in [i*10, i*10+10)
    all_keys = []
    for i in range(8):
        base = rng.integers(0, 50, 4000) * 100
        key = base + rng.integers(i * 10, i * 10 + 10, 4000)
        all_keys.append(key)
        pq.write_table(pa.table({"key": key, "val": rng.random(4000)}),
                       str(d / f"part-{i}.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dse", hs.MinMaxSketch("key % 100")))
    session.enable_hyperspace()
    q = df.filter((col("key") % 100) == 42).select("key", "val")
    plan = q.optimized_plan()
    leaf = plan.collect_leaves()[0]
    assert leaf.file_subset is not None and len(leaf.file_subset) == 1
    out = Executor(session).execute(plan)
    expected = sum(int((k % 100 == 42).sum()) for k in all_keys)
    assert out.num_rows == expected
    assert (out.tensor("key") % 100 == 42).all()
    # range over the expression
    q2 = df.filter((col("key") % 100) >= 55)
    leaf2 = q2.optimized_plan().collect_leaves()[0]
    # files 5,6,7 cover remainders [50,80)
    assert leaf2.file_subset is not None and len(leaf2.file_subset) == 3
    out2 = Executor(session).execute(q2.optimized_plan())
    expected2 = sum(int((k % 100 >= 55).sum()) for k in all_keys)
    assert out2.num_rows == expected2


def test_sketch_file_sizing(tmp_path, monkeypatch):
    """write() splits sketch rows to targetIndexDataFileSize (reference
    DataSkippingIndex writeImpl sizing, DataSkippingIndex.scala:187-206)
    and the split index still prunes correctly."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    data = tmp_path / "data"
    data.mkdir()
    rng = np.random.default_rng(3)
    for i in range(40):
        lo = i * 100
        pq.write_table(
            pa.table({"key": rng.integers(lo, lo + 100, 500),
                      "val": rng.random(500)}),
            str(data / f"part-{i:02d}.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    # ~40 sketch rows; force a tiny target so the write must split
    session.conf.set(
        hs.IndexConstants.DATASKIPPING_TARGET_INDEX_DATA_FILE_SIZE, 200)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    from hyperspace_amd.index.dataskipping.sketches import MinMaxSketch
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsz", MinMaxSketch("key")))
    entry = session.index_manager().get_index("dsz")
    files = [p for p in entry.content.os_files()
             if p.endswith(".parquet")]
    assert len(files) > 1, "expected the sketch data to split"
    session.enable_hyperspace()
    from hyperspace_amd.execution.executor import Executor
    ex = Executor(session)
    out = ex.execute(df.filter("key >= 3900").optimized_plan())
    t = pq.read_table(str(data))
    assert out.num_rows == int(
        (t.column("key").to_numpy() >= 3900).sum())
    assert ex.stats.scanned_files == 1  # only the last file can match


# ---------------------------------------------------------------------------
# Generalized sketch expressions (reference ExpressionUtils.scala:38-95:
# arbitrary deterministic scalar expressions over one column)
# ---------------------------------------------------------------------------

def test_parse_expr_precedence_and_canonical():
    from hyperspace_amd.index.dataskipping.sketches import (
        parse_expr_string, expr_to_string)
    from hyperspace_amd.plan.expr import Arith, Col, Lit, _expr_eq
    t = parse_expr_string("a * 2 + 1")
    assert _expr_eq(t, Arith("+", Arith("*", Col("a"), Lit(2)), Lit(1)))
    t2 = parse_expr_string("a * (2 + 1)")
    assert _expr_eq(t2, Arith("*", Col("a"), Arith("+", Lit(2), Lit(1))))
    assert expr_to_string(t) == "(a * 2) + 1"
    # single binary op keeps the round-1 flat naming
    assert expr_to_string(parse_expr_string("a % 10")) == "a % 10"


def test_sketch_expr_validation():
    from hyperspace_amd.exceptions import HyperspaceException
    with pytest.raises(HyperspaceException):
        hs.MinMaxSketch("a + b").base_column  # two columns
    with pytest.raises(HyperspaceException):
        hs.MinMaxSketch("1 + 2").base_column  # no column
    with pytest.raises(HyperspaceException):
        hs.MinMaxSketch("a +").base_column  # syntax


def test_minmax_over_composed_expression(env):
    """Sketch over key*2+1: a filter with the SAME expression as its
    LHS prunes files; results equal the unindexed run (oracle)."""
    from hyperspace_amd.plan.expr import Col
    session, h, df, data, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dse", hs.MinMaxSketch("key * 2 + 1")))
    session.enable_hyperspace()
    # key=2500 -> key*2+1 = 5001; only file 2 qualifies
    cond = (Col("key") * 2 + 1) == 5001
    q = df.filter(cond).select("key", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, Scan)]
    assert scans and scans[0].file_subset is not None, plan.pretty()
    assert scans[0].skipped_files == 7
    out = Executor(session).execute(plan)
    session.disable_hyperspace()
    base = q.collect()
    assert _rows(out, ["key", "val"]) == _rows(base, ["key", "val"])


def test_minmax_range_over_composed_expression(env):
    from hyperspace_amd.plan.expr import Col
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsr", hs.MinMaxSketch("(key + 500) % 4000")))
    session.enable_hyperspace()
    cond = ((Col("key") + 500) % 4000) < 200
    q = df.filter(cond).select("key")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, Scan)]
    assert scans and scans[0].skipped_files > 0, plan.pretty()
    out = Executor(session).execute(plan)
    session.disable_hyperspace()
    assert out.num_rows == q.collect().num_rows


def test_bloom_over_composed_expression(env):
    from hyperspace_amd.plan.expr import Col
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsb", hs.BloomFilterSketch("key * 3 + 2",
                                    expected_distinct=2000)))
    session.enable_hyperspace()
    cond = (Col("key") * 3 + 2) == 7502  # key = 2500 -> file 2 only
    q = df.filter(cond).select("key", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, Scan)]
    assert scans and scans[0].file_subset is not None
    assert scans[0].skipped_files >= 6  # bloom fpp may keep an extra
    out = Executor(session).execute(plan)
    session.disable_hyperspace()
    assert _rows(out, ["key", "val"]) == _rows(q.collect(),
                                               ["key", "val"])


def test_composed_expr_oracle_vs_numpy(env):
    """Build + prune correctness vs a numpy oracle on every file."""
    session, h, df, data, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dso", hs.MinMaxSketch("key % 100 * 2")))
    entry = session.index_manager().get_index("dso")
    idx = entry.derivedDataset
    sd = idx.load_sketch_data(entry)
    mn = sd.tensor("MinMax_(key % 100) * 2__min")
    mx = sd.tensor("MinMax_(key % 100) * 2__max")
    import pyarrow.parquet as pq
    files = sorted(str(data / f) for f in
                   __import__("os").listdir(data))
    for i, p in enumerate(sorted(
            f.name for f in entry.source_file_infos())):
        k = pq.read_table(p, columns=["key"]).column("key").to_numpy()
        v = (k % 100) * 2
        # sketch aggregates store normalize_key()-ordered values
        from hyperspace_amd.ops import cpu_ref
        vn = cpu_ref.normalize_key(torch.from_numpy(v))
        assert int(mn[i]) == int(vn.min())
        assert int(mx[i]) == int(vn.max())


def test_dataskipping_over_delta_table(tmp_path, monkeypatch):
    """Sketch index over a Delta table: per-file pruning composes with
    the transaction-log file listing (reference DataSkipping +
    DeltaLakeIntegration crossover)."""
    import torch
    from hyperspace_amd.sources.delta_source import DeltaTable
    from hyperspace_amd.execution.columnar import ColumnBatch
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(13)
    t = DeltaTable.create(str(tmp_path / "t"))
    for i in range(6):  # disjoint ranges -> minmax skipping bites
        t.append_batch(ColumnBatch({
            "key": torch.from_numpy(
                rng.integers(i * 100, (i + 1) * 100, 2000)),
            "val": torch.from_numpy(rng.random(2000))}))
    session = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(session)
    df = session.read_delta(str(tmp_path / "t"))
    h.create_index(df, hs.DataSkippingIndexConfig(
        "dsd", hs.MinMaxSketch("key")))
    session.enable_hyperspace()
    q = df.filter("key = 250").select("key", "val")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, Scan)]
    assert scans and scans[0].file_subset is not None
    assert scans[0].skipped_files == 5
    out = Executor(session).execute(plan)
    session.disable_hyperspace()
    assert out.num_rows == q.collect().num_rows


def test_covering_index_outscores_dataskipping(env):
    """When a covering filter index AND a data-skipping index both apply
    to the same Filter(Scan), the score-based optimizer must pick the
    covering rewrite (50 x coverage vs the DS flat score of 1;
    reference: ApplyDataSkippingIndex.scala score + ScoreBased
    optimizer)."""
    session, h, df, _, _ = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "ds_both", hs.MinMaxSketch("key")))
    h.create_index(df, hs.CoveringIndexConfig(
        "ci_both", ["key"], ["val"]))
    session.enable_hyperspace()
    q = df.filter("key = 1500").select("key", "val")
    plan = q.optimized_plan().pretty()
    assert "IndexScan(ci_both" in plan, plan
    assert "dataskipping" not in plan, plan
    # with the covering index gone, the DS index takes over
    h.delete_index("ci_both")
    h.vacuum_index("ci_both")
    q2 = df.filter("key = 1500").select("key", "val")
    plan2 = q2.optimized_plan().pretty()
    assert "dataskipping:-7files" in plan2, plan2
