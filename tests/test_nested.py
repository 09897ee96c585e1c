"""Nested column indexing: struct leaves flatten to dotted columns at
the Arrow boundary; the reference's ``__hs_nested.a.b.c`` spelling is an
accepted alias (reference util/ResolverUtils.scala nested resolution,
index/covering/CoveringIndexConfig nested support)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.execution.columnar import ColumnBatch
from hyperspace_amd.plan.nodes import IndexScan, Scan
from hyperspace_amd.sources.parquet_io import read_files_batch
from hyperspace_amd.utils.resolver import resolve, resolve_all

N = 20000


def _nested_table(rng, n):
    score = rng.integers(0, 1000, n)
    cnt = rng.integers(0, 50, n)
    val = rng.random(n)
    info = pa.StructArray.from_arrays(
        [pa.array(score), pa.array(cnt)], names=["score", "cnt"])
    return pa.table({"info": info, "v": pa.array(val)}), score, cnt, val


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(44)
    d = tmp_path / "data"
    d.mkdir()
    t, score, cnt, val = _nested_table(rng, N)
    pq.write_table(t.slice(0, N // 2), str(d / "part-0.parquet"))
    pq.write_table(t.slice(N // 2), str(d / "part-1.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    return session, h, df, score, cnt, val


def test_resolver_nested_alias():
    avail = ["info.score", "v"]
    assert resolve(avail, "__hs_nested.info.score") == "info.score"
    assert resolve(avail, "INFO.SCORE") == "info.score"
    assert resolve_all(avail, ["__hs_nested.info.score", "v"]) == \
        ["info.score", "v"]


def test_from_arrow_flattens_structs():
    rng = np.random.default_rng(1)
    t, score, cnt, val = _nested_table(rng, 100)
    b = ColumnBatch.from_arrow(t)
    assert set(b.names) == {"info.score", "info.cnt", "v"}
    assert np.array_equal(b.tensor("info.score").numpy(), score)


def test_struct_null_propagation():
    inner = pa.StructArray.from_arrays(
        [pa.array([1, 2, 3], type=pa.int64())], names=["x"],
        mask=pa.array([False, True, False]))  # row 1: struct itself null
    b = ColumnBatch.from_arrow(pa.table({"s": inner}))
    m = b.mask("s.x")
    assert m is not None and m.tolist() == [True, False, True]


def test_read_files_batch_nested_column_pruning(tmp_path):
    rng = np.random.default_rng(2)
    t, score, cnt, val = _nested_table(rng, 500)
    p = str(tmp_path / "n.parquet")
    pq.write_table(t, p)
    batch, counts = read_files_batch([p], columns=["info.score", "v"])
    assert counts == [500]
    assert set(batch.names) == {"info.score", "v"}
    assert np.array_equal(batch.tensor("info.score").numpy(), score)


def test_schema_contains_nested_leaves(env):
    session, h, df, *_ = env
    leaf = df.plan.collect_leaves()[0]
    names = leaf.relation.schema.field_names()
    assert "info.score" in names and "info.cnt" in names


def test_covering_index_on_nested_column(env):
    session, h, df, score, cnt, val = env
    h.create_index(df, hs.CoveringIndexConfig(
        "nst", ["info.score"], ["v"]))
    session.enable_hyperspace()
    q = df.filter("info.score = 77").select("info.score", "v")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves()), \
        plan.pretty()
    out = q.collect()
    keep = score == 77
    got = sorted(zip(out.tensor("info.score").tolist(),
                     np.round(out.tensor("v").numpy(), 9).tolist()))
    want = sorted(zip(score[keep].tolist(),
                      np.round(val[keep], 9).tolist()))
    assert got == want


def test_nested_config_via_reference_prefix(env):
    # configs written against the reference's __hs_nested. spelling work
    session, h, df, score, cnt, val = env
    h.create_index(df, hs.CoveringIndexConfig(
        "nstp", ["__hs_nested.info.cnt"], ["v"]))
    entry = session.index_manager().get_index("nstp")
    assert entry.derivedDataset.indexed_columns == ["info.cnt"]
    session.enable_hyperspace()
    out = df.filter("info.cnt = 7").select("info.cnt", "v").collect()
    assert out.num_rows == int((cnt == 7).sum())


def test_dataskipping_on_nested_column(env):
    session, h, df, score, cnt, val = env
    h.create_index(df, hs.DataSkippingIndexConfig(
        "nsk", hs.MinMaxSketch("info.score")))
    session.enable_hyperspace()
    out = df.filter("info.score >= 990")
    got = out.collect()
    assert got.num_rows == int((score >= 990).sum())


def test_nested_leaf_incremental_refresh(tmp_path, monkeypatch):
    """Incremental refresh over a nested-leaf index (reference
    RefreshIndexNestedTest): appended files with the same struct shape
    merge into the index and serve flattened queries."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(9)
    d = tmp_path / "src"
    d.mkdir()

    def table(n, base):
        return pa.table({
            "nested": pa.StructArray.from_arrays(
                [pa.array(rng.integers(base, base + 50, n)),
                 pa.array(rng.random(n))], names=["leaf", "w"]),
            "val": rng.random(n)})

    pq.write_table(table(3000, 0), str(d / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig(
        "nrx", ["nested.leaf"], ["val"]))
    session.enable_hyperspace()
    before = df.filter("nested.leaf = 7").collect().num_rows

    pq.write_table(table(3000, 0), str(d / "part-1.parquet"))
    h.refresh_index("nrx", "incremental")
    df2 = session.read_parquet(str(d))
    q = df2.filter("nested.leaf = 7").select("nested.leaf", "val")
    from hyperspace_amd.plan.nodes import IndexScan
    assert any(isinstance(l, IndexScan)
               for l in q.optimized_plan().collect_leaves())
    got = q.collect().num_rows
    session.disable_hyperspace()
    assert got == df2.filter("nested.leaf = 7").collect().num_rows
    assert got >= before


def test_nullable_struct_leaves_native_host(tmp_path):
    """Nullable struct with nullable leaf (max_def=2): definition
    levels are 2-bit; the native reader decodes level==max_def as the
    leaf mask with parent-null propagation (parquet-format.md nested
    encoding; reference reads via parquet-mr)."""
    from hyperspace_amd.sources.native_parquet import (read_native_host,
                                                       read_native_layout)
    n = 50_000
    structs = [None if i % 13 == 0 else
               {"x": (None if i % 7 == 0 else i), "y": float(i)}
               for i in range(n)]
    t = pa.table({"s": pa.array(structs,
                                type=pa.struct([("x", pa.int64()),
                                                ("y", pa.float64())])),
                  "k": np.arange(n)})
    p = str(tmp_path / "nest.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_version="1.0")
    lay = read_native_layout(p)
    assert lay is not None
    assert {c.name: c.max_def for c in lay[1]} == \
        {"s.x": 2, "s.y": 2, "k": 1}
    cols, masks = read_native_host(p)
    mx, vx = masks["s.x"], np.asarray(cols["s.x"])
    my, vy = masks["s.y"], np.asarray(cols["s.y"])
    for i in range(0, n, 313):
        if structs[i] is None or structs[i]["x"] is None:
            assert not mx[i]
        else:
            assert mx[i] and vx[i] == i
        if structs[i] is None:
            assert not my[i]
        else:
            assert my[i] and vy[i] == float(i)
    # repeated (list) leaves still decline
    tl = pa.table({"l": pa.array([[1, 2], [3]] * 100)})
    pl = str(tmp_path / "list.parquet")
    pq.write_table(tl, pl, compression="NONE")
    assert read_native_layout(pl) is None


def test_list_column_source_scalar_index(tmp_path, monkeypatch):
    """A source containing a LIST column (unsupported as an engine
    column type) must not break indexing/querying of its scalar
    columns — the column-pruned reads never touch the list."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "ix"))
    d = tmp_path / "t"
    d.mkdir()
    pq.write_table(pa.table({
        "k": np.arange(5000),
        "v": np.random.default_rng(0).random(5000),
        "tags": pa.array([[1, 2], [3]] * 2500)}),
        str(d / "p0.parquet"))
    s = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(s)
    df = s.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("lx", ["k"], ["v"]))
    s.enable_hyperspace()
    out = df.filter("k = 42").select("k", "v").collect()
    assert out.num_rows == 1
