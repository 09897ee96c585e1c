"""Nullable-column support: validity masks through the columnar batch,
native Parquet encode/decode (OPTIONAL + RLE def-levels), index build
(NULLS FIRST sort, Spark null bucketing), SQL null semantics in filters
and inner joins, and null-ignoring data-skipping sketches.

Reference behavior: Spark SQL null semantics, which the reference
inherits by indexing through DataFrames (e.g. sorts are ASC NULLS FIRST,
Murmur3Hash passes the seed through null children, comparisons never
match null, inner joins drop null keys)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.execution.columnar import ColumnBatch
from hyperspace_amd.ops import cpu_ref
from hyperspace_amd.plan.expr import col
from hyperspace_amd.plan.nodes import IndexScan
from hyperspace_amd.sources.native_parquet import (read_native_host,
                                                   write_parquet_native)
from hyperspace_amd.sources.parquet_io import read_files_batch

N = 20000


# ---------------------------------------------------------------------------
# ColumnBatch mask mechanics
# ---------------------------------------------------------------------------

def test_batch_mask_transforms():
    rng = np.random.default_rng(0)
    v = torch.from_numpy(rng.integers(0, 100, 1000))
    m = torch.from_numpy(rng.random(1000) > 0.3)
    b = ColumnBatch({"a": v, "b": v.clone()}, {"a": m})
    assert b.has_nulls("a") and not b.has_nulls("b")
    idx = torch.arange(0, 1000, 7)
    g = b.gather(idx)
    assert torch.equal(g.mask("a"), m[idx])
    s = b.slice(10, 200)
    assert torch.equal(s.mask("a"), m[10:200])
    assert b.select(["a"]).mask("a") is not None
    assert b.select(["b"]).mask("b") is None
    assert b.drop("a").mask("a") is None
    c = ColumnBatch.concat([b, b.slice(0, 50)])
    assert c.num_rows == 1050
    assert torch.equal(c.mask("a"), torch.cat([m, m[:50]]))
    # a batch without the mask contributes all-valid rows
    c2 = ColumnBatch.concat([b, ColumnBatch({"a": v[:5], "b": v[:5]})])
    assert bool(c2.mask("a")[1000:].all())


def test_arrow_mask_roundtrip():
    rng = np.random.default_rng(1)
    vals = rng.integers(0, 50, 500)
    mask = rng.random(500) > 0.2
    t = pa.table({"x": pa.array(vals, mask=~mask)})
    b = ColumnBatch.from_arrow(t)
    assert np.array_equal(b.mask("x").numpy(), mask)
    assert np.array_equal(b.tensor("x").numpy()[mask], vals[mask])
    back = b.to_arrow()
    assert back.column("x").null_count == int((~mask).sum())
    assert back.column("x").combine_chunks().is_valid().to_numpy(
        zero_copy_only=False).tolist() == mask.tolist()


# ---------------------------------------------------------------------------
# Native Parquet OPTIONAL encode/decode
# ---------------------------------------------------------------------------

def test_native_write_pyarrow_reads_nulls(tmp_path):
    rng = np.random.default_rng(2)
    vals = rng.integers(-1000, 1000, 5000)
    mask = rng.random(5000) > 0.15
    d = rng.random(5000)
    dmask = rng.random(5000) > 0.5
    p = str(tmp_path / "n.parquet")
    write_parquet_native({"k": vals, "d": d, "req": vals.copy()}, p,
                         masks={"k": mask, "d": dmask})
    t = pq.read_table(p)
    assert t.column("k").null_count == int((~mask).sum())
    got = t.column("k").to_numpy(zero_copy_only=False)
    assert np.array_equal(got[mask], vals[mask].astype(float))
    assert np.isnan(got[~mask]).all()
    assert t.column("req").null_count == 0
    # statistics ignore nulls
    st = None
    md = pq.ParquetFile(p).metadata.row_group(0)
    for i in range(md.num_columns):
        if md.column(i).path_in_schema == "k":
            st = md.column(i).statistics
    assert st.null_count == int((~mask).sum())
    assert st.min == vals[mask].min() and st.max == vals[mask].max()


def test_native_roundtrip_nullable(tmp_path):
    rng = np.random.default_rng(3)
    vals = rng.integers(0, 9, 3000).astype(np.int64)
    mask = rng.random(3000) > 0.25
    p = str(tmp_path / "n.parquet")
    write_parquet_native({"k": vals, "plainc": vals.copy()}, p,
                         masks={"k": mask})
    res = read_native_host(p)
    assert res is not None
    cols, masks = res
    assert np.array_equal(masks["k"], mask)
    assert np.array_equal(cols["k"][mask], vals[mask])
    assert (cols["k"][~mask] == 0).all()
    assert "plainc" not in masks
    # column subset
    cols2, masks2 = read_native_host(p, columns=["k"])
    assert list(cols2) == ["k"] and np.array_equal(masks2["k"], mask)


def test_native_reads_pyarrow_nullable(tmp_path):
    # pyarrow encodes def-levels with its own RLE/bit-packed mix
    rng = np.random.default_rng(4)
    vals = rng.integers(0, 100, 60000).astype(np.int64)
    mask = rng.random(60000) > 0.1
    mask[:5000] = True   # long all-valid prefix -> RLE run
    mask[5000:5100] = False  # long null run
    arr = pa.array(vals, mask=~mask)
    p = str(tmp_path / "pa.parquet")
    pq.write_table(pa.table({"k": arr}), p, compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
    res = read_native_host(p)
    assert res is not None
    cols, masks = res
    assert np.array_equal(masks["k"], mask)
    assert np.array_equal(cols["k"][mask], vals[mask])


def test_read_files_batch_mixed_null_files(tmp_path):
    rng = np.random.default_rng(5)
    v1 = rng.integers(0, 10, 100).astype(np.int64)
    m1 = rng.random(100) > 0.5
    v2 = rng.integers(0, 10, 80).astype(np.int64)
    p1, p2 = str(tmp_path / "a.parquet"), str(tmp_path / "b.parquet")
    write_parquet_native({"k": v1}, p1, masks={"k": m1})
    write_parquet_native({"k": v2}, p2)  # no nulls
    batch, counts = read_files_batch([p1, p2])
    assert counts == [100, 80]
    m = batch.mask("k")
    assert m is not None
    assert np.array_equal(m.numpy()[:100], m1)
    assert bool(m[100:].all())


def test_empty_after_mask_all_valid(tmp_path):
    # an all-valid mask degrades to the REQUIRED fast path
    p = str(tmp_path / "av.parquet")
    vals = np.arange(10, dtype=np.int64)
    write_parquet_native({"k": vals}, p,
                         masks={"k": np.ones(10, dtype=bool)})
    md = pq.ParquetFile(p).metadata
    assert md.row_group(0).column(0).statistics.null_count == 0
    cols, masks = read_native_host(p)
    assert masks == {} and np.array_equal(cols["k"], vals)


# ---------------------------------------------------------------------------
# Null bucketing (Spark HashPartitioning semantics)
# ---------------------------------------------------------------------------

def test_null_rows_bucket_to_pmod_seed():
    vals = torch.arange(100, dtype=torch.int64)
    mask = torch.ones(100, dtype=torch.bool)
    mask[::7] = False
    nb = 16
    b = cpu_ref.murmur3_bucket([vals], nb, [mask])
    expected_null_bucket = cpu_ref.SPARK_HASH_SEED % nb
    assert (b[~mask] == expected_null_bucket).all()
    # valid rows hash as if no mask existed
    b_plain = cpu_ref.murmur3_bucket([vals], nb)
    assert torch.equal(b[mask], b_plain[mask])


# ---------------------------------------------------------------------------
# End-to-end: filters, joins, sketches over nullable data
# ---------------------------------------------------------------------------

@pytest.fixture
def nullable_env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(7)
    data_dir = tmp_path / "data"
    data_dir.mkdir()
    key = rng.integers(0, 500, N)
    kmask = rng.random(N) > 0.1
    key[::97] = 0  # ensure literal-0 rows exist alongside nulls stored as 0
    val = (rng.random(N) * 100).astype(np.float64)
    for i in range(2):
        sl = slice(i * N // 2, (i + 1) * N // 2)
        t = pa.table({
            "k": pa.array(key[sl], mask=~kmask[sl]),
            "v": pa.array(val[sl]),
        })
        pq.write_table(t, str(data_dir / f"part-{i}.parquet"),
                       compression="NONE", use_dictionary=False,
                       data_page_version="1.0")
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data_dir))
    return session, h, df, key, kmask, val


def _expected(key, kmask, val, pred):
    keep = kmask & pred(key)
    return sorted(zip(key[keep].tolist(), val[keep].tolist()))


def _got(q):
    arrs = q.collect().to_numpy()
    return sorted(zip(arrs["k"].tolist(), arrs["v"].tolist()))


def test_filter_null_semantics_indexed(nullable_env):
    session, h, df, key, kmask, val = nullable_env
    h.create_index(df, hs.CoveringIndexConfig("nidx", ["k"], ["v"]))
    session.enable_hyperspace()

    q = df.filter("k = 0").select("k", "v")
    assert any(isinstance(l, IndexScan)
               for l in q.optimized_plan().collect_leaves())
    # null slots store 0: they must NOT match the literal 0
    assert _got(q) == _expected(key, kmask, val, lambda k: k == 0)

    assert _got(df.filter("k >= 400").select("k", "v")) == \
        _expected(key, kmask, val, lambda k: k >= 400)
    assert _got(df.filter("k < 3").select("k", "v")) == \
        _expected(key, kmask, val, lambda k: k < 3)

    # IS NULL / IS NOT NULL over the indexed scan
    n_null = int((~kmask).sum())
    got_null = df.filter(col("k").is_null()).select("k", "v").collect()
    assert got_null.num_rows == n_null
    got_nn = df.filter(col("k").is_not_null()).select("k", "v").collect()
    assert got_nn.num_rows == N - n_null

    # NOT(k = 0): SQL three-valued logic excludes nulls
    got_ne = df.filter(~(col("k") == 0)).select("k", "v")
    assert _got(got_ne) == _expected(key, kmask, val, lambda k: k != 0)

    # IN
    got_in = df.filter(col("k").isin([0, 5, 7])).select("k", "v")
    assert _got(got_in) == _expected(key, kmask, val,
                                     lambda k: np.isin(k, [0, 5, 7]))


def test_filter_null_baseline_equivalence(nullable_env):
    session, h, df, key, kmask, val = nullable_env
    h.create_index(df, hs.CoveringIndexConfig("nidx", ["k"], ["v"]))
    base = _got(df.filter("k <= 250").select("k", "v"))
    session.enable_hyperspace()
    accel = _got(df.filter("k <= 250").select("k", "v"))
    assert accel == base == _expected(key, kmask, val,
                                      lambda k: k <= 250)


def test_join_drops_null_keys(nullable_env, tmp_path):
    session, h, df, key, kmask, val = nullable_env
    rng = np.random.default_rng(8)
    dim_dir = tmp_path / "dim"
    dim_dir.mkdir()
    dk = np.arange(500, dtype=np.int64)
    ds = rng.integers(0, 3, 500)
    pq.write_table(pa.table({"k": dk, "s": ds}),
                   str(dim_dir / "part-0.parquet"))
    dim = session.read_parquet(str(dim_dir))
    h.create_index(df, hs.CoveringIndexConfig("jl", ["k"], ["v"]))
    h.create_index(dim, hs.CoveringIndexConfig("jr", ["k"], ["s"]))
    session.enable_hyperspace()
    out = df.join(dim, on="k").collect()
    # pandas-style inner join: null keys contribute no rows
    expected_rows = int(kmask.sum())  # dim covers every valid key 0..499
    assert out.num_rows == expected_rows


def test_nulls_first_in_index_files(nullable_env):
    session, h, df, key, kmask, val = nullable_env
    h.create_index(df, hs.CoveringIndexConfig("nidx", ["k"], ["v"]))
    entry = session.index_manager().get_index("nidx")
    files = entry.content.os_files()
    assert files
    checked = 0
    for f in files:
        res = read_native_host(f, columns=["k"])
        assert res is not None
        cols, masks = res
        if "k" not in masks:
            continue
        m = masks["k"]
        n_null = int((~m).sum())
        # NULLS FIRST: all nulls at the head of the (single-bucket) file
        assert not m[:n_null].any() and m[n_null:].all()
        ks = cols["k"][m]
        assert (np.diff(ks) >= 0).all()  # then sorted ascending
        checked += 1
    assert checked  # the null bucket exists


def test_minmax_sketch_ignores_nulls(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    d = tmp_path / "sk"
    d.mkdir()
    # file A: small values + nulls (stored 0); file B: large values
    va = np.arange(10, 20, dtype=np.int64)
    ma = np.ones(10, dtype=bool)
    ma[-3:] = False  # 17..19 are null -> file A max must be 16
    pq.write_table(pa.table({"a": pa.array(va, mask=~ma)}),
                   str(d / "pa.parquet"), compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
    pq.write_table(pa.table({"a": np.arange(100, 110, dtype=np.int64)}),
                   str(d / "pb.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.DataSkippingIndexConfig(
        "sk", hs.MinMaxSketch("a")))
    session.enable_hyperspace()
    plan = df.filter("a >= 100").optimized_plan()
    scans = [l for l in plan.collect_leaves()]
    kept = scans[0].file_subset
    assert kept is not None and len(kept) == 1
    assert "pb.parquet" in kept[0]
    # nulls are stored as 0 in the value buffer: if the sketch counted
    # them, file A's min would be 0 and a <= 5 could not skip it
    plan2 = df.filter("a <= 5").optimized_plan()
    leaf2 = plan2.collect_leaves()[0]
    assert leaf2.file_subset == [] and leaf2.skipped_files == 2
    # equivalence with nulls present
    out = df.filter("a <= 12").collect()
    assert sorted(out.tensor("a").tolist()) == [10, 11, 12]


def test_refresh_incremental_nullable(nullable_env, tmp_path):
    session, h, df, key, kmask, val = nullable_env
    h.create_index(df, hs.CoveringIndexConfig("nidx", ["k"], ["v"]))
    # append another nullable file
    rng = np.random.default_rng(9)
    k2 = rng.integers(0, 500, 1000)
    m2 = rng.random(1000) > 0.5
    src = df.plan.collect_leaves()[0].relation.root_paths[0]
    pq.write_table(
        pa.table({"k": pa.array(k2, mask=~m2),
                  "v": pa.array(rng.random(1000))}),
        src + "/part-9.parquet", compression="NONE",
        use_dictionary=False, data_page_version="1.0")
    h.refresh_index("nidx", mode="incremental")
    session.enable_hyperspace()
    df2 = session.read_parquet(src)
    got = df2.filter(col("k").is_null()).collect()
    assert got.num_rows == int((~kmask).sum()) + int((~m2).sum())
