"""Native Parquet writer/reader tests (K1/K3 encode-decode contract):
pyarrow interoperability, statistics, multi-page handling."""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from hyperspace_amd.sources.native_parquet import (
    read_native_host, read_native_layout, write_parquet_native)
from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                               write_batch_parquet)
from hyperspace_amd.execution.columnar import ColumnBatch


@pytest.fixture
def cols():
    rng = np.random.default_rng(0)
    return {
        "key": rng.integers(-(10**12), 10**12, 50_000),
        "val": rng.random(50_000),
        "f": rng.random(50_000).astype(np.float32),
        "i": rng.integers(-100, 100, 50_000).astype(np.int32),
    }


def test_pyarrow_reads_native_files(tmp_path, cols):
    p = str(tmp_path / "t.parquet")
    write_parquet_native(cols, p)
    t = pq.read_table(p)
    assert t.num_rows == 50_000
    for name, arr in cols.items():
        assert np.array_equal(t.column(name).to_numpy(), arr), name
    # column-chunk statistics present and exact (z-order pruning uses them)
    md = pq.ParquetFile(p).metadata.row_group(0)
    st = md.column(0).statistics
    assert st.min == cols["key"].min()
    assert st.max == cols["key"].max()


def test_native_reads_own_files(tmp_path, cols):
    p = str(tmp_path / "t.parquet")
    write_parquet_native(cols, p)
    back, masks = read_native_host(p)
    assert masks == {}
    for name, arr in cols.items():
        assert np.array_equal(back[name], arr), name
    sub, _ = read_native_host(p, columns=["key"])
    assert list(sub.keys()) == ["key"]


def test_native_reads_pyarrow_files_multipage(tmp_path, cols):
    # pyarrow splits into ~20k-row pages: exercises multi-page walking
    p = str(tmp_path / "t.parquet")
    pq.write_table(pa.table(cols), p, compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
    res = read_native_host(p)
    assert res is not None
    back, _ = res
    for name, arr in cols.items():
        assert np.array_equal(back[name], arr), name


def test_fallback_on_compressed(tmp_path, cols):
    # snappy now yields a device-decodable layout; the HOST decode path
    # still defers to pyarrow (device decompression is GPU-only), and
    # truly unsupported codecs yield no layout at all
    p = str(tmp_path / "t.parquet")
    pq.write_table(pa.table(cols), p, compression="SNAPPY")
    lay = read_native_layout(p)
    assert lay is not None
    assert all(c.encoding in ("plain_z", "dict_z") for c in lay[1])
    assert read_native_host(p) is None
    batch, counts = read_files_batch([p])
    assert batch.num_rows == 50_000 and counts == [50_000]
    p2 = str(tmp_path / "t2.parquet")
    # GZIP/ZSTD yield layouts too (host-codec pages on the native
    # assembly path)
    pq.write_table(pa.table(cols), p2, compression="GZIP")
    lay2 = read_native_layout(p2)
    assert lay2 is not None
    assert all(c.codec == "GZIP" for c in lay2[1])
    p3 = str(tmp_path / "t3.parquet")
    # LZ4 (raw-block pages) parses natively too
    pq.write_table(pa.table(cols), p3, compression="LZ4")
    lay3 = read_native_layout(p3)
    assert lay3 is not None
    assert all(c.codec == "LZ4" for c in lay3[1])
    batch2, counts2 = read_files_batch([p3])
    assert batch2.num_rows == 50_000 and counts2 == [50_000]
    # DataPageV2 parses natively too (levels uncompressed at payload
    # start, per-page is_compressed honored)
    p4 = str(tmp_path / "t4.parquet")
    pq.write_table(pa.table(cols), p4, data_page_version="2.0")
    assert read_native_layout(p4) is not None
    batch3, counts3 = read_files_batch([p4])
    assert batch3.num_rows == 50_000 and counts3 == [50_000]
    # a physical type outside the native set (FIXED_LEN_BYTE_ARRAY
    # decimal) yields no layout at all
    p5 = str(tmp_path / "t5.parquet")
    import decimal
    pq.write_table(pa.table({"d": pa.array(
        [decimal.Decimal("1.23")] * 100,
        type=pa.decimal128(20, 2))}), p5)
    assert read_native_layout(p5) is None


def test_write_batch_parquet_uses_native(tmp_path, cols):
    batch = ColumnBatch({k: torch.from_numpy(v) for k, v in cols.items()})
    p = str(tmp_path / "b.parquet")
    write_batch_parquet(batch, p)
    md = pq.ParquetFile(p).metadata
    assert md.created_by.startswith("hyperspace_amd")
    back, _ = read_files_batch([p])
    for name in cols:
        assert torch.equal(back.tensor(name), batch.tensor(name))


def test_empty_and_single_row(tmp_path):
    p = str(tmp_path / "one.parquet")
    write_parquet_native({"a": np.array([7], dtype=np.int64)}, p)
    assert pq.read_table(p).column("a").to_pylist() == [7]
    back, _ = read_native_host(p)
    assert back["a"].tolist() == [7]


def test_multi_rowgroup_native_read(tmp_path):
    rng = np.random.default_rng(9)
    cols = {"key": rng.integers(-(10**9), 10**9, 1_200_000),
            "val": rng.random(1_200_000)}
    p = str(tmp_path / "mrg.parquet")
    pq.write_table(pa.table(cols), p, compression="NONE",
                   use_dictionary=False, data_page_version="1.0",
                   row_group_size=250_000)
    assert pq.ParquetFile(p).metadata.num_row_groups > 1
    res = read_native_host(p)
    assert res is not None
    back, _ = res
    for k, v in cols.items():
        assert np.array_equal(back[k], v), k
    batch, counts = read_files_batch([p])
    assert counts == [1_200_000]


def test_multi_row_group_native_read(tmp_path):
    """Multi-row-group uncompressed files decode natively with correct
    per-row-group boundaries (the decode path splits one file's row
    groups across parallel units on GPU; the host path walks the same
    layout row-group-major)."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from hyperspace_amd.sources.parquet_io import read_files_batch
    rng = np.random.default_rng(3)
    n = 100_000
    key = rng.integers(0, 1000, n)
    val = rng.random(n)
    p = str(tmp_path / "mrg.parquet")
    pq.write_table(pa.table({"key": key, "val": val}), p,
                   compression="NONE", use_dictionary=False,
                   data_page_version="1.0", row_group_size=7_000)
    assert pq.ParquetFile(p).metadata.num_row_groups > 10
    batch, rc = read_files_batch([p])
    assert rc == [n]
    assert (batch.tensor("key").numpy() == key).all()
    assert np.allclose(batch.tensor("val").numpy(), val)


def test_split_row_groups_boundaries():
    """The GPU decode's row-group splitter: layout order is row-group-
    major, so a repeated column name starts a new group and row offsets
    accumulate per group."""
    from hyperspace_amd.sources.native_parquet import ColumnChunkLayout
    import numpy as np

    def chunk(name, nrows):
        return ColumnChunkLayout(name, np.dtype("int64"),
                                 [("plain", 0, nrows)], nrows)

    chunks = [chunk("a", 10), chunk("b", 10),
              chunk("a", 7), chunk("b", 7),
              chunk("a", 3), chunk("b", 3)]
    # reproduce the splitter used by read_files_batch_device
    groups = []
    cur, seen, row_off = [], set(), 0
    for c in chunks:
        if c.name in seen:
            groups.append((row_off, cur))
            row_off += cur[0].num_values
            cur, seen = [], set()
        cur.append(c)
        seen.add(c.name)
    if cur:
        groups.append((row_off, cur))
    assert [(off, [c.name for c in cs], cs[0].num_values)
            for off, cs in groups] == \
        [(0, ["a", "b"], 10), (10, ["a", "b"], 7), (17, ["a", "b"], 3)]


def test_data_page_v2_host_decode(tmp_path):
    """DataPageV2: definition levels live uncompressed at the payload
    start with a header-declared byte length (no 4-byte prefix); the
    host decode must match pyarrow for numerics, nullables and both
    string encodings (parquet-format.md DataPageHeaderV2)."""
    from hyperspace_amd.sources.native_parquet import read_native_host
    rng = np.random.default_rng(41)
    n = 80_000
    t = pa.table({
        "k": rng.integers(0, 5000, n),
        "v": rng.random(n),
        "nn": pa.array([None if i % 11 == 0 else int(i)
                        for i in range(n)], type=pa.int64()),
    })
    p = str(tmp_path / "v2.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_version="2.0")
    cols, masks = read_native_host(p)
    assert (np.asarray(cols["k"]) == t["k"].to_numpy()).all()
    assert np.allclose(np.asarray(cols["v"]), t["v"].to_numpy())
    m = masks["nn"]
    assert not m[0] and m[1]
    nn = np.asarray(cols["nn"])
    assert nn[1] == 1 and nn[12] == 12
    # dict + plain strings
    for ud in (True, False):
        ps = str(tmp_path / f"v2s{ud}.parquet")
        pq.write_table(pa.table({"s": [f"s{i % 997:04d}"
                                       for i in range(n)]}), ps,
                       compression="NONE", use_dictionary=ud,
                       data_page_version="2.0")
        cs, _ = read_native_host(ps)
        s = cs["s"]
        assert all(s.values[s.codes[i]] == f"s{i % 997:04d}"
                   for i in range(0, n, 997))


def test_spark_shaped_decimal_int64_native(tmp_path):
    """Spark writes decimal(p<=18) with INT64/INT32 physical type and a
    DECIMAL logical annotation; the native layout decodes the unscaled
    integers directly — exactly the engine's decimal representation
    (execution/columnar.py from_arrow).  FLBA-physical decimals
    (pyarrow default, p>18) decline to the pyarrow path."""
    import decimal
    vals = [decimal.Decimal(f"{i}.25") for i in range(10_000)]
    t = pa.table({"d": pa.array(vals, type=pa.decimal128(12, 2))})
    p = str(tmp_path / "dec64.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   store_decimal_as_integer=True,
                   data_page_version="1.0")
    md = pq.ParquetFile(p).metadata.row_group(0).column(0)
    assert md.physical_type == "INT64"
    cols, _ = read_native_host(p)
    unscaled = np.asarray(cols["d"])
    assert unscaled[5] == 525 and unscaled[9999] == 999925
    # FLBA-physical (pyarrow default): no native layout
    p2 = str(tmp_path / "decflba.parquet")
    pq.write_table(t, p2)
    assert read_native_layout(p2) is None


def test_decimal_filter_over_native_read(tmp_path):
    """End to end: a decimal filter binds its literal to the unscaled
    representation and serves from the natively-decoded INT64-physical
    column."""
    import decimal
    import hyperspace_amd as hs
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp_path / "ix")
    d = tmp_path / "t"
    d.mkdir()
    vals = [decimal.Decimal(f"{i}.25") for i in range(10_000)]
    pq.write_table(pa.table({"d": pa.array(vals,
                                           type=pa.decimal128(12, 2)),
                             "k": list(range(10_000))}),
                   str(d / "p0.parquet"), compression="NONE",
                   use_dictionary=False, store_decimal_as_integer=True,
                   data_page_version="1.0")
    s = hs.HyperspaceSession(device="cpu")
    df = s.read_parquet(str(d))
    out = df.filter("d > 9998.0").select("k", "d").collect()
    assert out.num_rows == 2


def test_mixed_native_and_fallback_files_in_one_batch(tmp_path):
    """A batch mixing native-layout files with a non-native one (FLBA
    decimal physical type) merges per file — the legacy file reads via
    pyarrow, the rest stay on the native path, content is exact."""
    import decimal
    rng = np.random.default_rng(13)
    p1 = str(tmp_path / "native.parquet")
    write_parquet_native({"k": np.arange(1000, dtype=np.int64),
                          "d": np.arange(1000, dtype=np.int64) * 100},
                         p1)
    # non-native: FLBA decimal (pyarrow default decimal128 layout),
    # same logical schema decimal(12,2) stored unscaled x100
    p2 = str(tmp_path / "legacy.parquet")
    pq.write_table(pa.table({
        "k": np.arange(1000, 2000),
        "d": pa.array([decimal.Decimal(f"{i}.00")
                       for i in range(1000, 2000)],
                      type=pa.decimal128(12, 2))}), p2)
    assert read_native_layout(p2) is None
    batch, counts = read_files_batch([p1, p2])
    assert counts == [1000, 1000]
    k = batch.tensor("k").numpy()
    d = batch.tensor("d").numpy()
    assert (k == np.arange(2000)).all()
    assert (d == np.arange(2000) * 100).all()


def test_numeric_dictionary_host_decode(tmp_path):
    """Spark's default shape (dictionary-encoded numerics) decodes on
    the HOST path too: RLE indices gathered through the PLAIN
    dictionary page, incl. mid-chunk overflow and nullable columns."""
    rng = np.random.default_rng(2)
    n = 300_000
    key = rng.integers(0, 1000, n)
    big = rng.integers(0, 1 << 40, n)  # dict overflows mid-chunk
    nn = pa.array([None if i % 9 == 0 else int(i % 777)
                   for i in range(n)], type=pa.int64())
    p = str(tmp_path / "numdict.parquet")
    pq.write_table(pa.table({"key": key, "big": big, "nn": nn}), p,
                   compression="NONE", use_dictionary=True,
                   data_page_version="1.0")
    lay = read_native_layout(p)
    kinds = {c.name: c.encoding for c in lay[1]}
    assert kinds["key"] == "dict"
    cols, masks = read_native_host(p)
    assert (np.asarray(cols["key"]) == key).all()
    assert (np.asarray(cols["big"]) == big).all()
    m, v = masks["nn"], np.asarray(cols["nn"])
    for i in range(0, n, 313):
        assert (not m[i]) if i % 9 == 0 else (m[i] and v[i] == i % 777)
    batch, rc = read_files_batch([p])
    assert rc == [n]
    assert (batch.tensor("big").numpy() == big).all()
