"""Native BYTE_ARRAY (string) parquet path: dictionary-encoded write +
host decode with no pyarrow on the data path.

Reference parity: CoveringIndex supports any column type
(index/covering/CoveringIndex.scala:140-192); round 1 routed strings
through pyarrow — this pins the native path.
"""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.execution.columnar import ColumnBatch, StringColumn
from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                               write_batch_parquet)


def _string_batch(rng, n=8000, nvals=200, with_nulls=False):
    vals = sorted(f"sku-{i:04d}" for i in range(nvals))
    codes = rng.integers(0, nvals, n).astype(np.int32)
    masks = {}
    if with_nulls:
        masks["s"] = torch.from_numpy(rng.random(n) > 0.15)
    return ColumnBatch(
        {"s": StringColumn(torch.from_numpy(codes), vals),
         "k": torch.from_numpy(rng.integers(0, 100, n)),
         "v": torch.from_numpy(rng.random(n))}, masks), vals, codes


def test_native_string_roundtrip_and_pyarrow_interop(tmp_path):
    rng = np.random.default_rng(5)
    b, vals, codes = _string_batch(rng)
    p = str(tmp_path / "s.parquet")
    write_batch_parquet(b, p)
    # dictionary-encoded BYTE_ARRAY chunk, not a pyarrow fallback file
    col = pq.ParquetFile(p).metadata.row_group(0).column(0)
    assert col.physical_type == "BYTE_ARRAY"
    assert "PLAIN_DICTIONARY" in col.encodings
    # pyarrow (a foreign reader) sees identical content
    t = pq.read_table(p)
    assert t.column("s").to_pylist() == \
        np.asarray(vals, dtype=object)[codes].tolist()
    # our host reader returns the dictionary layout
    rb, _ = read_files_batch([p])
    s = rb.column("s")
    assert isinstance(s, StringColumn)
    assert s.values == vals
    assert (s.codes.numpy() == codes).all()


def test_native_string_nullable_roundtrip(tmp_path):
    rng = np.random.default_rng(6)
    b, vals, codes = _string_batch(rng, with_nulls=True)
    mask = b.mask("s").numpy()
    p = str(tmp_path / "sn.parquet")
    write_batch_parquet(b, p)
    exp = np.asarray(vals, dtype=object)[codes]
    exp[~mask] = None
    assert pq.read_table(p).column("s").to_pylist() == exp.tolist()
    rb, _ = read_files_batch([p])
    assert (rb.mask("s").numpy() == mask).all()
    got = rb.column("s").to_numpy()[mask]
    assert (got == exp[mask]).all()


def test_reads_pyarrow_dict_string_natively(tmp_path):
    """pyarrow-written dictionary string files decode on the native
    path (no content fallback)."""
    words = ["alpha", "beta", "gamma", "delta"]
    data = [words[i % 4] for i in range(2000)]
    p = str(tmp_path / "pa.parquet")
    pq.write_table(pa.table({"s": pa.array(data).dictionary_encode()}),
                   p, compression="NONE", use_dictionary=True,
                   data_page_version="1.0")
    rb, _ = read_files_batch([p])
    s = rb.column("s")
    assert isinstance(s, StringColumn)
    assert (s.to_numpy() == np.array(data, dtype=object)).all()
    # parquet dictionaries are insertion-ordered ("alpha, beta, gamma,
    # delta" here); StringColumn codes MUST follow lex order or every
    # order-dependent op (join merge, z-order, range compare) breaks
    assert s.values == sorted(s.values)
    import torch as _torch
    order_ok = s.codes[np.array(data, dtype=object) == "delta"]
    assert (order_ok == s.values.index("delta")).all()


def test_string_index_build_without_pyarrow_data_path(tmp_path,
                                                      monkeypatch):
    """String-keyed covering index build + filter + join with
    pq.read_table/pa write disabled: the native reader/writer must carry
    the whole hot path (pyarrow is allowed only for footer metadata)."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(7)
    src = tmp_path / "src"
    src.mkdir()
    vocab = sorted(f"sku-{i:04d}" for i in range(500))
    expected_eq = 0
    for i in range(3):
        codes = rng.integers(0, 500, 4000).astype(np.int32)
        expected_eq += int((codes == 42).sum())
        b = ColumnBatch({"sku": StringColumn(torch.from_numpy(codes),
                                             list(vocab)),
                         "v": torch.from_numpy(rng.random(4000))})
        write_batch_parquet(b, str(src / f"part-{i}.parquet"))
    dim = tmp_path / "dim"
    dim.mkdir()
    dcodes = np.arange(500, dtype=np.int32)
    write_batch_parquet(
        ColumnBatch({"sku": StringColumn(torch.from_numpy(dcodes),
                                         list(vocab)),
                     "status": torch.from_numpy(
                         rng.integers(0, 3, 500))}),
        str(dim / "part-0.parquet"))

    import pyarrow.parquet as _pq

    def _no_read(*a, **k):
        raise AssertionError("pq.read_table on the hot path")

    def _no_write(*a, **k):
        raise AssertionError("pq.write_table on the hot path")

    monkeypatch.setattr(_pq, "read_table", _no_read)
    monkeypatch.setattr(_pq, "write_table", _no_write)

    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    fact = session.read_parquet(str(src))
    dimdf = session.read_parquet(str(dim))
    h.create_index(fact, hs.CoveringIndexConfig("sfx", ["sku"], ["v"]))
    h.create_index(dimdf, hs.CoveringIndexConfig("sdx", ["sku"],
                                                 ["status"]))
    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     True)
    out = fact.filter("sku = 'sku-0042'").select("sku", "v").collect()
    assert out.num_rows == expected_eq
    j = fact.select("sku", "v").join(dimdf.select("sku", "status"),
                                     on="sku")
    assert j.collect().num_rows == 12000  # every fact row matches


def test_empty_string_column_roundtrip(tmp_path):
    b = ColumnBatch({"s": StringColumn(torch.empty(0, dtype=torch.int32),
                                       []),
                     "k": torch.empty(0, dtype=torch.int64)})
    p = str(tmp_path / "e.parquet")
    write_batch_parquet(b, p)
    rb, _ = read_files_batch([p])
    assert rb.num_rows == 0
    assert isinstance(rb.column("s"), StringColumn)


def test_compressed_string_dict_overflow_layout(tmp_path):
    """A SNAPPY string chunk whose dictionary overflows mid-chunk mixes
    dict-index and PLAIN byte-array pages; the layout records the PLAIN
    pages as splain_z (device path: decompress + host byte-array
    parse), never as numeric codes.  The HOST reader falls back to
    pyarrow for compressed chunks, content exact."""
    from hyperspace_amd.sources.native_parquet import (read_native_host,
                                                       read_native_layout)
    vals = [f"s{i:07d}" for i in range(300_000)]
    p = str(tmp_path / "ovf.parquet")
    pq.write_table(pa.table({"s": vals}), p, compression="SNAPPY",
                   use_dictionary=True,
                   dictionary_pagesize_limit=64 * 1024,
                   data_page_version="1.0")
    encs = set(pq.ParquetFile(p).metadata.row_group(0).column(0)
               .encodings)
    assert "PLAIN" in encs and encs & {"PLAIN_DICTIONARY",
                                       "RLE_DICTIONARY"}, encs
    lay = read_native_layout(p)
    assert lay is not None
    kinds = {pg[0] for c in lay[1] for pg in c.pages}
    assert kinds == {"dict_z", "splain_z"}, kinds
    assert read_native_host(p) is None  # compressed: device-only
    rb, counts = read_files_batch([p])
    assert counts == [300_000]
    got = rb.column("s")
    assert isinstance(got, StringColumn)
    assert got.to_numpy()[:3].tolist() == ["s0000000", "s0000001",
                                           "s0000002"]


def _plain_string_file(tmp_path, name, vals, **kw):
    p = str(tmp_path / name)
    pq.write_table(pa.table({"s": vals}), p, compression="NONE",
                   data_page_version="1.0", **kw)
    return p


def test_plain_string_pages_host_native(tmp_path):
    """PLAIN (non-dictionary) byte-array pages parse natively: host
    byte-array walk (parse_byte_arrays) + arrow dictionary-encode to
    the sorted StringColumn contract (reference reads strings through
    parquet-mr; here K1's splain path)."""
    from hyperspace_amd.sources.native_parquet import (read_native_host,
                                                       read_native_layout)
    vals = [f"v{i % 977:04d}" for i in range(120_000)]
    p = _plain_string_file(tmp_path, "sp.parquet", vals,
                           use_dictionary=False)
    lay = read_native_layout(p)
    assert lay is not None
    assert lay[1][0].encoding == "splain"
    cols, masks = read_native_host(p)
    s = cols["s"]
    got = [s.values[c] for c in s.codes[::997]]
    assert got == vals[::997]
    assert all(s.values[i] <= s.values[i + 1]
               for i in range(len(s.values) - 1))
    rb, counts = read_files_batch([p])
    assert counts == [120_000]
    assert rb.column("s").to_numpy()[::997].tolist() == vals[::997]


def test_mixed_dict_plain_string_chunk_host(tmp_path):
    """Uncompressed dictionary-overflow chunks (dict pages then PLAIN
    byte-array pages in ONE chunk) decode natively with per-run
    dictionary merge."""
    from hyperspace_amd.sources.native_parquet import (read_native_host,
                                                       read_native_layout)
    vals = [f"s{i:07d}" for i in range(300_000)]
    p = _plain_string_file(tmp_path, "mx.parquet", vals,
                           use_dictionary=True,
                           dictionary_pagesize_limit=64 * 1024)
    lay = read_native_layout(p)
    assert lay is not None
    kinds = {pg[0] for c in lay[1] for pg in c.pages}
    assert kinds == {"dict", "splain"}, kinds
    cols, _ = read_native_host(p)
    s = cols["s"]
    for i in (0, 1, 99_999, 100_000, 299_999):
        assert s.values[s.codes[i]] == vals[i]
    assert all(s.values[i] <= s.values[i + 1]
               for i in range(len(s.values) - 1))


def test_plain_string_pages_nullable_host(tmp_path):
    from hyperspace_amd.sources.native_parquet import read_native_host
    vals = [None if i % 7 == 0 else f"x{i % 50}" for i in range(50_000)]
    p = _plain_string_file(tmp_path, "nl.parquet", vals,
                           use_dictionary=False)
    cols, masks = read_native_host(p)
    s, m = cols["s"], masks["s"]
    assert not m[0] and m[1]
    for i in range(0, 50_000, 317):
        assert (vals[i] is None and not m[i]) or \
            s.values[s.codes[i]] == vals[i]
