"""Device snappy decompression (K1): pyarrow SNAPPY files decode on
device bit-identically to the host read; masked/dictionary snappy
chunks fall back cleanly."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from hyperspace_amd.ops import native
from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                               read_files_batch_device)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert native.available()


def test_snappy_kernel_roundtrip_vs_codec(tmp_path):
    # raw-block compress with pyarrow's codec, decompress with the kernel
    rng = np.random.default_rng(3)
    parts = []
    for i in range(6):
        if i % 2:  # compressible: repeated patterns -> copies
            parts.append(np.tile(rng.integers(0, 255, 50,
                                              dtype=np.uint8), 997))
        else:      # incompressible: literals
            parts.append(rng.integers(0, 255, 40_000, dtype=np.uint8))
    codec = pa.Codec("snappy")
    comp = [codec.compress(p.tobytes()).to_pybytes() for p in parts]
    blob = b"".join(comp)
    dev = torch.device("cuda:0")
    src = torch.frombuffer(bytearray(blob), dtype=torch.uint8).to(dev)
    offs = np.concatenate([[0], np.cumsum([len(c) for c in comp])])
    uncs = [len(p) for p in parts]
    d_off = np.concatenate([[0], np.cumsum(uncs)])
    dst = torch.empty(int(d_off[-1]) + 4, dtype=torch.uint8, device=dev)
    ext = native.ext()
    st = ext.snappy_decompress(
        src, torch.tensor(offs[:-1]), torch.tensor(offs[1:]), dst,
        torch.tensor(d_off[:-1]), torch.tensor(uncs))
    torch.cuda.synchronize()
    assert (st == 0).all(), st.cpu()
    got = dst[:int(d_off[-1])].cpu().numpy()
    want = np.concatenate(parts)
    assert np.array_equal(got, want)


def test_snappy_parquet_device_read(tmp_path):
    rng = np.random.default_rng(4)
    # compressible int64s (small range -> back-references) + doubles
    key = rng.integers(0, 50, 500_000)
    val = rng.random(500_000)
    paths = []
    for i in range(2):
        p = str(tmp_path / f"s{i}.parquet")
        pq.write_table(pa.table({"key": key, "val": val}), p,
                       compression="SNAPPY", use_dictionary=False,
                       data_page_version="1.0")
        paths.append(p)
    dev_batch, counts = read_files_batch_device(
        paths, torch.device("cuda:0"))
    host_batch, hcounts = read_files_batch(paths)
    assert counts == hcounts == [500_000, 500_000]
    for cname in ("key", "val"):
        assert torch.equal(dev_batch.tensor(cname).cpu(),
                           host_batch.tensor(cname)), cname


def test_snappy_nullable_falls_back(tmp_path):
    rng = np.random.default_rng(5)
    vals = rng.integers(0, 9, 10_000)
    mask = rng.random(10_000) > 0.3
    p = str(tmp_path / "n.parquet")
    pq.write_table(pa.table({"k": pa.array(vals, mask=~mask)}), p,
                   compression="SNAPPY", use_dictionary=False)
    batch, counts = read_files_batch_device([p], torch.device("cuda:0"))
    assert counts == [10_000]
    m = batch.mask("k")
    assert m is not None and int((~m.cpu()).sum()) == int((~mask).sum())


def test_snappy_dictionary_device_read(tmp_path):
    # Spark's default output shape: SNAPPY + dictionary encoding
    rng = np.random.default_rng(6)
    key = rng.integers(0, 1000, 400_000)   # dict-encodable
    val = rng.random(400_000)              # dict overflows -> mixed
    paths = []
    for i in range(2):
        p = str(tmp_path / f"sd{i}.parquet")
        pq.write_table(pa.table({"key": key, "val": val}), p,
                       compression="SNAPPY", use_dictionary=True,
                       data_page_version="1.0")
        paths.append(p)
    dev_batch, counts = read_files_batch_device(
        paths, torch.device("cuda:0"))
    host_batch, hcounts = read_files_batch(paths)
    assert counts == hcounts == [400_000, 400_000]
    for cname in ("key", "val"):
        assert torch.equal(dev_batch.tensor(cname).cpu(),
                           host_batch.tensor(cname)), cname


def test_snappy_kernel_rejects_garbage():
    """Malformed compressed pages (corrupt files) must set a nonzero
    status without out-of-bounds access or hangs."""
    rng = np.random.default_rng(7)
    dev = torch.device("cuda:0")
    ext = native.ext()
    blobs = [rng.integers(0, 255, n, dtype=np.uint8).tobytes()
             for n in (1, 7, 300, 5000)]
    # also a truncated VALID stream
    codec = pa.Codec("snappy")
    good = codec.compress(np.arange(10000, dtype=np.uint8)
                          .tobytes()).to_pybytes()
    blobs.append(good[: len(good) // 2])
    blob = b"".join(blobs)
    src = torch.frombuffer(bytearray(blob), dtype=torch.uint8).to(dev)
    offs = np.concatenate([[0], np.cumsum([len(b) for b in blobs])])
    uncs = [64, 64, 1024, 16384, 10000]
    d_off = np.concatenate([[0], np.cumsum(uncs)])
    dst = torch.empty(int(d_off[-1]) + 4, dtype=torch.uint8, device=dev)
    st = ext.snappy_decompress(
        src, torch.tensor(offs[:-1]), torch.tensor(offs[1:]), dst,
        torch.tensor(d_off[:-1]), torch.tensor(uncs))
    torch.cuda.synchronize()
    assert (st.cpu() != 0).all(), st.cpu()


def test_corrupt_snappy_file_falls_back(tmp_path):
    """A parquet file claiming SNAPPY whose page bytes are corrupt must
    fall back to the host read's error, not crash the device path."""
    rng = np.random.default_rng(8)
    p = str(tmp_path / "c.parquet")
    pq.write_table(pa.table({"k": rng.integers(0, 9, 50_000)}), p,
                   compression="SNAPPY", use_dictionary=False,
                   data_page_version="1.0")
    data = bytearray(open(p, "rb").read())
    # stomp bytes in the middle of the page payload region
    for i in range(200, 1200):
        data[i] ^= 0xFF
    open(p, "wb").write(bytes(data))
    try:
        batch, counts = read_files_batch_device([p],
                                                torch.device("cuda:0"))
        # if pyarrow tolerated it, fine — just require no crash
    except Exception:
        pass  # host fallback raising on corrupt data is acceptable


def test_snappy_string_dictionary_device_read(tmp_path):
    """Snappy-compressed dictionary STRING chunks decode on device: the
    dictionary page D2Hs once for value parsing, the RLE index pages
    become codes in HBM; content equals the pyarrow read."""
    rng = np.random.default_rng(17)
    words = sorted(f"w{i:04d}" for i in range(800))
    data = [words[i] for i in rng.integers(0, 800, 400_000)]
    p = str(tmp_path / "sdz.parquet")
    pq.write_table(pa.table({"s": pa.array(data).dictionary_encode(),
                             "k": rng.integers(0, 99, 400_000)}),
                   p, compression="SNAPPY", use_dictionary=True,
                   data_page_version="1.0")
    from hyperspace_amd.execution.columnar import StringColumn
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    assert s.codes.device.type == "cuda"
    got = s.to_numpy()
    assert (got == np.array(data, dtype=object)).all()
    assert int(batch.tensor("k").sum()) == int(
        pq.read_table(p).column("k").to_numpy().sum())


def test_snappy_mixed_dict_plain_chunk(tmp_path):
    """pyarrow's dictionary overflow mid-chunk produces MIXED dict+PLAIN
    snappy pages in one column chunk; the batched fast path must decode
    both kinds with exact content (the incremental-refresh file shape)."""
    rng = np.random.default_rng(23)
    n = 2_000_000
    key = rng.integers(0, 1 << 24, n)  # huge keyspace -> dict overflow
    val = rng.random(n)
    p = str(tmp_path / "mix.parquet")
    pq.write_table(pa.table({"key": key, "val": val}), p,
                   compression="SNAPPY", use_dictionary=True,
                   dictionary_pagesize_limit=64 * 1024,
                   data_page_version="1.0")
    encs = set(pq.ParquetFile(p).metadata.row_group(0).column(0)
               .encodings)
    assert "PLAIN" in encs and encs & {"PLAIN_DICTIONARY",
                                       "RLE_DICTIONARY"}, encs
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [n]
    assert (batch.tensor("key").cpu().numpy() == key).all()
    assert np.allclose(batch.tensor("val").cpu().numpy(), val)


@pytest.mark.parametrize("codec", ["zstd", "gzip", "lz4"])
def test_zstd_gzip_pages_host_codec_path(tmp_path, codec):
    """ZSTD/GZIP/LZ4 pages keep the native page-assembly path: host codec
    threads decompress into device scratch, everything downstream
    (copy_unaligned, dict decode) is unchanged."""
    rng = np.random.default_rng(29)
    n = 1_500_000
    key = rng.integers(0, 10_000, n)
    val = rng.random(n)
    p = str(tmp_path / f"{codec}.parquet")
    pq.write_table(pa.table({"key": key, "val": val}), p,
                   compression=codec.upper(), use_dictionary=True,
                   data_page_version="1.0")
    from hyperspace_amd.sources import parquet_io as pio
    orig = pio.read_files_batch
    called = []

    def spy(*a, **k):
        called.append(1)
        return orig(*a, **k)
    pio.read_files_batch = spy
    try:
        batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    finally:
        pio.read_files_batch = orig
    assert not called, "fell back to the host table path"
    assert rc == [n]
    assert (batch.tensor("key").cpu().numpy() == key).all()
    assert np.allclose(batch.tensor("val").cpu().numpy(), val)


def test_plain_string_pages_device_read(tmp_path):
    """PLAIN (non-dictionary) byte-array string chunks decode through
    the native device path: host parse_byte_arrays + arrow dict-encode,
    codes uploaded; content equals the written values."""
    from hyperspace_amd.execution.columnar import StringColumn
    rng = np.random.default_rng(31)
    vals = [f"v{i % 977:04d}" for i in rng.integers(0, 1 << 30, 500_000)]
    p = str(tmp_path / "spd.parquet")
    pq.write_table(pa.table({"s": vals, "k": np.arange(500_000)}), p,
                   compression="NONE", use_dictionary=False,
                   data_page_version="1.0")
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [500_000]
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    assert s.codes.device.type == "cuda"
    assert (s.to_numpy() == np.array(vals, dtype=object)).all()
    assert (batch.tensor("k").cpu().numpy() == np.arange(500_000)).all()


def test_mixed_dict_plain_string_chunk_device_read(tmp_path):
    """Uncompressed dictionary-overflow STRING chunks (dict + splain
    pages in one chunk) decode on device with per-run dictionary
    entries merged to one sorted dictionary."""
    from hyperspace_amd.execution.columnar import StringColumn
    vals = [f"s{i:07d}" for i in range(300_000)]
    p = str(tmp_path / "mxd.parquet")
    pq.write_table(pa.table({"s": vals}), p, compression="NONE",
                   use_dictionary=True,
                   dictionary_pagesize_limit=64 * 1024,
                   data_page_version="1.0")
    from hyperspace_amd.sources.native_parquet import read_native_layout
    lay = read_native_layout(p)
    kinds = {pg[0] for c in lay[1] for pg in c.pages}
    assert kinds == {"dict", "splain"}, kinds
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [300_000]
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    assert (s.to_numpy() == np.array(vals, dtype=object)).all()


def test_plain_string_nullable_device_read(tmp_path):
    from hyperspace_amd.execution.columnar import StringColumn
    vals = [None if i % 7 == 0 else f"x{i % 50}" for i in range(200_000)]
    p = str(tmp_path / "nld.parquet")
    pq.write_table(pa.table({"s": vals}), p, compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [200_000]
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    m = batch.mask("s").cpu().numpy()
    assert not m[0] and m[1]
    got = s.to_numpy()
    for i in range(0, 200_000, 317):
        assert (vals[i] is None and not m[i]) or got[i] == vals[i]


def test_snappy_string_dict_overflow_device_read(tmp_path):
    """SNAPPY string chunks with mid-chunk dictionary overflow (dict_z
    + splain_z pages in one chunk) decode natively: dict-index pages
    as codes, PLAIN byte-array pages decompressed then host-parsed."""
    from hyperspace_amd.execution.columnar import StringColumn
    vals = [f"s{i:07d}" for i in range(300_000)]
    p = str(tmp_path / "ovfz.parquet")
    pq.write_table(pa.table({"s": vals}), p, compression="SNAPPY",
                   use_dictionary=True,
                   dictionary_pagesize_limit=64 * 1024,
                   data_page_version="1.0")
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [300_000]
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    assert (s.to_numpy() == np.array(vals, dtype=object)).all()


def test_snappy_plain_string_chunk_device_read(tmp_path):
    """Whole-chunk SNAPPY PLAIN (non-dictionary) strings: pure splain_z
    chunks decode through decompression + host byte-array parse."""
    from hyperspace_amd.execution.columnar import StringColumn
    rng = np.random.default_rng(37)
    vals = [f"w{i % 1251:05d}" for i in rng.integers(0, 1 << 30,
                                                     400_000)]
    p = str(tmp_path / "spz.parquet")
    pq.write_table(pa.table({"s": vals, "k": np.arange(400_000)}), p,
                   compression="SNAPPY", use_dictionary=False,
                   data_page_version="1.0")
    from hyperspace_amd.sources.native_parquet import read_native_layout
    lay = read_native_layout(p)
    assert lay is not None
    assert any(c.encoding == "splain_z" for c in lay[1])
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [400_000]
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    assert (s.to_numpy() == np.array(vals, dtype=object)).all()
    assert (batch.tensor("k").cpu().numpy() == np.arange(400_000)).all()


def test_snappy_plain_string_nullable_device_read(tmp_path):
    from hyperspace_amd.execution.columnar import StringColumn
    vals = [None if i % 5 == 0 else f"x{i % 40}" for i in range(150_000)]
    p = str(tmp_path / "spzn.parquet")
    pq.write_table(pa.table({"s": vals}), p, compression="SNAPPY",
                   use_dictionary=False, data_page_version="1.0")
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [150_000]
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    m = batch.mask("s").cpu().numpy()
    assert not m[0] and m[1]
    got = s.to_numpy()
    for i in range(0, 150_000, 311):
        assert (vals[i] is None and not m[i]) or got[i] == vals[i]


def test_data_page_v2_snappy_device_read(tmp_path):
    """Compressed DataPageV2 chunks (levels uncompressed in-file, data
    sections per-page compressed or identity) decode on device with
    exact content: numerics, nullable and dictionary strings."""
    from hyperspace_amd.execution.columnar import StringColumn
    rng = np.random.default_rng(43)
    n = 600_000
    key = rng.integers(0, 5000, n)
    words = sorted(f"w{i:04d}" for i in range(700))
    sv = [words[i] for i in rng.integers(0, 700, n)]
    nn = [None if i % 13 == 0 else int(i) for i in range(n)]
    p = str(tmp_path / "v2z.parquet")
    pq.write_table(pa.table({
        "k": key, "s": sv,
        "nn": pa.array(nn, type=pa.int64())}), p,
        compression="SNAPPY", use_dictionary=True,
        data_page_version="2.0")
    from hyperspace_amd.sources.native_parquet import read_native_layout
    lay = read_native_layout(p)
    assert lay is not None
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [n]
    assert (batch.tensor("k").cpu().numpy() == key).all()
    s = batch.column("s")
    assert isinstance(s, StringColumn)
    assert (s.to_numpy() == np.array(sv, dtype=object)).all()
    m = batch.mask("nn").cpu().numpy()
    assert not m[0] and m[1]
    got = batch.tensor("nn").cpu().numpy()
    for i in range(0, n, 431):
        assert (nn[i] is None and not m[i]) or got[i] == nn[i]


def test_mixed_native_and_host_files_device_read(tmp_path):
    """A device batch mixing native files with a DELTA_BINARY_PACKED
    file: the exotic file pyarrow-reads on a worker thread into the
    shared preallocated tensors while its neighbors decode on device."""
    rng = np.random.default_rng(53)
    n = 200_000
    paths = []
    all_k, all_v = [], []
    for i in range(3):
        k = rng.integers(0, 10_000, n)
        v = rng.random(n)
        p = str(tmp_path / f"m{i}.parquet")
        if i == 1:  # exotic encoding: no native layout
            pq.write_table(pa.table({"k": k, "v": v}), p,
                           use_dictionary=False, compression="NONE",
                           column_encoding={"k": "DELTA_BINARY_PACKED",
                                            "v": "PLAIN"})
            from hyperspace_amd.sources.native_parquet import \
                read_native_layout
            assert read_native_layout(p) is None
        else:
            pq.write_table(pa.table({"k": k, "v": v}), p,
                           compression="SNAPPY", use_dictionary=True,
                           data_page_version="1.0")
        paths.append(p)
        all_k.append(k)
        all_v.append(v)
    batch, rc = read_files_batch_device(paths, torch.device("cuda:0"))
    assert rc == [n, n, n]
    assert (batch.tensor("k").cpu().numpy()
            == np.concatenate(all_k)).all()
    assert np.allclose(batch.tensor("v").cpu().numpy(),
                       np.concatenate(all_v))


def test_flba_decimal_file_in_device_batch(tmp_path):
    """FLBA-physical decimals (pyarrow's default decimal128 layout)
    ride the per-file host fallback inside a device batch: unscaled
    int64 in the shared tensor, native neighbors unaffected."""
    import decimal
    rng = np.random.default_rng(59)
    n = 100_000
    p1 = str(tmp_path / "native.parquet")
    pq.write_table(pa.table({"k": np.arange(n),
                             "d": np.arange(n) * 100}), p1,
                   compression="SNAPPY", use_dictionary=True,
                   data_page_version="1.0")
    p2 = str(tmp_path / "flba.parquet")
    pq.write_table(pa.table({
        "k": np.arange(n, 2 * n),
        "d": pa.array([decimal.Decimal(f"{i}.00")
                       for i in range(n, 2 * n)],
                      type=pa.decimal128(12, 2))}), p2)
    batch, rc = read_files_batch_device([p1, p2],
                                        torch.device("cuda:0"))
    assert rc == [n, n]
    assert (batch.tensor("k").cpu().numpy() == np.arange(2 * n)).all()
    assert (batch.tensor("d").cpu().numpy()
            == np.arange(2 * n) * 100).all()


def test_nullable_struct_leaves_snappy_device_read(tmp_path):
    """Compressed nullable-struct leaves (2-bit def levels) decode on
    device with parent-null propagation."""
    n = 300_000
    structs = [None if i % 13 == 0 else
               {"x": (None if i % 7 == 0 else i), "y": float(i)}
               for i in range(n)]
    t = pa.table({"s": pa.array(structs,
                                type=pa.struct([("x", pa.int64()),
                                                ("y", pa.float64())])),
                  "k": np.arange(n)})
    p = str(tmp_path / "nestz.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=False,
                   data_page_version="1.0")
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    assert rc == [n]
    mx = batch.mask("s.x").cpu().numpy()
    vx = batch.tensor("s.x").cpu().numpy()
    my = batch.mask("s.y").cpu().numpy()
    vy = batch.tensor("s.y").cpu().numpy()
    for i in range(0, n, 431):
        if structs[i] is None or structs[i]["x"] is None:
            assert not mx[i], i
        else:
            assert mx[i] and vx[i] == i, i
        if structs[i] is None:
            assert not my[i], i
        else:
            assert my[i] and vy[i] == float(i), i


def test_column_projection_on_new_page_kinds(tmp_path):
    """columns= projection must hold for splain/splain_z/V2 chunks: only
    requested columns decode and the content stays exact."""
    rng = np.random.default_rng(67)
    n = 200_000
    vals = [f"p{i % 301:03d}" for i in range(n)]
    p1 = str(tmp_path / "a.parquet")
    pq.write_table(pa.table({"s": vals, "k": np.arange(n),
                             "drop": rng.random(n)}), p1,
                   compression="SNAPPY", use_dictionary=False,
                   data_page_version="2.0")
    batch, rc = read_files_batch_device([p1], torch.device("cuda:0"),
                                        columns=["k", "s"])
    assert rc == [n]
    assert set(batch.columns.keys()) == {"k", "s"}
    assert (batch.tensor("k").cpu().numpy() == np.arange(n)).all()
    s = batch.column("s")
    assert (s.to_numpy()[::733] == np.array(vals[::733],
                                            dtype=object)).all()
