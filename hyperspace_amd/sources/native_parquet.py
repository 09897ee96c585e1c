"""Native Parquet writer/reader for the device pipeline (K1/K3).

The hot path writes index data and synthetic sources as uncompressed
PLAIN pages of REQUIRED numeric columns: the page payload is then exactly
the little-endian column buffer, so
  - the write side assembles [PAR1][pages][footer][len][PAR1] with a
    minimal Thrift compact-protocol encoder and hands large contiguous
    buffers to os.write (no per-row work, no pyarrow overhead), and
  - the read side locates page payloads (pyarrow footer metadata + a tiny
    Thrift PageHeader parse) and decodes ON DEVICE with the
    unaligned-copy kernel — bytes go disk -> pinned host -> HBM,
    untouched by the CPU.

Files are standard Parquet: pyarrow reads them (tests assert equality),
column-chunk statistics (min/max) are written so the z-order stats
pruning works.  Strings/nullable columns fall back to the pyarrow path.

Parquet format reference: github.com/apache/parquet-format
(Thrift compact protocol; REQUIRED fields carry no def/rep levels).
"""

from __future__ import annotations

import os
import struct
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

MAGIC = b"PAR1"

# parquet physical types
T_INT32, T_INT64, T_FLOAT, T_DOUBLE, T_BYTE_ARRAY = 1, 2, 4, 5, 6
_NP_TO_PARQUET = {
    np.dtype("int64"): T_INT64,
    np.dtype("int32"): T_INT32,
    np.dtype("float64"): T_DOUBLE,
    np.dtype("float32"): T_FLOAT,
}
_PARQUET_TO_NP = {v: k for k, v in _NP_TO_PARQUET.items()}

ENC_PLAIN = 0
ENC_PLAIN_DICTIONARY = 2
ENC_RLE = 3
ENC_RLE_DICTIONARY = 8
CODEC_UNCOMPRESSED = 0
PAGE_DATA = 0
PAGE_DICTIONARY = 2
REP_REQUIRED = 0
REP_OPTIONAL = 1
CONV_UTF8 = 0


class StrCol:
    """Dictionary-encoded string column for the native writer/reader:
    int32 codes into a sorted list of unique utf-8 values (the in-memory
    layout of execution.columnar.StringColumn, minus torch)."""

    __slots__ = ("codes", "values")

    def __init__(self, codes: np.ndarray, values: List[str]):
        self.codes = np.ascontiguousarray(codes, dtype=np.int32)
        self.values = values

    def __len__(self):
        return len(self.codes)


# ---------------------------------------------------------------------------
# Thrift compact protocol
# ---------------------------------------------------------------------------

CT_STOP, CT_TRUE, CT_FALSE, CT_BYTE, CT_I16, CT_I32, CT_I64, CT_DOUBLE, \
    CT_BINARY, CT_LIST, CT_SET, CT_MAP, CT_STRUCT = range(13)


def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _zigzag(n: int) -> int:
    return (n << 1) ^ (n >> 63)


class TWriter:
    """Struct-oriented Thrift compact writer."""

    def __init__(self):
        self.buf = bytearray()
        self._last_fid = [0]

    # struct nesting
    def struct_begin(self):
        self._last_fid.append(0)

    def struct_end(self):
        self.buf.append(0x00)
        self._last_fid.pop()

    def _field(self, fid: int, ctype: int):
        delta = fid - self._last_fid[-1]
        if 0 < delta <= 15:
            self.buf.append((delta << 4) | ctype)
        else:
            self.buf.append(ctype)
            self.buf += _varint(_zigzag(fid))
        self._last_fid[-1] = fid

    def field_i32(self, fid: int, v: int):
        self._field(fid, CT_I32)
        self.buf += _varint(_zigzag(v))

    def field_i64(self, fid: int, v: int):
        self._field(fid, CT_I64)
        self.buf += _varint(_zigzag(v))

    def field_binary(self, fid: int, data: bytes):
        self._field(fid, CT_BINARY)
        self.buf += _varint(len(data))
        self.buf += data

    def field_string(self, fid: int, s: str):
        self.field_binary(fid, s.encode("utf-8"))

    def field_list_begin(self, fid: int, etype: int, size: int):
        self._field(fid, CT_LIST)
        if size < 15:
            self.buf.append((size << 4) | etype)
        else:
            self.buf.append(0xF0 | etype)
            self.buf += _varint(size)

    def field_struct_begin(self, fid: int):
        self._field(fid, CT_STRUCT)
        self.struct_begin()

    def i32_elem(self, v: int):
        self.buf += _varint(_zigzag(v))

    def struct_elem_begin(self):
        self.struct_begin()


class TReader:
    """Minimal Thrift compact reader (PageHeader subset)."""

    def __init__(self, data: bytes, pos: int = 0):
        self.data = data
        self.pos = pos

    def _read_varint(self) -> int:
        out = 0
        shift = 0
        while True:
            b = self.data[self.pos]
            self.pos += 1
            out |= (b & 0x7F) << shift
            if not b & 0x80:
                return out
            shift += 7

    def _read_zigzag(self) -> int:
        n = self._read_varint()
        return (n >> 1) ^ -(n & 1)

    def read_struct(self) -> Dict[int, object]:
        """Parse one struct into {field id: value}; nested structs become
        dicts, lists become lists."""
        out: Dict[int, object] = {}
        last_fid = 0
        while True:
            b = self.data[self.pos]
            self.pos += 1
            if b == 0:
                return out
            delta = b >> 4
            ctype = b & 0x0F
            fid = (last_fid + delta) if delta else self._read_zigzag()
            last_fid = fid
            out[fid] = self._read_value(ctype)

    def _read_value(self, ctype: int):
        if ctype in (CT_TRUE, CT_FALSE):
            return ctype == CT_TRUE
        if ctype in (CT_BYTE,):
            v = self.data[self.pos]
            self.pos += 1
            return v
        if ctype in (CT_I16, CT_I32, CT_I64):
            return self._read_zigzag()
        if ctype == CT_DOUBLE:
            v = struct.unpack_from("<d", self.data, self.pos)[0]
            self.pos += 8
            return v
        if ctype == CT_BINARY:
            n = self._read_varint()
            v = self.data[self.pos:self.pos + n]
            self.pos += n
            return v
        if ctype == CT_LIST:
            h = self.data[self.pos]
            self.pos += 1
            size = h >> 4
            etype = h & 0x0F
            if size == 15:
                size = self._read_varint()
            return [self._read_value(etype) for _ in range(size)]
        if ctype == CT_STRUCT:
            return self.read_struct()
        raise ValueError(f"unsupported compact type {ctype}")


# ---------------------------------------------------------------------------
# Writer
# ---------------------------------------------------------------------------

def _page_header(num_values: int, nbytes: int,
                 has_def_levels: bool = False,
                 encoding: int = ENC_PLAIN) -> bytes:
    w = TWriter()
    w.struct_begin()
    w.field_i32(1, PAGE_DATA)
    w.field_i32(2, nbytes)
    w.field_i32(3, nbytes)
    w.field_struct_begin(5)  # DataPageHeader
    w.field_i32(1, num_values)
    w.field_i32(2, encoding)
    # definition_level_encoding: RLE for OPTIONAL columns
    w.field_i32(3, ENC_RLE if has_def_levels else ENC_PLAIN)
    w.field_i32(4, ENC_PLAIN)  # repetition_level_encoding
    w.struct_end()
    w.struct_end()
    return bytes(w.buf)


def _dict_page_header(num_values: int, nbytes: int) -> bytes:
    w = TWriter()
    w.struct_begin()
    w.field_i32(1, PAGE_DICTIONARY)
    w.field_i32(2, nbytes)
    w.field_i32(3, nbytes)
    w.field_struct_begin(7)  # DictionaryPageHeader
    w.field_i32(1, num_values)
    w.field_i32(2, ENC_PLAIN_DICTIONARY)
    w.struct_end()
    w.struct_end()
    return bytes(w.buf)


def _bitpack(vals: np.ndarray, width: int) -> bytes:
    """Bit-pack values LSB-first at ``width`` bits each (the RLE-hybrid
    bit-packed run layout; values padded to a multiple of 8)."""
    n = len(vals)
    pad = (-n) % 8
    if pad:
        vals = np.concatenate(
            [vals, np.zeros(pad, dtype=vals.dtype)])
    bits = ((vals[:, None].astype(np.uint64)
             >> np.arange(width, dtype=np.uint64)) & 1).astype(np.uint8)
    return np.packbits(bits.reshape(-1), bitorder="little").tobytes()


def _dict_indices_payload(codes: np.ndarray, num_dict: int) -> Tuple[
        bytes, int]:
    """[u8 bit-width][one bit-packed RLE-hybrid run] + the bit width."""
    bw = max(1, (num_dict - 1).bit_length()) if num_dict > 1 else 1
    groups = (len(codes) + 7) // 8
    body = _varint((groups << 1) | 1) + _bitpack(codes, bw)
    return bytes([bw]) + body, bw


def _plain_byte_array_dict(values: List[str]) -> bytes:
    """PLAIN-encoded BYTE_ARRAY dictionary page payload."""
    out = bytearray()
    for v in values:
        b = v.encode("utf-8")
        out += struct.pack("<I", len(b))
        out += b
    return bytes(out)


def _statistics(arr: np.ndarray) -> Tuple[bytes, bytes]:
    """(min_value, max_value) PLAIN-encoded."""
    return arr.min().tobytes(), arr.max().tobytes()


def _def_levels_payload(mask: np.ndarray) -> bytes:
    """4-byte-length-prefixed RLE-hybrid encoding of max_def=1 levels:
    one bit-packed run (groups of 8, LSB-first) covering all rows."""
    groups = (len(mask) + 7) // 8
    packed = np.packbits(mask.astype(np.uint8), bitorder="little")
    body = _varint((groups << 1) | 1) + packed.tobytes()
    return struct.pack("<I", len(body)) + body


# Layout cache: (path, size, mtime_ns) -> (num_rows, [ColumnChunkLayout]).
# The native writer knows the page layout it just wrote, so the first
# device read of a freshly built index skips the footer + page-header
# parse entirely (the reference relies on Spark's FileIndex/footer
# caching for the same effect).  Bounded FIFO.
_LAYOUT_CACHE: "Dict[str, Tuple[Tuple[int, int], int, list]]" = {}
_LAYOUT_CACHE_MAX = 16384


def layout_cache_get(path: str):
    """(num_rows, layouts) if the cached entry matches the file's current
    size+mtime, else None."""
    ent = _LAYOUT_CACHE.get(path)
    if ent is None:
        return None
    try:
        st = os.stat(path)
    except OSError:
        return None
    if ent[0] != (st.st_size, st.st_mtime_ns):
        _LAYOUT_CACHE.pop(path, None)
        return None
    return ent[1], ent[2]


def _layout_cache_put(path: str, num_rows: int, layouts: list) -> None:
    if len(_LAYOUT_CACHE) >= _LAYOUT_CACHE_MAX:
        # FIFO evict ~1/8th
        for k in list(_LAYOUT_CACHE)[:_LAYOUT_CACHE_MAX // 8]:
            _LAYOUT_CACHE.pop(k, None)
    st = os.stat(path)
    _LAYOUT_CACHE[path] = ((st.st_size, st.st_mtime_ns), num_rows, layouts)


def write_parquet_native(columns: "Dict[str, np.ndarray]", path: str,
                         masks: "Optional[Dict[str, np.ndarray]]" = None
                         ) -> Tuple[int, int]:
    """Write numeric columns as one row group, one PLAIN page per column.
    ``masks``: optional name -> bool validity array; masked columns are
    written OPTIONAL with RLE def-levels and compacted values.
    Returns (size, mtime_ms)."""
    names = list(columns.keys())
    arrays: List[Any] = []
    for n in names:
        c = columns[n]
        arrays.append(c if isinstance(c, StrCol)
                      else np.ascontiguousarray(c))
    num_rows = len(arrays[0]) if arrays else 0
    col_masks: List[Optional[np.ndarray]] = []
    for n, a in zip(names, arrays):
        if not isinstance(a, StrCol) and a.dtype not in _NP_TO_PARQUET:
            raise ValueError(f"dtype {a.dtype} not supported natively")
        assert len(a) == num_rows
        m = (masks or {}).get(n)
        if m is not None and bool(m.all()):
            m = None  # all-valid mask: write the cheaper REQUIRED form
        col_masks.append(m)

    chunks: List[Any] = []  # bytes | memoryview (os.write takes both)
    col_meta: List[Tuple] = []
    layouts: List[ColumnChunkLayout] = []
    offset = 4  # after magic
    for name, arr, mask in zip(names, arrays, col_masks):
        if isinstance(arr, StrCol):
            # dictionary-encoded BYTE_ARRAY chunk: PLAIN dictionary page
            # + one bit-packed RLE_DICTIONARY-style data page — the
            # on-disk twin of the in-memory StringColumn (codes travel,
            # values parse once)
            dict_payload = _plain_byte_array_dict(arr.values)
            dict_header = _dict_page_header(len(arr.values),
                                            len(dict_payload))
            dict_off = offset
            offset += len(dict_header) + len(dict_payload)
            chunks.append(dict_header)
            chunks.append(dict_payload)

            codes = arr.codes if mask is None else arr.codes[mask]
            null_count = int(num_rows - len(codes))
            idx_payload, bw = _dict_indices_payload(codes,
                                                    len(arr.values))
            lvl_len = 0
            if mask is None:
                payload: Any = idx_payload
            else:
                levels = _def_levels_payload(mask)
                lvl_len = len(levels)
                payload = levels + idx_payload
            header = _page_header(num_rows, len(payload), mask is not None,
                                  encoding=ENC_PLAIN_DICTIONARY)
            col_meta.append((name, T_BYTE_ARRAY, offset,
                             offset - dict_off + len(header) + len(payload),
                             num_rows, b"", b"", null_count,
                             mask is not None, dict_off))
            layouts.append(ColumnChunkLayout(
                name, np.dtype("int32"),
                [("dict", offset + len(header) + lvl_len + 1,
                  offset + len(header) + len(payload), num_rows, bw)],
                num_rows, "dict",
                (dict_off + len(dict_header), len(arr.values)), [mask],
                is_string=True, str_values=list(arr.values)))
            chunks.append(header)
            chunks.append(payload)
            offset += len(header) + len(payload)
            continue
        lvl_len = 0
        if mask is None:
            # zero-copy: the PLAIN payload IS the little-endian column
            # buffer; write straight from the array's memory
            payload = arr.data.cast("B")
            null_count = 0
            valid = arr
        else:
            valid = arr[mask]
            levels = _def_levels_payload(mask)
            lvl_len = len(levels)
            payload = levels + valid.tobytes()
            null_count = int(num_rows - len(valid))
        nbytes = len(payload)
        header = _page_header(num_rows, nbytes, mask is not None)
        mn, mx = _statistics(valid) if len(valid) else (b"", b"")
        col_meta.append((name, _NP_TO_PARQUET[arr.dtype], offset,
                         len(header) + nbytes, num_rows, mn, mx,
                         null_count, mask is not None, None))
        layouts.append(ColumnChunkLayout(
            name, arr.dtype,
            [("plain", offset + len(header) + lvl_len, num_rows)],
            num_rows, "plain", None, [mask]))
        chunks.append(header)
        chunks.append(payload)
        offset += len(header) + nbytes

    # FileMetaData
    w = TWriter()
    w.struct_begin()
    w.field_i32(1, 2)  # version
    # schema: root + fields
    w.field_list_begin(2, CT_STRUCT, 1 + len(names))
    w.struct_elem_begin()  # root
    w.field_string(4, "schema")
    w.field_i32(5, len(names))
    w.struct_end()
    for name, ptype, off, nbytes, nvals, mn, mx, null_count, optional, \
            dict_off in col_meta:
        w.struct_elem_begin()
        w.field_i32(1, ptype)
        w.field_i32(3, REP_OPTIONAL if optional else REP_REQUIRED)
        w.field_string(4, name)
        if ptype == T_BYTE_ARRAY:
            w.field_i32(6, CONV_UTF8)  # converted_type: UTF8
        w.struct_end()
    w.field_i64(3, num_rows)
    # row_groups
    w.field_list_begin(4, CT_STRUCT, 1)
    w.struct_elem_begin()
    total_bytes = sum(m[3] for m in col_meta)
    w.field_list_begin(1, CT_STRUCT, len(col_meta))
    for name, ptype, off, nbytes, nvals, mn, mx, null_count, optional, \
            dict_off in col_meta:
        w.struct_elem_begin()  # ColumnChunk
        w.field_i64(2, dict_off if dict_off is not None else off)
        w.field_struct_begin(3)  # ColumnMetaData
        w.field_i32(1, ptype)
        encs = ([ENC_PLAIN_DICTIONARY, ENC_RLE]
                if dict_off is not None
                else ([ENC_PLAIN, ENC_RLE] if optional
                      else [ENC_PLAIN]))
        w.field_list_begin(2, CT_I32, len(encs))
        for e in encs:
            w.i32_elem(e)
        w.field_list_begin(3, CT_BINARY, 1)
        w.buf += _varint(len(name.encode()))
        w.buf += name.encode()
        w.field_i32(4, CODEC_UNCOMPRESSED)
        w.field_i64(5, nvals)
        w.field_i64(6, nbytes)
        w.field_i64(7, nbytes)
        w.field_i64(9, off)  # data_page_offset
        if dict_off is not None:
            w.field_i64(11, dict_off)  # dictionary_page_offset
        if mn or null_count:
            w.field_struct_begin(12)  # Statistics
            if mn:
                w.field_binary(1, mx)  # max (legacy)
                w.field_binary(2, mn)  # min (legacy)
            w.field_i64(3, null_count)
            if mn:
                w.field_binary(5, mx)  # max_value
                w.field_binary(6, mn)  # min_value
            w.struct_end()
        w.struct_end()
        w.struct_end()
    w.field_i64(2, total_bytes)
    w.field_i64(3, num_rows)
    w.struct_end()
    w.field_string(6, "hyperspace_amd 0.1")
    w.struct_end()
    footer = bytes(w.buf)

    with open(path, "wb") as f:
        f.write(MAGIC)
        for c in chunks:
            f.write(c)
        f.write(footer)
        f.write(struct.pack("<I", len(footer)))
        f.write(MAGIC)
    _layout_cache_put(path, num_rows, layouts)
    st = os.stat(path)
    return st.st_size, int(st.st_mtime * 1000)


# ---------------------------------------------------------------------------
# Reader (page layout for the device decode path)
# ---------------------------------------------------------------------------

class ColumnChunkLayout:
    """Per-column page layout.

    encoding "plain": pages = [("plain", values byte offset, num values)]
    encoding "dict":  pages = [("dict", payload start, payload end,
                      num values, bit width), ...] of RLE/bit-packed index
                      payloads, plus dict_page = (values offset, dict size)

    ``page_masks`` parallels ``pages``: a bool validity array per page for
    OPTIONAL columns with nulls in that page, else None.  A page's stored
    values are compacted (non-null only); num values counts rows.

    ``is_string``: BYTE_ARRAY dictionary chunk — decoded values are int32
    codes into ``str_values`` (parsed lazily from the dictionary page
    when not supplied by the writer's layout cache).
    """
    __slots__ = ("name", "np_dtype", "pages", "num_values", "encoding",
                 "dict_page", "page_masks", "is_string", "str_values",
                 "codec", "max_def")

    def __init__(self, name, np_dtype, pages, num_values,
                 encoding="plain", dict_page=None, page_masks=None,
                 is_string=False, str_values=None,
                 codec="UNCOMPRESSED", max_def=1):
        self.name = name
        self.np_dtype = np_dtype
        self.pages = pages
        self.num_values = num_values
        self.encoding = encoding
        self.dict_page = dict_page
        self.page_masks = page_masks or [None] * len(pages)
        self.is_string = is_string
        self.str_values = str_values
        self.codec = codec
        self.max_def = max_def

    def dict_values(self, data) -> List[str]:
        """The string dictionary: cached from the writer, else parsed
        from the PLAIN dictionary page payload in ``data``."""
        if self.str_values is None:
            off, n = self.dict_page
            vals = []
            pos = off
            for _ in range(n):
                ln = struct.unpack_from("<I", data, pos)[0]
                pos += 4
                vals.append(bytes(data[pos:pos + ln]).decode("utf-8"))
                pos += ln
            self.str_values = vals
        return self.str_values

    @property
    def has_nulls(self) -> bool:
        return any(m is not None for m in self.page_masks)


_DEF_DECODER = None


def _native_def_decoder():
    """C++ def-level decoder when the extension is importable (the
    Python loop costs ~1 ms per page on pyarrow's many-run encodings);
    cached; None -> pure-Python fallback."""
    global _DEF_DECODER
    if _DEF_DECODER is None:
        try:
            from ..ops import native as _native
            _DEF_DECODER = (_native.ext().decode_def_levels,)
        except Exception:  # noqa: BLE001
            _DEF_DECODER = (None,)
    return _DEF_DECODER[0]


def _decode_defs(data, off: int, length: int, n: int,
                 max_def: int = 1) -> Optional[np.ndarray]:
    """Decode RLE-hybrid definition levels into a bool validity array
    (level == max_def <=> leaf present; any smaller level is a null at
    the leaf or an ancestor struct), or None when every row is valid."""
    if max_def > 1:
        # multi-bit levels (nullable-struct leaves): decode the level
        # integers with the generic RLE/bit-packed reader
        bw = max_def.bit_length()
        levels = _decode_rle_indices(data, off, off + length, n, bw)
        mask = levels == max_def
        return None if bool(mask.all()) else mask
    dec = _native_def_decoder()
    if dec is not None:
        mask = dec(data, off, length, n).numpy()
        return None if bool(mask.all()) else mask
    end = off + length
    out = np.empty(n, dtype=np.uint8)
    pos = off
    filled = 0
    while filled < n and pos < end:
        header = 0
        shift = 0
        while True:
            b = data[pos]
            pos += 1
            header |= (b & 0x7F) << shift
            if not b & 0x80:
                break
            shift += 7
        if header & 1:  # bit-packed run: (header>>1) groups of 8 levels
            groups = header >> 1
            cnt = min(groups * 8, n - filled)
            packed = np.frombuffer(data, dtype=np.uint8, count=groups,
                                   offset=pos)
            out[filled:filled + cnt] = np.unpackbits(
                packed, bitorder="little")[:cnt]
            pos += groups
            filled += cnt
        else:  # RLE run: one value byte (bit_width 1 -> 1 byte)
            run = min(header >> 1, n - filled)
            v = data[pos]
            pos += 1
            out[filled:filled + run] = v
            filled += run
    if filled < n:
        out[filled:] = 1
    mask = out.astype(bool)
    return None if bool(mask.all()) else mask


def read_native_layout(path: str,
                       columns: Optional[List[str]] = None,
                       data: Optional[bytes] = None,
                       meta=None, pf_schema=None
                       ) -> Optional[Tuple[bytes, List[ColumnChunkLayout]]]:
    """If ``path`` decodes natively (uncompressed PLAIN pages of numeric
    columns — our writer's files and pyarrow's NONE/PLAIN files), return
    (raw file bytes, per-column page layouts); else None (caller falls
    back to pyarrow).  ``data`` can supply pre-read file bytes (any
    buffer protocol object) to avoid a second disk read; ``meta`` /
    ``pf_schema`` supply already-parsed footer metadata to avoid a
    second footer parse."""
    if meta is None or pf_schema is None:
        import pyarrow.parquet as pq
        try:
            pf = pq.ParquetFile(path)
            md = pf.metadata
            pf_schema = pf.schema
        except Exception:  # noqa: BLE001
            return None
    else:
        md = meta
    if data is None:
        with open(path, "rb") as f:
            data = f.read()
    out: List[ColumnChunkLayout] = []
    want = {c.lower() for c in columns} if columns is not None else None
    # row-group-major walk: large files carry many row groups; each
    # column chunk (and its dictionary) is per-row-group
    for rg_i in range(md.num_row_groups):
        rg = md.row_group(rg_i)
        lay = _walk_row_group(rg, pf_schema, data, want)
        if lay is None:
            return None
        out.extend(lay)
    return data, out


def _walk_row_group(rg, pf_schema, data, want
                    ) -> Optional[List[ColumnChunkLayout]]:
    out: List[ColumnChunkLayout] = []
    col_index = -1
    for i in range(rg.num_columns):
        col = rg.column(i)
        col_index += 1
        name = col.path_in_schema
        if want is not None and name.lower() not in want:
            continue
        codec = col.compression.upper()
        # SNAPPY decodes on device (or host for copy-dense pages);
        # GZIP/ZSTD/BROTLI/LZ4 pages decode on host codec threads but
        # keep the native page-assembly path (no pyarrow table
        # fallback).  Parquet "LZ4"/"LZ4_RAW" pages are raw LZ4 blocks
        # (arrow writes block format for both) -> pa.Codec("lz4_raw").
        if codec not in ("UNCOMPRESSED", "SNAPPY", "GZIP", "ZSTD",
                         "BROTLI", "LZ4", "LZ4_RAW"):
            return None
        encs = set(col.encodings)
        is_dict = bool(encs & {"PLAIN_DICTIONARY", "RLE_DICTIONARY"})
        if encs - {"PLAIN", "RLE", "BIT_PACKED", "PLAIN_DICTIONARY",
                   "RLE_DICTIONARY"}:
            return None
        is_string = False
        if col.physical_type == "BYTE_ARRAY":
            # dictionary-encoded strings decode natively (codes + one
            # dictionary parse), uncompressed or compressed; PLAIN
            # byte-array pages parse host-side into codes ("splain" /
            # "splain_z" after decompression)
            is_string = True
            np_dtype = np.dtype("int32")
        else:
            ptype = {"INT64": T_INT64, "INT32": T_INT32, "FLOAT": T_FLOAT,
                     "DOUBLE": T_DOUBLE}.get(col.physical_type)
            if ptype is None:
                return None
            np_dtype = _PARQUET_TO_NP[ptype]
        # OPTIONAL columns carry a def-level prefix per page; when the
        # chunk statistics prove null_count == 0 the levels are skipped
        # without decoding, else each page's levels become a validity
        # mask.  Nested leaves (max_def > 1 or repeated) use wider level
        # encodings: pyarrow fallback.
        col_schema = pf_schema.column(col_index)
        if col_schema.max_repetition_level > 0:
            return None  # repeated (list) leaves: pyarrow
        max_def = col_schema.max_definition_level
        has_levels = max_def > 0
        st = col.statistics
        chunk_all_valid = st is not None and st.null_count == 0

        dict_page = None
        pos = col.data_page_offset
        if is_dict:
            dict_off = col.dictionary_page_offset
            if dict_off is None:
                return None
            r = TReader(data, dict_off)
            try:
                hdr = r.read_struct()
            except Exception:  # noqa: BLE001
                return None
            if hdr.get(1) != 2:  # DICTIONARY_PAGE
                return None
            dict_n = hdr.get(7, {}).get(1)
            if dict_n is None:
                return None
            if codec != "UNCOMPRESSED":
                if hdr.get(2) is None:
                    return None
                dict_page = ("z", r.pos, r.pos + hdr.get(3), dict_n,
                             hdr.get(2))
            else:
                dict_page = (r.pos, dict_n)
            if dict_off >= pos:
                # dictionary physically precedes data pages
                pos = r.pos + hdr.get(3)

        seen = 0
        pages: List[Tuple] = []
        page_masks: List[Optional[np.ndarray]] = []
        while seen < col.num_values:
            r = TReader(data, pos)
            try:
                hdr = r.read_struct()
            except Exception:  # noqa: BLE001
                return None
            page_bytes = hdr.get(3)
            if hdr.get(1) == 2:  # stray dictionary page: skip
                pos = r.pos + page_bytes
                continue
            if hdr.get(1) == 3:  # DataPageV2
                # levels are stored UNCOMPRESSED at the payload start
                # with a KNOWN byte length (no 4-byte prefix); only the
                # data section after them is subject to the codec
                # (parquet-format PageHeader.data_page_header_v2)
                d2 = hdr.get(8, {})
                num_values = d2.get(1)
                dl_len = d2.get(5, 0) or 0
                rl_len = d2.get(6, 0) or 0
                if num_values is None or rl_len:
                    return None  # repeated leaves: pyarrow
                enc2 = d2.get(4)
                is_comp = codec != "UNCOMPRESSED" and \
                    d2.get(7, True) is not False
                values_off = r.pos
                page_end = r.pos + page_bytes
                mask = None
                if dl_len:
                    if not chunk_all_valid:
                        mask = _decode_defs(data, values_off, dl_len,
                                            num_values, max_def)
                    values_off += dl_len
                n_valid = int(mask.sum()) if mask is not None \
                    else num_values
                if codec != "UNCOMPRESSED":
                    # under a codec, every V2 page becomes a z page so
                    # the chunk stays uniform (the dictionary page is
                    # always compressed); per-page is_compressed=false
                    # pages carry is_comp=False (7th element) and are
                    # identity-copied into scratch at decode
                    if is_comp:
                        unc = hdr.get(2)
                        if unc is None:
                            return None
                        unc_data = unc - dl_len - rl_len
                    else:
                        unc_data = page_end - values_off
                    if is_dict and enc2 in (2, 8):
                        pages.append(("dict_z", values_off, page_end,
                                      num_values, unc_data, False,
                                      is_comp))
                    elif enc2 == ENC_PLAIN:
                        pages.append((
                            "splain_z" if is_string else "plain_z",
                            values_off, page_end, num_values, unc_data,
                            False, is_comp))
                    else:
                        return None
                else:
                    if is_dict and enc2 in (2, 8):
                        bit_width = data[values_off]
                        pages.append(("dict", values_off + 1, page_end,
                                      num_values, bit_width))
                    elif is_string and enc2 == ENC_PLAIN:
                        pages.append(("splain", values_off, page_end,
                                      num_values))
                    elif enc2 == ENC_PLAIN:
                        expected = n_valid * np_dtype.itemsize
                        if values_off + expected > page_end:
                            return None
                        pages.append(("plain", values_off, num_values))
                    else:
                        return None
                page_masks.append(mask)
                seen += num_values
                pos = page_end
                continue
            if hdr.get(1) != PAGE_DATA:
                return None
            dph = hdr.get(5, {})
            num_values = dph.get(1)
            if num_values is None:
                return None
            values_off = r.pos
            page_end = r.pos + page_bytes
            mask = None
            if codec != "UNCOMPRESSED":
                # compressed page: record the compressed extent +
                # uncompressed size; the device decompresses, then (for
                # OPTIONAL all-valid chunks) the level prefix is skipped
                # post-decompression.  Dictionary-index pages D2H their
                # small decompressed payload for the host run parser.
                unc = hdr.get(2)
                if unc is None:
                    return None
                pe = dph.get(2)
                if is_dict and pe in (2, 8):
                    pages.append(("dict_z", values_off, page_end,
                                  num_values, unc, has_levels))
                elif pe == ENC_PLAIN:
                    # compressed PLAIN byte-array pages (whole-chunk
                    # PLAIN strings, or a mid-chunk dictionary
                    # overflow) decompress like any page, then the
                    # payload D2Hs for the host byte-array parse
                    kind = "splain_z" if is_string else "plain_z"
                    pages.append((kind, values_off, page_end,
                                  num_values, unc, has_levels))
                else:
                    return None
                page_masks.append(None)
                seen += num_values
                pos = page_end
                continue
            if has_levels:
                lvl_len = struct.unpack_from("<I", data, values_off)[0]
                if not chunk_all_valid:
                    mask = _decode_defs(data, values_off + 4, lvl_len,
                                        num_values, max_def)
                values_off += 4 + lvl_len
            n_valid = int(mask.sum()) if mask is not None else num_values
            page_enc = dph.get(2)
            if is_dict and page_enc in (2, 8):  # PLAIN_DICT / RLE_DICT
                bit_width = data[values_off]
                pages.append(("dict", values_off + 1, page_end,
                              num_values, bit_width))
            elif is_string and page_enc == ENC_PLAIN:
                # PLAIN byte-array page: a whole-chunk PLAIN string
                # column, or pyarrow's mid-chunk dictionary overflow.
                # The (compacted) length-prefixed values parse on host
                # (parse_byte_arrays) and dictionary-encode per chunk.
                pages.append(("splain", values_off, page_end,
                              num_values))
            elif is_string:
                return None
            elif page_enc == ENC_PLAIN:
                # also reached as the writer's mid-chunk fallback when a
                # dictionary overflows: later pages switch to PLAIN
                expected = n_valid * np_dtype.itemsize
                if values_off + expected > page_end:
                    return None
                pages.append(("plain", values_off, num_values))
            else:
                return None
            page_masks.append(mask)
            seen += num_values
            pos = page_end
        if seen != col.num_values:
            return None
        if any(pg[0] == "dict" for pg in pages):
            enc_kind = "dict"
        elif any(pg[0] == "splain" for pg in pages):
            enc_kind = "splain"
        elif any(pg[0] == "dict_z" for pg in pages):
            enc_kind = "dict_z"
        elif any(pg[0] == "splain_z" for pg in pages):
            enc_kind = "splain_z"
        elif any(pg[0] == "plain_z" for pg in pages):
            enc_kind = "plain_z"
        else:
            enc_kind = "plain"
        out.append(ColumnChunkLayout(name, np_dtype, pages,
                                     col.num_values, enc_kind, dict_page,
                                     page_masks, is_string=is_string,
                                     codec=codec, max_def=max_def))
    return out


def _decode_rle_indices(data, off: int, end: int, n: int, bw: int
                        ) -> np.ndarray:
    """Host RLE/bit-packed hybrid decode of dictionary indices at
    ``bw`` bits (testing / CPU path; the device twin is k_rle_decode)."""
    out = np.empty(n, dtype=np.int32)
    pos = off
    filled = 0
    vbytes = (bw + 7) // 8
    while filled < n and pos < end:
        header = 0
        shift = 0
        while True:
            b = data[pos]
            pos += 1
            header |= (b & 0x7F) << shift
            if not b & 0x80:
                break
            shift += 7
        if header & 1:  # bit-packed run of (header>>1) groups of 8
            groups = header >> 1
            nbytes = groups * bw
            packed = np.frombuffer(data, dtype=np.uint8, count=nbytes,
                                   offset=pos)
            bits = np.unpackbits(packed, bitorder="little")
            vals = bits.reshape(-1, bw).astype(np.uint32)
            vals = (vals << np.arange(bw, dtype=np.uint32)).sum(
                axis=1)
            cnt = min(groups * 8, n - filled)
            out[filled:filled + cnt] = vals[:cnt]
            pos += nbytes
            filled += cnt
        else:  # RLE run: repeated value of ceil(bw/8) bytes LE
            run = min(header >> 1, n - filled)
            v = int.from_bytes(bytes(data[pos:pos + vbytes]), "little")
            pos += vbytes
            out[filled:filled + run] = v
            filled += run
    if filled < n:
        out[filled:] = 0
    return out


def decode_splain_pages(buf, starts: List[int], ends: List[int],
                        counts: List[int]
                        ) -> Tuple[np.ndarray, List[str]]:
    """PLAIN byte-array pages -> (int32 codes over all pages' values
    concatenated, SORTED dictionary).  ``buf`` is a cpu uint8 torch
    tensor or a bytes-like.  The length-prefixed values parse in one
    GIL-released C++ call (parse_byte_arrays) when the extension is
    available, else a python walk; dictionary-encoding runs in arrow
    C++ and the insertion-order dictionary is canonicalized to the
    sorted StringColumn contract."""
    import torch
    from ..ops import native as _native
    n = int(sum(counts))
    if _native.available():
        if not isinstance(buf, torch.Tensor):
            import warnings
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                buf = torch.frombuffer(buf, dtype=torch.uint8)
        offs, bts = _native.ext().parse_byte_arrays(
            buf, torch.tensor(starts, dtype=torch.int64),
            torch.tensor(ends, dtype=torch.int64),
            torch.tensor(counts, dtype=torch.int64))
        import pyarrow as pa
        arr = pa.Array.from_buffers(
            pa.utf8(), n,
            [None, pa.py_buffer(offs.numpy()), pa.py_buffer(bts.numpy())])
        denc = arr.dictionary_encode()
        vals = denc.dictionary.to_pylist()
        codes = denc.indices.to_numpy(zero_copy_only=False).astype(
            np.int32, copy=False)
    else:  # pure-python fallback (CPU testing without the extension)
        data = buf.numpy().tobytes() if hasattr(buf, "numpy") else buf
        raw: List[str] = []
        for st, end, cnt in zip(starts, ends, counts):
            pos = st
            for _ in range(cnt):
                ln = struct.unpack_from("<I", data, pos)[0]
                pos += 4
                raw.append(data[pos:pos + ln].decode("utf-8"))
                pos += ln
        vals = list(dict.fromkeys(raw))
        vi0 = {v: i for i, v in enumerate(vals)}
        codes = np.fromiter((vi0[v] for v in raw), np.int32, n)
    if any(vals[i] > vals[i + 1] for i in range(len(vals) - 1)):
        order = sorted(range(len(vals)), key=vals.__getitem__)
        svals = [vals[i] for i in order]
        lut = np.empty(len(vals), np.int32)
        for rank, i in enumerate(order):
            lut[i] = rank
        codes = lut[codes]
        return codes, svals
    return np.ascontiguousarray(codes), list(vals)


def read_native_host(path: str, columns: Optional[List[str]] = None
                     ) -> Optional[Tuple[Dict[str, np.ndarray],
                                         Dict[str, np.ndarray]]]:
    """Host-side decode of a native-layout file (testing / CPU path).
    Returns (columns, validity masks); masks holds entries only for
    columns containing nulls (null slots in the value array are 0).
    String columns come back as StrCol (codes + dictionary)."""
    layout = read_native_layout(path, columns)
    if layout is None:
        return None
    data, chunks = layout
    if any(not (c.encoding in ("plain", "dict")
                or (c.is_string and c.encoding == "splain"))
           for c in chunks):
        return None  # compressed chunks decode on the device path
    acc: Dict[str, list] = {}
    macc: Dict[str, list] = {}
    any_null: Dict[str, bool] = {}
    str_dicts: Dict[str, list] = {}
    for c in chunks:  # row-group-major order
        if not c.pages:  # zero-row chunk still contributes its column
            acc.setdefault(c.name, []).append(
                np.empty(0, dtype=c.np_dtype))
            macc.setdefault(c.name, []).append(np.ones(0, dtype=bool))
            if c.is_string:
                str_dicts.setdefault(c.name, []).append([])
            continue
        if c.is_string:
            values = c.dict_values(data) if c.dict_page else []
            for page, mask in zip(c.pages, c.page_masks):
                if page[0] == "splain":
                    _, p_start, p_end, nv = page
                    n_valid = int(mask.sum()) if mask is not None else nv
                    codes, page_vals = decode_splain_pages(
                        data, [p_start], [p_end], [n_valid])
                else:
                    _, p_start, p_end, nv, bw = page
                    n_valid = int(mask.sum()) if mask is not None else nv
                    codes = _decode_rle_indices(data, p_start, p_end,
                                                n_valid, bw)
                    page_vals = values
                if mask is None:
                    part = codes
                    pm = np.ones(nv, dtype=bool)
                else:
                    part = np.zeros(nv, dtype=np.int32)
                    part[mask] = codes
                    pm = mask
                    any_null[c.name] = True
                acc.setdefault(c.name, []).append(part)
                macc.setdefault(c.name, []).append(pm)
                str_dicts.setdefault(c.name, []).append(page_vals)
            continue
        dvals = None
        if c.dict_page is not None:
            doff, dn = c.dict_page
            dvals = np.frombuffer(data, dtype=c.np_dtype, count=dn,
                                  offset=doff)
        for page, mask in zip(c.pages, c.page_masks):
            if page[0] == "dict":
                # numeric dictionary page (Spark's default output
                # shape): decode indices, gather through the PLAIN
                # dictionary values
                _, p_start, p_end, nv, bw = page
                n_valid = int(mask.sum()) if mask is not None else nv
                idx = _decode_rle_indices(data, p_start, p_end,
                                          n_valid, bw)
                vals = (dvals[idx] if n_valid
                        else np.empty(0, dtype=c.np_dtype))
            else:
                _, off, nv = page
                n_valid = int(mask.sum()) if mask is not None else nv
                vals = np.frombuffer(data, dtype=c.np_dtype,
                                     count=n_valid, offset=off)
            if mask is None:
                part = vals
                pm = np.ones(nv, dtype=bool)
            else:
                part = np.zeros(nv, dtype=c.np_dtype)
                part[mask] = vals
                pm = mask
                any_null[c.name] = True
            acc.setdefault(c.name, []).append(part)
            macc.setdefault(c.name, []).append(pm)
    cols: Dict[str, Any] = {}
    for name, parts in acc.items():
        if name in str_dicts:
            dicts = str_dicts[name]
            if all(d == dicts[0] for d in dicts[1:]) and \
                    all(dicts[0][i] <= dicts[0][i + 1]
                        for i in range(len(dicts[0]) - 1)):
                merged_vals = dicts[0]
                codes = (np.concatenate(parts) if len(parts) > 1
                         else parts[0].copy())
            else:  # per-row-group dictionaries: merge + remap
                merged_vals = sorted(set().union(*map(set, dicts)))
                vi = {v: i for i, v in enumerate(merged_vals)}
                remapped = []
                for part, d in zip(parts, dicts):
                    lut = np.array([vi[v] for v in d] or [0],
                                   dtype=np.int32)
                    remapped.append(lut[part])
                codes = (np.concatenate(remapped) if len(remapped) > 1
                         else remapped[0])
            if not merged_vals and len(codes):
                # all-null column: masked slots carry code 0 — give it
                # a value to dereference
                merged_vals = [""]
            cols[name] = StrCol(codes, list(merged_vals))
        else:
            cols[name] = (np.concatenate(parts) if len(parts) > 1
                          else (parts[0] if parts[0].flags.writeable
                                else parts[0].copy()))
    masks = {name: np.concatenate(macc[name]) if len(macc[name]) > 1
             else macc[name][0]
             for name in acc if any_null.get(name)}
    return cols, masks
