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
from typing import Dict, List, Optional, Tuple

import numpy as np

MAGIC = b"PAR1"

# parquet physical types
T_INT32, T_INT64, T_FLOAT, T_DOUBLE = 1, 2, 4, 5
_NP_TO_PARQUET = {
    np.dtype("int64"): T_INT64,
    np.dtype("int32"): T_INT32,
    np.dtype("float64"): T_DOUBLE,
    np.dtype("float32"): T_FLOAT,
}
_PARQUET_TO_NP = {v: k for k, v in _NP_TO_PARQUET.items()}

ENC_PLAIN = 0
CODEC_UNCOMPRESSED = 0
PAGE_DATA = 0
REP_REQUIRED = 0


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

def _page_header(num_values: int, nbytes: int) -> bytes:
    w = TWriter()
    w.struct_begin()
    w.field_i32(1, PAGE_DATA)
    w.field_i32(2, nbytes)
    w.field_i32(3, nbytes)
    w.field_struct_begin(5)  # DataPageHeader
    w.field_i32(1, num_values)
    w.field_i32(2, ENC_PLAIN)
    w.field_i32(3, ENC_PLAIN)  # definition_level_encoding (unused: REQUIRED)
    w.field_i32(4, ENC_PLAIN)  # repetition_level_encoding
    w.struct_end()
    w.struct_end()
    return bytes(w.buf)


def _statistics(arr: np.ndarray) -> Tuple[bytes, bytes]:
    """(min_value, max_value) PLAIN-encoded."""
    return arr.min().tobytes(), arr.max().tobytes()


def write_parquet_native(columns: "Dict[str, np.ndarray]", path: str
                         ) -> Tuple[int, int]:
    """Write numeric columns as one row group, one PLAIN page per column.
    Returns (size, mtime_ms)."""
    names = list(columns.keys())
    arrays = [np.ascontiguousarray(columns[n]) for n in names]
    num_rows = len(arrays[0]) if arrays else 0
    for a in arrays:
        if a.dtype not in _NP_TO_PARQUET:
            raise ValueError(f"dtype {a.dtype} not supported natively")
        assert len(a) == num_rows

    chunks: List[bytes] = []
    col_meta: List[Tuple[str, int, int, int, int, bytes, bytes]] = []
    offset = 4  # after magic
    for name, arr in zip(names, arrays):
        payload = arr.tobytes()  # PLAIN little-endian
        header = _page_header(num_rows, len(payload))
        mn, mx = _statistics(arr) if num_rows else (b"", b"")
        col_meta.append((name, _NP_TO_PARQUET[arr.dtype], offset,
                         len(header) + len(payload), num_rows, mn, mx))
        chunks.append(header)
        chunks.append(payload)
        offset += len(header) + len(payload)

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
    for name, ptype, *_ in col_meta:
        w.struct_elem_begin()
        w.field_i32(1, ptype)
        w.field_i32(3, REP_REQUIRED)
        w.field_string(4, name)
        w.struct_end()
    w.field_i64(3, num_rows)
    # row_groups
    w.field_list_begin(4, CT_STRUCT, 1)
    w.struct_elem_begin()
    total_bytes = sum(m[3] for m in col_meta)
    w.field_list_begin(1, CT_STRUCT, len(col_meta))
    for name, ptype, off, nbytes, nvals, mn, mx in col_meta:
        w.struct_elem_begin()  # ColumnChunk
        w.field_i64(2, off)  # file_offset
        w.field_struct_begin(3)  # ColumnMetaData
        w.field_i32(1, ptype)
        w.field_list_begin(2, CT_I32, 1)
        w.i32_elem(ENC_PLAIN)
        w.field_list_begin(3, CT_BINARY, 1)
        w.buf += _varint(len(name.encode()))
        w.buf += name.encode()
        w.field_i32(4, CODEC_UNCOMPRESSED)
        w.field_i64(5, nvals)
        w.field_i64(6, nbytes)
        w.field_i64(7, nbytes)
        w.field_i64(9, off)  # data_page_offset
        if mn:
            w.field_struct_begin(12)  # Statistics
            w.field_binary(1, mx)  # max (legacy)
            w.field_binary(2, mn)  # min (legacy)
            w.field_i64(3, 0)      # null_count
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
    st = os.stat(path)
    return st.st_size, int(st.st_mtime * 1000)


# ---------------------------------------------------------------------------
# Reader (page layout for the device decode path)
# ---------------------------------------------------------------------------

class ColumnChunkLayout:
    """Per-column page layout.

    encoding "plain": pages = [(values byte offset, num values), ...]
    encoding "dict":  pages = [(payload start, payload end, num values,
                      bit width), ...] of RLE/bit-packed index payloads,
                      plus dict_page = (values byte offset, dict size)
    """
    __slots__ = ("name", "np_dtype", "pages", "num_values", "encoding",
                 "dict_page")

    def __init__(self, name, np_dtype, pages, num_values,
                 encoding="plain", dict_page=None):
        self.name = name
        self.np_dtype = np_dtype
        self.pages = pages
        self.num_values = num_values
        self.encoding = encoding
        self.dict_page = dict_page


def read_native_layout(path: str,
                       columns: Optional[List[str]] = None,
                       data: Optional[bytes] = None
                       ) -> Optional[Tuple[bytes, List[ColumnChunkLayout]]]:
    """If ``path`` decodes natively (uncompressed PLAIN pages of numeric
    columns — our writer's files and pyarrow's NONE/PLAIN files), return
    (raw file bytes, per-column page layouts); else None (caller falls
    back to pyarrow).  ``data`` can supply pre-read file bytes (any
    buffer protocol object) to avoid a second disk read."""
    import pyarrow.parquet as pq
    try:
        pf = pq.ParquetFile(path)
        md = pf.metadata
        pf_schema = pf.schema
    except Exception:  # noqa: BLE001
        return None
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
        if col.compression.upper() != "UNCOMPRESSED":
            return None
        encs = set(col.encodings)
        is_dict = bool(encs & {"PLAIN_DICTIONARY", "RLE_DICTIONARY"})
        if encs - {"PLAIN", "RLE", "BIT_PACKED", "PLAIN_DICTIONARY",
                   "RLE_DICTIONARY"}:
            return None
        ptype = {"INT64": T_INT64, "INT32": T_INT32, "FLOAT": T_FLOAT,
                 "DOUBLE": T_DOUBLE}.get(col.physical_type)
        if ptype is None:
            return None
        np_dtype = _PARQUET_TO_NP[ptype]
        # nulls unsupported: skip the def-level prefix only when levels
        # exist; reject columns that actually contain nulls
        has_levels = pf_schema.column(col_index).max_definition_level > 0
        st = col.statistics
        if has_levels and (st is None or st.null_count not in (0, None)):
            if st is None or st.null_count != 0:
                return None

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
            dict_page = (r.pos, dict_n)
            if dict_off >= pos:
                # dictionary physically precedes data pages
                pos = r.pos + hdr.get(3)

        seen = 0
        pages: List[Tuple] = []
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
            if hdr.get(1) != PAGE_DATA:
                return None
            dph = hdr.get(5, {})
            num_values = dph.get(1)
            if num_values is None:
                return None
            values_off = r.pos
            page_end = r.pos + page_bytes
            if has_levels:
                lvl_len = struct.unpack_from("<I", data, values_off)[0]
                values_off += 4 + lvl_len
            page_enc = dph.get(2)
            if is_dict and page_enc in (2, 8):  # PLAIN_DICT / RLE_DICT
                bit_width = data[values_off]
                pages.append(("dict", values_off + 1, page_end,
                              num_values, bit_width))
            elif page_enc == ENC_PLAIN:
                # also reached as the writer's mid-chunk fallback when a
                # dictionary overflows: later pages switch to PLAIN
                expected = num_values * np_dtype.itemsize
                if values_off + expected > page_end:
                    return None
                pages.append(("plain", values_off, num_values))
            else:
                return None
            seen += num_values
            pos = page_end
        if seen != col.num_values:
            return None
        enc_kind = ("dict" if any(pg[0] == "dict" for pg in pages)
                    else "plain")
        out.append(ColumnChunkLayout(name, np_dtype, pages,
                                     col.num_values, enc_kind, dict_page))
    return out


def read_native_host(path: str, columns: Optional[List[str]] = None
                     ) -> Optional[Dict[str, np.ndarray]]:
    """Host-side decode of a native-layout file (testing / CPU path)."""
    layout = read_native_layout(path, columns)
    if layout is None:
        return None
    data, chunks = layout
    if any(c.encoding != "plain" for c in chunks):
        return None  # dictionary decode is the device path; host->pyarrow
    acc: Dict[str, list] = {}
    for c in chunks:  # row-group-major order
        parts = [np.frombuffer(data, dtype=c.np_dtype, count=nv,
                               offset=off) for _, off, nv in c.pages]
        acc.setdefault(c.name, []).extend(parts)
    return {name: (np.concatenate(parts) if len(parts) > 1
                   else parts[0].copy())
            for name, parts in acc.items()}
