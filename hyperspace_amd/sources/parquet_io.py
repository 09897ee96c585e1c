"""Parquet read/write for the build pipeline (host path).

Write side keeps the reference's bucket-id-in-filename contract:
``part-<task>-<uuid>_<bucketid>.c000.parquet`` — OptimizeAction groups
files by the bucket id parsed from the name
(reference: actions/OptimizeAction.scala:109-111; Spark bucketed file
naming from saveWithBuckets, index/DataFrameWriterExtensions.scala:50-80).

Index data files are written UNCOMPRESSED with PLAIN encoding so the
device Parquet page-decode kernel (K1) has a direct path; pyarrow handles
footer/metadata assembly.
"""

from __future__ import annotations

import os
import re
import sys
import uuid
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

_BUCKET_RE = re.compile(r".*_(\d+)(?:\.\w+)*\.parquet$")

# Pinned host staging buffers are expensive to allocate (~60 ms/GB —
# comparable to the copy itself), so the device read path recycles them
# through a size-classed pool.  Classes are 8 MiB-granular above 8 MiB
# (pow2 wastes up to 2x of the budget on ~40-60 MB parquet files) and
# power-of-two below; total bounded per process (8 ranks share a node's
# host RAM).
_PINNED_POOL: Dict[int, List["torch.Tensor"]] = {}
_PINNED_POOL_BYTES = 0
# Default 32 GiB: a build-then-serve step keeps BOTH the source-staging
# classes (~few large buffers) and the index-read classes (~hundreds of
# bucket files) resident; a 16 GiB cap evicted one set every step and
# re-paid ~190 ms/GB of pinned allocation on the cold query read
# (profiles/cold_read_probe: full path 8.3 GB/s pool-cold vs 36 GB/s
# pool-warm, H2D-bound).  HS_PINNED_POOL_GB overrides.
_PINNED_POOL_MAX = int(
    float(os.environ.get("HS_PINNED_POOL_GB", "32")) * (1 << 30))
import threading as _threading

_pool_lock = _threading.Lock()


def _pinned_size_class(nbytes: int) -> int:
    if nbytes > (8 << 20):
        gran = 8 << 20
        return (nbytes + gran - 1) // gran * gran
    return 1 << max(12, (nbytes - 1).bit_length())


def _pinned_get(nbytes: int) -> "torch.Tensor":
    global _PINNED_POOL_BYTES
    size = _pinned_size_class(nbytes)
    with _pool_lock:
        # exact class first, then up to 2x oversized (a 48 MB index-file
        # read can reuse a 64 MB source-staging buffer instead of paying
        # a fresh pinned allocation)
        cls = size
        while cls <= 2 * size:
            lst = _PINNED_POOL.get(cls)
            if lst:
                _PINNED_POOL_BYTES -= cls
                return lst.pop()
            cls += (8 << 20) if cls >= (8 << 20) else cls
    return torch.empty(size, dtype=torch.uint8, pin_memory=True)


def _pinned_put(buf: "torch.Tensor") -> None:
    global _PINNED_POOL_BYTES
    size = buf.numel()
    with _pool_lock:
        if _PINNED_POOL_BYTES + size <= _PINNED_POOL_MAX:
            _PINNED_POOL.setdefault(size, []).append(buf)
            _PINNED_POOL_BYTES += size


_CODEC_POOL = None
_codec_tl = _threading.local()
# snappy pages with comp >= ratio*unc (near-incompressible, giant
# literals) decode on device; the rest are copy-dense -> host codec
# pool.  0.93 measured best (see profiles/r2_summary.md); 0 = all
# device, >1 = all host (sweep knob).
_DEV_RATIO = float(os.environ.get("HS_SNAPPY_DEV_RATIO", "0.93"))


def _codec_pool():
    """Shared host-codec thread pool (gzip/zstd/lz4/copy-dense-snappy
    page decompression; the codecs release the GIL)."""
    global _CODEC_POOL
    if _CODEC_POOL is None:
        from concurrent.futures import ThreadPoolExecutor
        _CODEC_POOL = ThreadPoolExecutor(
            max_workers=int(os.environ.get("HS_CODEC_WORKERS", "16")),
            thread_name_prefix="hs-codec")
    return _CODEC_POOL


def _tl_decompress(name: str, view, unc: int):
    """Decompress with a THREAD-LOCAL pyarrow codec: a pa.Codec shared
    across threads segfaults (reproduced with zstd at 16 threads), so
    each pool worker keeps its own instance per codec name."""
    c = getattr(_codec_tl, name, None)
    if c is None:
        import pyarrow as _pa
        c = _pa.Codec(name)
        setattr(_codec_tl, name, c)
    return c.decompress(view, unc)


def bucket_id_of_file(path: str) -> Optional[int]:
    m = _BUCKET_RE.match(os.path.basename(path))
    return int(m.group(1)) if m else None


def bucket_file_name(task_id: int, bucket_id: int, token: str = "") -> str:
    token = token or uuid.uuid4().hex[:8]
    return f"part-{task_id:05d}-{token}_{bucket_id:05d}.c000.parquet"


def write_batch_parquet(batch, path: str, compression: Optional[str] = None,
                        use_dictionary: bool = False) -> Tuple[int, int]:
    """Write a ColumnBatch to one parquet file.  Returns (size, mtime_ms).

    All-numeric batches use the native writer (uncompressed PLAIN pages
    assembled directly from the column buffers — K3 encode side); string
    columns fall back to pyarrow.
    """
    from ..execution.columnar import StringColumn
    if compression is None:
        from .native_parquet import (write_parquet_native, StrCol,
                                     _NP_TO_PARQUET)
        cols = {name: (StrCol(c.codes.numpy(), c.values)
                       if isinstance(c, StringColumn) else c.numpy())
                for name, c in batch.columns.items()}
        if all(isinstance(a, StrCol) or a.dtype in _NP_TO_PARQUET
               for a in cols.values()):
            masks = {name: m.numpy()
                     for name, m in batch.masks.items()} \
                if batch.masks else None
            return write_parquet_native(cols, path, masks)
    import pyarrow.parquet as pq
    table = batch.to_arrow()
    pq.write_table(table, path, compression=compression or "NONE",
                   use_dictionary=use_dictionary,
                   write_statistics=True,
                   data_page_version="1.0")
    st = os.stat(path)
    return st.st_size, int(st.st_mtime * 1000)


def _resolve_read_columns(p, columns):
    """Map requested column names onto a file schema: a dotted request
    ("a.b.c") may be a nested leaf (read the top-level struct and let
    from_arrow flatten it); flat columns whose NAME contains dots (our
    index data for nested leaves) match the file schema directly."""
    import pyarrow.parquet as pq
    if columns is None:
        return None
    top = {n.lower(): n for n in pq.ParquetFile(p).schema_arrow.names}
    read_cols = []
    for c in columns:
        if c.lower() in top:
            read_cols.append(top[c.lower()])
        elif "." in c and c.split(".")[0].lower() in top:
            root = top[c.split(".")[0].lower()]
            if root not in read_cols:
                read_cols.append(root)
        else:
            read_cols.append(c)
    return read_cols


def _pyarrow_file_dict(p, columns):
    """One non-native file read through pyarrow into the per-file
    (columns, masks) dict shape of read_native_host, so a batch can mix
    native and fallback files."""
    import pyarrow.parquet as pq
    from ..execution.columnar import ColumnBatch, StringColumn
    from .native_parquet import StrCol
    t = pq.read_table(p, columns=_resolve_read_columns(p, columns))
    b = ColumnBatch.from_arrow(t)
    if columns is not None:
        b = b.select(columns)
    cols = {}
    for name, c in b.columns.items():
        cols[name] = (StrCol(c.codes.numpy(), c.values)
                      if isinstance(c, StringColumn) else c.numpy())
    masks = {name: m.numpy() for name, m in (b.masks or {}).items()}
    return cols, masks, t.num_rows


def read_files_batch(paths: List[str], columns: Optional[List[str]] = None):
    """Read parquet files into a single host ColumnBatch, plus per-file row
    counts (for lineage / per-file segmentation).  Native-layout files
    (uncompressed PLAIN numeric) bypass pyarrow; non-native files decode
    per-file through pyarrow and merge with the native ones (a single
    legacy file no longer drops the whole batch to the slow path)."""
    import numpy as np
    from ..execution.columnar import ColumnBatch
    from .native_parquet import read_native_host

    per_file = []
    per_file_masks = []
    row_counts = []
    native_ok = True
    try:
        for p in paths:
            res = read_native_host(p, columns)
            if res is None:
                cols, fmasks, n = _pyarrow_file_dict(p, columns)
            else:
                cols, fmasks = res
                n = len(next(iter(cols.values()))) if cols else 0
            per_file.append(cols)
            per_file_masks.append(fmasks)
            row_counts.append(n)
    except Exception:  # exotic schema mix: whole-batch pyarrow path
        native_ok = False
        row_counts = []
    if native_ok and paths:
        names = list(per_file[0].keys())
        if columns is not None:
            order = {c.lower(): i for i, c in enumerate(columns)}
            names.sort(key=lambda n: order.get(n.lower(), 99))
        from .native_parquet import StrCol
        from ..execution.columnar import StringColumn
        merged = {}
        merged_masks = {}
        for name in names:
            arrs = [f[name] for f in per_file]
            if any(isinstance(a, StrCol) for a in arrs):
                # merge per-file dictionaries, remap codes (unsorted
                # parquet insertion-order dictionaries force the remap:
                # StringColumn codes must follow lex order)
                dicts = [a.values for a in arrs]
                if all(d == dicts[0] for d in dicts[1:]) and \
                        all(dicts[0][i] <= dicts[0][i + 1]
                            for i in range(len(dicts[0]) - 1)):
                    mvals = list(dicts[0])
                    codes = (np.concatenate([a.codes for a in arrs])
                             if len(arrs) > 1 else arrs[0].codes)
                else:
                    mvals = sorted(set().union(*map(set, dicts)))
                    vi = {v: i for i, v in enumerate(mvals)}
                    parts = []
                    for a in arrs:
                        lut = np.array([vi[v] for v in a.values] or [0],
                                       dtype=np.int32)
                        parts.append(lut[a.codes])
                    codes = (np.concatenate(parts) if len(parts) > 1
                             else parts[0])
                merged[name] = StringColumn(
                    torch.from_numpy(np.ascontiguousarray(codes)), mvals)
            else:
                merged[name] = torch.from_numpy(
                    np.concatenate(arrs) if len(arrs) > 1 else arrs[0])
            if any(name in fm for fm in per_file_masks):
                mparts = [fm.get(name, np.ones(rc, dtype=bool))
                          for fm, rc in zip(per_file_masks, row_counts)]
                merged_masks[name] = torch.from_numpy(
                    np.concatenate(mparts) if len(mparts) > 1
                    else mparts[0])
        return ColumnBatch(merged, merged_masks), row_counts

    import pyarrow.parquet as pq
    import pyarrow as pa
    tables = []
    row_counts = []
    for p in paths:
        t = pq.read_table(p, columns=_resolve_read_columns(p, columns))
        tables.append(t)
        row_counts.append(t.num_rows)
    if not tables:
        return ColumnBatch({}), []
    table = pa.concat_tables(tables, promote_options="default")
    batch = ColumnBatch.from_arrow(table)
    if columns is not None:
        batch = batch.select(columns)
    return batch, row_counts


def read_files_batch_device(paths: List[str], device,
                            columns: Optional[List[str]] = None):
    """Device-decoded parquet read (K1): raw file bytes -> HBM -> the
    unaligned-copy decode kernel assembles each column.  Falls back to the
    host path + H2D for non-native files.  Returns (ColumnBatch on
    ``device``, per-file row counts)."""
    import numpy as np
    import torch as _torch
    from ..execution.columnar import ColumnBatch
    from ..ops import native as native_ext
    from .native_parquet import layout_cache_get, read_native_layout

    # Fully-overlapped pipeline: footer metadata (cheap) sizes the output
    # tensors up front; worker threads then read each file into a pinned
    # buffer, upload async and launch the decode kernels into their
    # pre-assigned regions — disk reads, PCIe transfers and decode
    # overlap across files.
    import pyarrow.parquet as pq
    from concurrent.futures import ThreadPoolExecutor

    _PHYS_TO_NP = {"INT64": np.dtype("int64"), "INT32": np.dtype("int32"),
                   "DOUBLE": np.dtype("float64"),
                   "FLOAT": np.dtype("float32")}

    def fallback():
        batch, rc = read_files_batch(paths, columns)
        return batch.to(device), rc

    want = {c.lower() for c in columns} if columns is not None else None

    # write-time layout cache: every file our native writer produced in
    # this process skips the footer + page-header parse (the dominant
    # GIL-bound cost of a ~200-file cold load)
    cached = []
    for p in paths:
        ent = layout_cache_get(p)
        if ent is None:
            cached = None
            break
        cached.append(ent)

    metas = []
    schemas = []
    per_file_layouts = None
    string_cols = set()
    if cached is not None:
        row_counts = [ent[0] for ent in cached]
        lay0 = cached[0][1]
        names = [c.name for c in lay0
                 if want is None or c.name.lower() in want]
        dtypes = {c.name: c.np_dtype for c in lay0}
        nullable_cols = {c.name for ent in cached for c in ent[1]
                         if c.has_nulls}
        string_cols = {c.name for ent in cached for c in ent[1]
                       if c.is_string}
        per_file_layouts = [
            [c for c in ent[1]
             if want is None or c.name.lower() in want]
            for ent in cached]
    else:
        for p in paths:
            try:
                pf = pq.ParquetFile(p)
            except Exception:  # noqa: BLE001
                return fallback()
            metas.append(pf.metadata)
            schemas.append(pf.schema)

        # column nullability from chunk statistics (sizes the
        # preallocated masks); an OPTIONAL chunk without statistics
        # could hide nulls -> the host path decides instead
        nullable_cols = set()
        for md, sch in zip(metas, schemas):
            for rg_i in range(md.num_row_groups):
                rg = md.row_group(rg_i)
                for i in range(rg.num_columns):
                    col = rg.column(i)
                    st = col.statistics
                    if st is None or st.null_count is None:
                        if sch.column(i).max_definition_level > 0:
                            return fallback()
                    elif st.null_count > 0:
                        nullable_cols.add(col.path_in_schema)

        rg0 = metas[0].row_group(0)
        names = []
        dtypes = {}
        for i in range(rg0.num_columns):
            col = rg0.column(i)
            if want is not None and \
                    col.path_in_schema.lower() not in want:
                continue
            npd = _PHYS_TO_NP.get(col.physical_type)
            if npd is None:
                if col.physical_type == "BYTE_ARRAY":
                    # dictionary strings decode to int32 codes; the
                    # layout parse validates the encoding (else the
                    # worker falls back)
                    npd = np.dtype("int32")
                    string_cols.add(col.path_in_schema)
                elif col.physical_type == "FIXED_LEN_BYTE_ARRAY" and                         schemas[0].column(i).logical_type.type ==                         "DECIMAL" and                         schemas[0].column(i).precision <= 18:
                    # FLBA decimal(p<=18): engine representation is the
                    # unscaled int64; files carrying it have no native
                    # layout and read per file on host workers
                    npd = np.dtype("int64")
                elif col.physical_type == "INT96":
                    # legacy timestamp: per-file host read -> int64
                    npd = np.dtype("int64")
                else:
                    return fallback()
            names.append(col.path_in_schema)
            dtypes[col.path_in_schema] = npd
        row_counts = [md.num_rows for md in metas]
    if columns is not None:
        order = {c.lower(): i for i, c in enumerate(columns)}
        names.sort(key=lambda n: order.get(n.lower(), 99))

    file_base = np.concatenate([[0], np.cumsum(row_counts)[:-1]])
    total_rows = int(sum(row_counts))

    np_to_torch = {np.dtype("int64"): _torch.int64,
                   np.dtype("int32"): _torch.int32,
                   np.dtype("float64"): _torch.float64,
                   np.dtype("float32"): _torch.float32}
    out = {n: _torch.empty(total_rows, dtype=np_to_torch[dtypes[n]],
                           device=device) for n in names}
    out_masks = {n: _torch.ones(total_rows, dtype=_torch.bool,
                                device=device)
                 for n in names if n in nullable_cols}
    ext = native_ext.ext()
    statuses: List["_torch.Tensor"] = []  # snappy per-page status words
    # (name, abs row start, abs row end, dictionary values) per string
    # chunk — merged into one dictionary per column after the sync
    str_chunks: List[Tuple[str, int, int, List[str]]] = []

    # per-worker HIP streams: each file's H2D copy and decode kernels run
    # on their own stream, so copies overlap other files' decodes instead
    # of serializing on the default stream (xfers are the cold-load bound)
    n_streams = 16  # row-group units of a single file fan out too
    streams = [_torch.cuda.Stream(device=device) for _ in range(n_streams)]

    def load_file(i):
        """Read + layout-parse one file (no decode)."""
        p = paths[i]
        size = os.path.getsize(p)
        buf = _pinned_get(size + 4)
        view = memoryview(buf.numpy())
        with open(p, "rb", buffering=0) as f:
            f.readinto(view[:size])
        if per_file_layouts is not None:
            chunks = per_file_layouts[i]
        else:
            lay = read_native_layout(p, columns, data=view[:size],
                                     meta=metas[i], pf_schema=schemas[i])
            if lay is None:
                _pinned_put(buf)
                return None
            chunks = lay[1]
        return buf, size, chunks

    def split_row_groups(chunks):
        """[(row_offset, chunks-of-one-row-group)] — layout order is
        row-group-major, so a repeated column name starts a new group."""
        groups = []
        cur: list = []
        seen = set()
        row_off = 0
        for c in chunks:
            if c.name in seen:
                groups.append((row_off, cur))
                row_off += cur[0].num_values
                cur = []
                seen = set()
            cur.append(c)
            seen.add(c.name)
        if cur:
            groups.append((row_off, cur))
        return groups

    def _decode_on_stream(i, buf, size, chunks, dev_bytes, row_off):
        cursors = {n: int(file_base[i]) + row_off for n in names}
        for c in chunks:
            itemsize = c.np_dtype.itemsize
            written = cursors[c.name]
            if _dt == "3":
                t_c0 = _time.perf_counter()
            if c.encoding in ("plain_z", "dict_z", "splain_z"):
                # snappy chunk (K1): one wave per page decompresses into
                # a scratch buffer; PLAIN pages copy straight from
                # scratch, dictionary-index pages D2H their (small)
                # decompressed payload for the host run parser and
                # gather through the decompressed dictionary on device.
                has_zdict = c.encoding == "dict_z"
                segs = []
                idn = []  # V2 is_compressed=false page: identity copy
                if has_zdict:
                    _, dz_off, dz_end, dict_n, dict_unc = c.dict_page
                    segs.append((dz_off, dz_end, dict_unc))
                    idn.append(False)
                for p in c.pages:
                    segs.append((p[1], p[2], p[4]))
                    idn.append(len(p) > 6 and not p[6])
                uncs_all = [s[2] for s in segs]
                doff_all = np.concatenate([[0], np.cumsum(uncs_all)])
                scratch = _torch.empty(int(doff_all[-1]) + 4,
                                       dtype=_torch.uint8, device=device)
                # latency/bandwidth split: a page that actually
                # COMPRESSED is copy-dense — a serial ~150-500 ns/op
                # chain on a single wave (a 1 MB dictionary page of
                # small-valued int64s measured 137 ms as 264k tiny ops,
                # profiles/r2_summary.md) — so it decodes on the HOST
                # C++ codec (~GB/s per worker thread) and uploads;
                # near-incompressible pages (giant literals) stay on
                # device where decode is a bandwidth-bound copy.
                host_bytes: Dict[int, object] = {}
                if c.codec == "SNAPPY":
                    dev_idx = [i for i, s in enumerate(segs)
                               if not idn[i]
                               and (s[1] - s[0]) >= _DEV_RATIO * s[2]]
                else:
                    dev_idx = []  # gzip/zstd/brotli: host codec only
                host_idx = [i for i in range(len(segs))
                            if i not in dev_idx and not idn[i]]
                for i, s0 in enumerate(segs):
                    if idn[i]:  # stored uncompressed: straight copy
                        # (torch u8 slice copy: arbitrary alignment,
                        # async on the decode stream)
                        d0 = int(doff_all[i])
                        scratch[d0:d0 + s0[2]] = \
                            dev_bytes[s0[0]:s0[0] + s0[2]]
                if dev_idx:
                    st = ext.snappy_decompress(
                        dev_bytes,
                        _torch.tensor([segs[i][0] for i in dev_idx],
                                      dtype=_torch.int64),
                        _torch.tensor([segs[i][1] for i in dev_idx],
                                      dtype=_torch.int64),
                        scratch,
                        _torch.tensor([int(doff_all[i])
                                       for i in dev_idx],
                                      dtype=_torch.int64),
                        _torch.tensor([segs[i][2] for i in dev_idx],
                                      dtype=_torch.int64))
                    statuses.append(st)
                if host_idx:
                    # parquet LZ4 / LZ4_RAW pages are raw LZ4 blocks
                    codec_name = ("lz4_raw"
                                  if c.codec in ("LZ4", "LZ4_RAW")
                                  else c.codec.lower())
                    hview = buf.numpy()
                    # stage all host-decoded pages in ONE pinned buffer:
                    # the codec reads the compressed page zero-copy from
                    # the read buffer, one host memcpy lands it in
                    # pinned memory, and the H2D is a true async copy
                    # (the old per-page pageable copy_ serialized the
                    # decode stream on every page)
                    total_host = sum(segs[i][2] for i in host_idx)
                    hstage = _pinned_get(total_host)
                    keepalive.append(hstage)
                    # fan the per-page codec calls AND the pinned
                    # staging memcpy over the shared codec pool — the
                    # pyarrow codecs release the GIL, and host_memcpy
                    # releases it for the copy (a numpy slice
                    # assignment held it for ~GB of memcpy per read,
                    # serializing all 16 workers)
                    hoffs = []
                    hoff = 0
                    for i in host_idx:
                        hoffs.append(hoff)
                        hoff += segs[i][2]

                    def _stage(i2, off2):
                        a2, b2, unc2 = segs[i2]
                        dec = _tl_decompress(codec_name,
                                             hview[a2:b2], unc2)
                        if len(dec) != unc2:
                            raise ValueError("short page")
                        ext.host_memcpy(hstage, off2, dec)
                        return dec

                    futs = [_codec_pool().submit(_stage, i2, off2)
                            for i2, off2 in zip(host_idx, hoffs)]
                    for i, fut, off2 in zip(host_idx, futs, hoffs):
                        unc2 = segs[i][2]
                        try:
                            # host-decoded payloads stay available for
                            # the header/run parse below — it reads them
                            # WITHOUT a device round-trip (the old
                            # scratch D2H sync serialized every chunk
                            # on its own decompression)
                            host_bytes[i] = fut.result()
                        except Exception:  # noqa: BLE001
                            statuses.append(_torch.ones(
                                1, dtype=_torch.int32, device=device))
                            continue
                        scratch[int(doff_all[i]):
                                int(doff_all[i]) + unc2].copy_(
                            hstage[off2:off2 + unc2],
                            non_blocking=True)
                page_base = doff_all[1:-1] if has_zdict \
                    else doff_all[:-1]
                dict_vals = None
                z_is_str = c.is_string
                has_spz = any(p[0] == "splain_z" for p in c.pages)
                vals_list = None
                if has_zdict and z_is_str:
                    # string dictionary: small D2H of the decompressed
                    # dict page, parsed to values on host; indices stay
                    # on device as the column's codes
                    hb0 = host_bytes.get(0)
                    draw = (np.frombuffer(hb0, dtype=np.uint8)
                            if hb0 is not None
                            else scratch[:dict_unc].cpu().numpy())
                    vals_list = []
                    pos = 0
                    for _ in range(dict_n):
                        ln = int(np.frombuffer(draw, "<u4", 1, pos)[0])
                        pos += 4
                        vals_list.append(
                            draw[pos:pos + ln].tobytes().decode("utf-8"))
                        pos += ln
                    if not has_spz:
                        str_chunks.append(
                            (c.name, written,
                             written + sum(p[3] for p in c.pages),
                             vals_list))
                    # with dictionary-overflow PLAIN pages in the chunk
                    # the dictionary spans become per-run entries below
                elif has_zdict:
                    dict_vals = _torch.empty(
                        dict_n + 1, dtype=out[c.name].dtype,
                        device=device)
                    ext.copy_unaligned(scratch, 0, dict_vals, 0,
                                       dict_n * itemsize)
                    dict_vals = dict_vals[:dict_n].contiguous()

                nullable_chunk = c.name in out_masks
                if not nullable_chunk:
                    # FAST PATH (no nulls): one cat + ONE D2H sync per
                    # chunk; all page headers (level-length prefixes,
                    # bit widths) are then read from HOST memory and the
                    # dictionary-index runs of every page parse in one
                    # GIL-released C++ call per bit width.  The round-1
                    # shape (per-page torch slicing + a level-prefix
                    # stack().cpu() sync per chunk) serialized 26-unit
                    # decodes on the GIL.
                    pieces = []
                    dev_parts = []   # scratch slices still needing D2H
                    dev_marks = []   # their slots in pieces
                    roff = []  # page -> offset of its span in hb
                    cur = 0
                    for j, page in enumerate(c.pages):
                        base = int(page_base[j])
                        if page[0] == "dict_z":
                            ln_span = int(page[4])
                        elif page[5]:
                            ln_span = 4  # level-length prefix only
                        else:
                            roff.append(-1)
                            continue
                        roff.append(cur)
                        hbuf = host_bytes.get(
                            j + 1 if has_zdict else j)
                        if hbuf is not None:
                            pieces.append(np.frombuffer(
                                hbuf, dtype=np.uint8,
                                count=ln_span))
                        else:
                            dev_marks.append(len(pieces))
                            pieces.append(None)
                            dev_parts.append(
                                scratch[base:base + ln_span])
                        cur += ln_span
                    if dev_parts:
                        dv = _torch.cat(dev_parts).cpu().numpy()
                        off3 = 0
                        for m, t3 in zip(dev_marks, dev_parts):
                            ln3 = t3.numel()
                            pieces[m] = dv[off3:off3 + ln3]
                            off3 += ln3
                    hb = (np.concatenate(pieces) if pieces
                          else np.empty(0, dtype=np.uint8))
                    if _dt == "3":
                        tA = tB = _time.perf_counter()

                    groups: Dict[int, list] = {}
                    sp_zp = []  # splain_z: (j, abs_row, nv, lo, hi)
                    row = written
                    run_lo = written  # dict-page run (string chunks)
                    ok = True
                    for j, page in enumerate(c.pages):
                        nv = page[3]
                        base = int(page_base[j])
                        r0 = roff[j]
                        skip = 0
                        if page[5]:
                            ln = int(np.frombuffer(hb, "<u4", 1, r0)[0])
                            skip = 4 + ln
                        if page[0] == "dict_z":
                            bw = int(hb[r0 + skip])
                            if bw <= 0 or skip + 1 >= int(page[4]):
                                ok = False
                                break
                            groups.setdefault(bw, []).append(
                                (j, row, nv, r0 + skip + 1,
                                 r0 + int(page[4]), (base - r0) * 8))
                        elif page[0] == "splain_z":
                            if has_spz and vals_list is not None \
                                    and row > run_lo:
                                str_chunks.append(
                                    (c.name, run_lo, row, vals_list))
                            sp_zp.append((j, row, nv, base + skip,
                                          base + int(page[4])))
                            run_lo = row + nv
                        else:
                            ext.copy_unaligned(
                                scratch, base + skip, out[c.name],
                                row * itemsize, nv * itemsize)
                        row += nv
                    if ok and has_spz and vals_list is not None \
                            and row > run_lo:
                        str_chunks.append(
                            (c.name, run_lo, row, vals_list))
                    if ok:
                        hb_t = _torch.from_numpy(hb)

                        def t64(x):
                            return _torch.tensor(x, dtype=_torch.int64)
                        for bw, pages_g in groups.items():
                            starts = [g[3] for g in pages_g]
                            ends = [g[4] for g in pages_g]
                            nvs = [g[2] for g in pages_g]
                            bshift = [g[5] for g in pages_g]
                            outs = []
                            drows = 0
                            dplaces = []
                            for _, abs_row, nv, _, _, _ in pages_g:
                                outs.append(drows)
                                dplaces.append((abs_row, drows, nv))
                                drows += nv
                            kind, ooff, ln_t, boff, val, _cnt = \
                                ext.parse_rle_runs_batch(
                                    hb_t, t64(starts), t64(ends),
                                    t64([bw] * len(starts)), t64(nvs),
                                    t64(outs), t64(bshift))
                            idx = ext.rle_decode(scratch, kind, ooff,
                                                 ln_t, boff, val, bw,
                                                 drows)
                            vals = idx if z_is_str else \
                                ext.gather_rows(dict_vals,
                                                idx.to(_torch.int64))
                            for abs_row, drow, nv in dplaces:
                                out[c.name][abs_row:abs_row + nv] = \
                                    vals[drow:drow + nv]
                        if sp_zp:
                            # one D2H of the decompressed byte-array
                            # payloads, one host parse + dict-encode
                            from .native_parquet import \
                                decode_splain_pages
                            pb = _torch.cat(
                                [scratch[a:b]
                                 for _, _, _, a, b in sp_zp]).cpu()
                            offs = []
                            cur0 = 0
                            for _, _, nv2, a, b in sp_zp:
                                offs.append((cur0, cur0 + (b - a), nv2))
                                cur0 += b - a
                            codes_all, spz_vals = decode_splain_pages(
                                pb, [o[0] for o in offs],
                                [o[1] for o in offs],
                                [o[2] for o in offs])
                            cpos = 0
                            for (j2, abs_row, nv2, _, _) in sp_zp:
                                seg2 = codes_all[cpos:cpos + nv2]
                                out[c.name][abs_row:abs_row + nv2] = \
                                    _torch.from_numpy(
                                        np.ascontiguousarray(seg2)).to(
                                        device, non_blocking=True)
                                str_chunks.append(
                                    (c.name, abs_row, abs_row + nv2,
                                     spz_vals))
                                cpos += nv2
                        written = row
                        cursors[c.name] = written
                        if _dt == "3":
                            print(f"[hs-chunk] {c.name} "
                                  f"pages={len(c.pages)} "
                                  f"pre={tA-t_c0:.3f} "
                                  f"fast={_time.perf_counter()-tB:.3f}",
                                  file=sys.stderr)
                        continue
                    # rare malformed/empty-dict page: fall through to
                    # the general per-page path below

                # level-length prefixes for every leveled page in one
                # small D2H (4 bytes each); the index streams and level
                # bytes D2H per page below — NEVER the whole chunk
                # (plain page payloads can be hundreds of MB)
                lvl_skips = [0] * len(c.pages)
                with_lvl = [j for j, p in enumerate(c.pages) if p[5]]
                if with_lvl:
                    pref = _torch.stack([
                        scratch[int(page_base[j]):
                                int(page_base[j]) + 4]
                        for j in with_lvl]).cpu()
                    lens = pref.numpy().view("<u4").ravel()
                    for j, ln in zip(with_lvl, lens):
                        lvl_skips[j] = 4 + int(ln)
                regions = {}  # (j, tag) -> (start in cat, length, abs)
                cat_parts = []
                cur = 0
                for j, page in enumerate(c.pages):
                    base = int(page_base[j])
                    spans = []
                    if page[5] and nullable_chunk:
                        spans.append(("lvl", base + 4,
                                      base + lvl_skips[j]))
                    if page[0] == "dict_z":
                        skip_j = lvl_skips[j] if page[5] else 0
                        spans.append(("idx", base + skip_j,
                                      base + int(page[4])))
                    for tag, a2, b2 in spans:
                        regions[(j, tag)] = (cur, b2 - a2, a2)
                        cat_parts.append(scratch[a2:b2])
                        cur += b2 - a2
                hb_all = (_torch.cat(cat_parts).cpu() if cat_parts
                          else None)
                # batch unmasked dict_z pages: one rle_decode + gather
                # per (chunk, bit-width) instead of per page
                zbatch: Dict[int, list] = {}
                for j, page in enumerate(c.pages):
                    nv = page[3]
                    base = int(page_base[j])
                    skip = 0
                    # V2 pages carry their mask from the layout parse
                    # (levels live uncompressed in the file); V1 pages
                    # decode the level prefix from the decompressed
                    # payload here
                    pmask = c.page_masks[j]
                    if pmask is None and page[5]:
                        ln = lvl_skips[j] - 4
                        skip = 4 + ln
                        if nullable_chunk:
                            s0, rln, _ = regions[(j, "lvl")]
                            lvl_bytes = hb_all[s0:s0 + rln].numpy()
                            from .native_parquet import _decode_defs
                            pmask = _decode_defs(
                                lvl_bytes.tobytes(), 0, ln, nv,
                                getattr(c, "max_def", 1))
                    n_valid = int(pmask.sum()) if pmask is not None \
                        else nv
                    if page[0] == "dict_z":
                        s0, rln, absbase = regions[(j, "idx")]
                        bw = int(hb_all[s0])
                        runs = list(ext.parse_rle_runs(
                            hb_all, s0 + 1, s0 + rln, bw, n_valid))
                        # bit-offsets: cat-local -> scratch coordinates
                        runs[3] = runs[3] + (absbase - s0) * 8
                        if pmask is None:
                            lst = zbatch.setdefault(
                                bw, {"runs": [], "rows": 0, "pages": []})
                            runs[1] = runs[1] + lst["rows"]
                            lst["runs"].append(runs)
                            lst["pages"].append(
                                (written, nv, lst["rows"]))
                            lst["rows"] += nv
                            written += nv
                            continue
                        idx = ext.rle_decode(scratch, *runs, bw,
                                             n_valid)
                        vals = idx if z_is_str else ext.gather_rows(
                            dict_vals, idx.to(_torch.int64))
                        if z_is_str and has_spz and \
                                vals_list is not None:
                            str_chunks.append((c.name, written,
                                               written + nv, vals_list))
                    elif page[0] == "splain_z":
                        # decompressed PLAIN byte-array payload: D2H,
                        # host parse + dict-encode, codes back up
                        from .native_parquet import decode_splain_pages
                        pb = scratch[base + skip:
                                     base + int(page[4])].cpu()
                        codes_np, pvals = decode_splain_pages(
                            pb, [0], [int(pb.numel())], [n_valid])
                        vals = _torch.from_numpy(
                            np.ascontiguousarray(codes_np)).to(device)
                        str_chunks.append((c.name, written,
                                           written + nv, pvals))
                    else:
                        if pmask is None:
                            ext.copy_unaligned(scratch, base + skip,
                                               out[c.name],
                                               written * itemsize,
                                               nv * itemsize)
                            written += nv
                            continue
                        vals = _torch.empty(
                            n_valid + 1, dtype=out[c.name].dtype,
                            device=device)
                        ext.copy_unaligned(scratch, base + skip, vals,
                                           0, n_valid * itemsize)
                        vals = vals[:n_valid]
                    if pmask is None:
                        out[c.name][written:written + nv] = vals
                    else:
                        mask_dev = _torch.from_numpy(pmask).to(device)
                        dst = out[c.name][written:written + nv]
                        dst.zero_()
                        dst[mask_dev] = vals
                        out_masks[c.name][written:written + nv] = \
                            mask_dev
                    written += nv
                for bw, lst in zbatch.items():
                    cat = [_torch.cat([r[k] for r in lst["runs"]])
                           for k in range(5)]
                    idx = ext.rle_decode(scratch, *cat, bw,
                                         lst["rows"])
                    vals = idx if z_is_str else ext.gather_rows(
                        dict_vals, idx.to(_torch.int64))
                    for woff, nv, rbase in lst["pages"]:
                        out[c.name][woff:woff + nv] = \
                            vals[rbase:rbase + nv]
                        if z_is_str and has_spz and \
                                vals_list is not None:
                            str_chunks.append((c.name, woff,
                                               woff + nv, vals_list))
                cursors[c.name] = written
                continue
            dict_vals = None
            is_str = c.is_string
            chunk_dict = None
            sp_codes: Dict[int, np.ndarray] = {}
            sp_vals = None
            if is_str:
                # string chunk (K1): indices ARE the column (int32 codes
                # into the host-parsed dictionary) — no value gather;
                # per-run str_chunks spans merge after the sync below.
                # PLAIN byte-array pages (whole-chunk PLAIN, or the
                # writer's mid-chunk dictionary overflow) parse host-
                # side in one GIL-released call and dictionary-encode
                # per chunk; their codes upload directly.
                chunk_dict = (c.dict_values(buf.numpy())
                              if c.dict_page else None)
                sp = [(j, p) for j, p in enumerate(c.pages)
                      if p[0] == "splain"]
                if sp:
                    from .native_parquet import decode_splain_pages
                    sp_counts = [
                        (int(c.page_masks[j].sum())
                         if c.page_masks[j] is not None else p[3])
                        for j, p in sp]
                    all_codes, sp_vals = decode_splain_pages(
                        buf, [p[1] for _, p in sp],
                        [p[2] for _, p in sp], sp_counts)
                    cur0 = 0
                    for (j, _), nv2 in zip(sp, sp_counts):
                        sp_codes[j] = all_codes[cur0:cur0 + nv2]
                        cur0 += nv2
            elif c.encoding == "dict":
                # K1 dictionary path: decode the dictionary page once;
                # pages gather through it (a chunk can also carry PLAIN
                # pages — the writer's mid-chunk dictionary-overflow
                # fallback)
                dict_off, dict_n = c.dict_page
                dict_vals = _torch.empty(
                    dict_n + 1,  # +1 slack for the 4B-overread
                    dtype=out[c.name].dtype, device=device)
                ext.copy_unaligned(dev_bytes, dict_off, dict_vals, 0,
                                   dict_n * itemsize)
                dict_vals = dict_vals[:dict_n].contiguous()
            if not any(m is not None for m in c.page_masks):
                run_lo = written  # current dict-page run (string chunks)
                for j, page in enumerate(c.pages):
                    nv = page[3] if page[0] in ("dict", "splain") \
                        else page[2]
                    if page[0] == "dict":
                        _, p_start, p_end, _, bw = page
                        runs = ext.parse_rle_runs(buf, p_start, p_end, bw,
                                                  nv)
                        idx = ext.rle_decode(dev_bytes, *runs, bw, nv)
                        if is_str:
                            out[c.name][written:written + nv] = idx
                        else:
                            out[c.name][written:written + nv] = \
                                ext.gather_rows(dict_vals,
                                                idx.to(_torch.int64))
                    elif page[0] == "splain":
                        if written > run_lo:
                            str_chunks.append((c.name, run_lo, written,
                                               chunk_dict))
                        out[c.name][written:written + nv] = \
                            _torch.from_numpy(
                                np.ascontiguousarray(sp_codes[j])).to(
                                device, non_blocking=True)
                        str_chunks.append((c.name, written,
                                           written + nv, sp_vals))
                        run_lo = written + nv
                    else:
                        _, off, _ = page
                        ext.copy_unaligned(dev_bytes, off, out[c.name],
                                           written * itemsize,
                                           nv * itemsize)
                    written += nv
                if is_str and written > run_lo:
                    str_chunks.append((c.name, run_lo, written,
                                       chunk_dict))
            else:
                # nullable chunk: decode every page's compacted values
                # into ONE temp, then a single chunk-wide scatter (per-
                # page boolean indexing costs a kernel pair per page —
                # thousands of launches on multi-page files)
                n_rows = sum(p[3] if p[0] in ("dict", "splain")
                             else p[2] for p in c.pages)
                valid_counts = [
                    (int(m.sum()) if m is not None
                     else (p[3] if p[0] in ("dict", "splain")
                           else p[2]))
                    for p, m in zip(c.pages, c.page_masks)]
                total_valid = sum(valid_counts)
                tmp = _torch.empty(total_valid + 1,
                                   dtype=out[c.name].dtype, device=device)
                cur = 0
                row_pos = written
                run_lo = written
                for j, (page, n_valid) in enumerate(
                        zip(c.pages, valid_counts)):
                    if page[0] == "dict":
                        _, p_start, p_end, _, bw = page
                        runs = ext.parse_rle_runs(buf, p_start, p_end, bw,
                                                  n_valid)
                        idx = ext.rle_decode(dev_bytes, *runs, bw,
                                             n_valid)
                        tmp[cur:cur + n_valid] = idx if is_str else \
                            ext.gather_rows(dict_vals,
                                            idx.to(_torch.int64))
                        row_pos += page[3]
                    elif page[0] == "splain":
                        if row_pos > run_lo:
                            str_chunks.append((c.name, run_lo, row_pos,
                                               chunk_dict))
                        tmp[cur:cur + n_valid] = _torch.from_numpy(
                            np.ascontiguousarray(sp_codes[j])).to(
                            device, non_blocking=True)
                        str_chunks.append((c.name, row_pos,
                                           row_pos + page[3], sp_vals))
                        row_pos += page[3]
                        run_lo = row_pos
                    else:
                        _, off, _ = page
                        ext.copy_unaligned(dev_bytes, off, tmp,
                                           cur * itemsize,
                                           n_valid * itemsize)
                        row_pos += page[2]
                    cur += n_valid
                if is_str and row_pos > run_lo:
                    str_chunks.append((c.name, run_lo, row_pos,
                                       chunk_dict))
                chunk_mask = np.concatenate([
                    (m if m is not None
                     else np.ones(p[3] if p[0] in ("dict", "splain")
                                  else p[2], dtype=bool))
                    for p, m in zip(c.pages, c.page_masks)])
                mask_dev = _torch.from_numpy(chunk_mask).to(device)
                dst = out[c.name][written:written + n_rows]
                dst.zero_()
                dst[mask_dev] = tmp[:total_valid]
                out_masks[c.name][written:written + n_rows] = mask_dev
                written += n_rows
            cursors[c.name] = written
        # pinned buffer must stay alive until the stream drains; the
        # caller-side synchronize below holds them via `bufs`
        return buf

    # phase 1+2 pipeline: file reads stream into decode units as each
    # read completes (a strict all-reads-then-all-decodes barrier left
    # the H2D + decode tail serialized after ~0.46 s of source reads)
    import time as _time
    _dt = os.environ.get("HS_DECODE_TIMING")
    _t0 = _time.perf_counter()
    infos: List[Optional[tuple]] = [None] * len(paths)
    split_rgs = len(paths) < n_streams

    import threading
    dev_bufs: List[Optional["_torch.Tensor"]] = [None] * len(paths)
    keepalive: List["_torch.Tensor"] = []  # pinned codec staging buffers
    upload_events: List[Optional["_torch.cuda.Event"]] = \
        [None] * len(paths)
    upload_lock = threading.Lock()

    def host_unit(i):
        """Non-native file inside a mixed batch: pyarrow-read on this
        worker thread, upload into the preallocated slices (the native
        files around it stay on the device decode path)."""
        from .native_parquet import StrCol
        cols, fmasks, n = _pyarrow_file_dict(paths[i], columns)
        assert n == row_counts[i], (paths[i], n, row_counts[i])
        low = {k.lower(): v for k, v in cols.items()}
        mlow = {k.lower(): v for k, v in fmasks.items()}
        base = int(file_base[i])
        stream = streams[i % n_streams]
        with _torch.cuda.stream(stream):
            for name in names:
                arr = low[name.lower()]
                if isinstance(arr, StrCol):
                    codes = _torch.from_numpy(
                        np.ascontiguousarray(arr.codes))
                    out[name][base:base + n] = codes.to(
                        device, non_blocking=True)
                    str_chunks.append((name, base, base + n,
                                       arr.values))
                else:
                    t = _torch.from_numpy(np.ascontiguousarray(
                        arr.astype(dtypes[name], copy=False)))
                    out[name][base:base + n] = t.to(device,
                                                    non_blocking=True)
                if name in out_masks:
                    m = mlow.get(name.lower())
                    if m is not None:
                        out_masks[name][base:base + n] =                             _torch.from_numpy(m).to(device,
                                                    non_blocking=True)
        return True

    def decode_unit(u):
        uid, (i, row_off, chunks) = u
        buf, size, _ = infos[i]
        if _dt == "2":
            t0u = _time.perf_counter()
        stream = streams[uid % n_streams]
        with _torch.cuda.stream(stream):
            with upload_lock:
                if dev_bufs[i] is None:
                    # upload only the file's bytes (+4B decode slack),
                    # not the whole pooled size class
                    dev_bufs[i] = buf[:size + 4].to(device,
                                                    non_blocking=True)
                    ev = _torch.cuda.Event()
                    ev.record(stream)
                    upload_events[i] = ev
                else:
                    stream.wait_event(upload_events[i])
            _decode_on_stream(i, buf, size, chunks, dev_bufs[i], row_off)
        if _dt == "2":
            print(f"[hs-unit] u{uid} rows={chunks[0].num_values} "
                  f"chunks={[(c.name, c.encoding, len(c.pages)) for c in chunks]} "
                  f"{_time.perf_counter()-t0u:.3f}s", file=sys.stderr)
        return True

    _t1 = _time.perf_counter()
    _dw = int(os.environ.get("HS_DECODE_WORKERS", "16"))
    n_units = 0
    failed = False
    if len(paths) > 2 and _dw > 1:
        from concurrent.futures import as_completed
        import itertools
        uid_gen = itertools.count()
        dec_futs = []
        with ThreadPoolExecutor(max_workers=16) as read_pool, \
                ThreadPoolExecutor(max_workers=_dw) as dec_pool:
            read_futs = {read_pool.submit(load_file, i): i
                         for i in range(len(paths))}
            for fut in as_completed(read_futs):
                i = read_futs[fut]
                inf = fut.result()
                if inf is None:
                    # no native layout: per-file pyarrow host read
                    infos[i] = ("host", 0, None)
                    dec_futs.append(dec_pool.submit(host_unit, i))
                    n_units += 1
                    continue
                infos[i] = inf
                if failed:
                    continue  # drain remaining reads; fallback below
                if split_rgs:
                    for row_off, rg_chunks in split_row_groups(inf[2]):
                        dec_futs.append(dec_pool.submit(
                            decode_unit,
                            (next(uid_gen), (i, row_off, rg_chunks))))
                        n_units += 1
                else:
                    dec_futs.append(dec_pool.submit(
                        decode_unit, (next(uid_gen), (i, 0, inf[2]))))
                    n_units += 1
            if not failed:
                for f in dec_futs:
                    f.result()
    else:
        for i in range(len(paths)):
            infos[i] = load_file(i)
            if infos[i] is None:
                infos[i] = ("host", 0, None)
        if not failed:
            uid = 0
            for i in range(len(paths)):
                if infos[i][0] == "host":
                    host_unit(i)
                    n_units += 1
                    continue
                unit_list = (split_row_groups(infos[i][2]) if split_rgs
                             else [(0, infos[i][2])])
                for row_off, rg_chunks in unit_list:
                    decode_unit((uid, (i, row_off, rg_chunks)))
                    uid += 1
                    n_units += 1
    bufs = [inf[0] if inf is not None else None for inf in infos]
    if failed:
        for bq in bufs:
            if isinstance(bq, _torch.Tensor):
                _pinned_put(bq)
        for bq in keepalive:
            _pinned_put(bq)
        return fallback()
    _t2 = _time.perf_counter()    # order the default stream after every worker stream, then host-sync
    # so the pinned buffers can be recycled
    cur = _torch.cuda.current_stream()
    for s in streams:
        cur.wait_stream(s)
    cur.synchronize()
    if _dt:
        print(f"[hs-decode] files={len(paths)} units={n_units} "
              f"read+decode(pipe) {_t2-_t1:.3f}s "
              f"sync {_time.perf_counter()-_t2:.3f}s", file=sys.stderr)
    # one device reduction + sync for every chunk's status words (a
    # per-tensor .any() paid ~1.4 ms of sync each across ~50 chunks)
    snappy_bad = bool(_torch.cat(statuses).ne(0).any().item()) \
        if statuses else False
    if snappy_bad or \
            not all(b is not None and b is not False for b in bufs):
        for b in bufs:
            if isinstance(b, _torch.Tensor):
                _pinned_put(b)
        for b in keepalive:
            _pinned_put(b)
        return fallback()
    for b in bufs:
        if isinstance(b, _torch.Tensor):
            _pinned_put(b)
    for b in keepalive:
        _pinned_put(b)
    if string_cols:
        from ..execution.columnar import StringColumn
        by_name: Dict[str, list] = {}
        for name, lo, hi, vals in str_chunks:
            by_name.setdefault(name, []).append((lo, hi, vals))
        for name in string_cols:
            chunks_n = by_name.get(name, [])
            dicts = [v for _, _, v in chunks_n]
            # parquet dictionaries are in INSERTION order; StringColumn
            # requires SORTED values (code order = lex order), so any
            # unsorted dictionary forces the remap path
            if dicts and all(d == dicts[0] for d in dicts[1:]) and \
                    all(dicts[0][i] <= dicts[0][i + 1]
                        for i in range(len(dicts[0]) - 1)):
                merged = list(dicts[0])
            else:
                merged = sorted(set().union(*map(set, dicts))) \
                    if dicts else []
                if not merged:
                    merged = [""]  # all-null column: masked 0-codes
                vi = {v: i for i, v in enumerate(merged)}
                for lo, hi, vals in chunks_n:
                    if vals == merged:
                        continue
                    lut = _torch.tensor(
                        [vi[v] for v in vals] or [0],
                        dtype=_torch.int32, device=device)
                    seg = out[name][lo:hi]
                    out[name][lo:hi] = lut[seg.long()]
            out[name] = StringColumn(out[name], merged)
    return ColumnBatch(out, out_masks), row_counts
