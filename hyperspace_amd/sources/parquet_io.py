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
import uuid
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

_BUCKET_RE = re.compile(r".*_(\d+)(?:\.\w+)*\.parquet$")


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
    if compression is None and not any(
            isinstance(c, StringColumn) for c in batch.columns.values()):
        from .native_parquet import write_parquet_native
        cols = {name: t.numpy() for name, t in batch.columns.items()}
        return write_parquet_native(cols, path)
    import pyarrow.parquet as pq
    table = batch.to_arrow()
    pq.write_table(table, path, compression=compression or "NONE",
                   use_dictionary=use_dictionary,
                   write_statistics=True,
                   data_page_version="1.0")
    st = os.stat(path)
    return st.st_size, int(st.st_mtime * 1000)


def read_files_batch(paths: List[str], columns: Optional[List[str]] = None):
    """Read parquet files into a single host ColumnBatch, plus per-file row
    counts (for lineage / per-file segmentation).  Native-layout files
    (uncompressed PLAIN numeric) bypass pyarrow."""
    import numpy as np
    from ..execution.columnar import ColumnBatch
    from .native_parquet import read_native_host

    per_file = []
    row_counts = []
    native_ok = True
    for p in paths:
        cols = read_native_host(p, columns)
        if cols is None:
            native_ok = False
            break
        per_file.append(cols)
        n = len(next(iter(cols.values()))) if cols else 0
        row_counts.append(n)
    if native_ok and paths:
        names = list(per_file[0].keys())
        if columns is not None:
            order = {c.lower(): i for i, c in enumerate(columns)}
            names.sort(key=lambda n: order.get(n.lower(), 99))
        merged = {}
        for name in names:
            arrs = [f[name] for f in per_file]
            merged[name] = torch.from_numpy(
                np.concatenate(arrs) if len(arrs) > 1 else arrs[0])
        return ColumnBatch(merged), row_counts

    import pyarrow.parquet as pq
    import pyarrow as pa
    tables = []
    row_counts = []
    for p in paths:
        t = pq.read_table(p, columns=columns)
        tables.append(t)
        row_counts.append(t.num_rows)
    if not tables:
        return ColumnBatch({}), []
    table = pa.concat_tables(tables, promote_options="default")
    return ColumnBatch.from_arrow(table), row_counts


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
    from .native_parquet import read_native_layout

    # read each file ONCE into a pinned buffer (+4B slack for the decode
    # kernel), parse the layout from the same bytes, upload async; file
    # reads overlap on a thread pool (readinto releases the GIL)
    from concurrent.futures import ThreadPoolExecutor

    def load_one(p):
        size = os.path.getsize(p)
        buf = _torch.empty(size + 4, dtype=_torch.uint8, pin_memory=True)
        view = memoryview(buf.numpy())
        with open(p, "rb", buffering=0) as f:
            f.readinto(view[:size])
        lay = read_native_layout(p, columns, data=view[:size])
        return buf, (lay[1] if lay is not None else None)

    if len(paths) > 2:
        with ThreadPoolExecutor(max_workers=16) as pool:
            loaded = list(pool.map(load_one, paths))
    else:
        loaded = [load_one(p) for p in paths]
    if any(lay is None for _, lay in loaded):
        batch, row_counts = read_files_batch(paths, columns)
        return batch.to(device), row_counts
    bufs = [b for b, _ in loaded]
    layouts = [lay for _, lay in loaded]

    ext = native_ext.ext()
    # column structure from the first file
    names = [c.name for c in layouts[0]]
    if columns is not None:
        order = {c.lower(): i for i, c in enumerate(columns)}
        names.sort(key=lambda n: order.get(n.lower(), 99))
    dtypes = {c.name: c.np_dtype for c in layouts[0]}
    totals = {n: 0 for n in names}
    row_counts = []
    for chunks in layouts:
        nrows = chunks[0].num_values if chunks else 0
        row_counts.append(nrows)
        for c in chunks:
            totals[c.name] += c.num_values

    np_to_torch = {np.dtype("int64"): _torch.int64,
                   np.dtype("int32"): _torch.int32,
                   np.dtype("float64"): _torch.float64,
                   np.dtype("float32"): _torch.float32}
    out = {n: _torch.empty(totals[n], dtype=np_to_torch[dtypes[n]],
                           device=device) for n in names}
    written = {n: 0 for n in names}
    for buf, chunks in zip(bufs, layouts):
        dev_bytes = buf.to(device, non_blocking=True)
        for c in chunks:
            itemsize = c.np_dtype.itemsize
            for off, nv in c.pages:
                ext.copy_unaligned(dev_bytes, off, out[c.name],
                                   written[c.name] * itemsize,
                                   nv * itemsize)
                written[c.name] += nv
    return ColumnBatch(out), row_counts
