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

    Defaults to uncompressed PLAIN pages (device-decodable).
    """
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
    counts (for lineage / per-file segmentation)."""
    import pyarrow.parquet as pq
    import pyarrow as pa
    from ..execution.columnar import ColumnBatch
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
