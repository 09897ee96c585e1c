"""Benchmark helpers: synthetic Parquet generation shaped per
BASELINE.json's configs (covering index on (key, val); lineitem⋈orders-
shaped join tables)."""

from __future__ import annotations

import os
from typing import List, Optional

import numpy as np


def generate_fact_parquet(out_dir: str, total_bytes: int, seed: int = 0,
                          key_hi: Optional[int] = None,
                          rows_per_file: int = 4_000_000) -> List[str]:
    """Fact table: (key int64, val float64) rows ≈16B each, uncompressed
    PLAIN pages (device-decodable).  Returns written file paths."""
    from .sources.native_parquet import write_parquet_native
    os.makedirs(out_dir, exist_ok=True)
    bytes_per_row = 16
    n_rows = max(1, total_bytes // bytes_per_row)
    if key_hi is None:
        key_hi = max(1000, n_rows // 8)
    rng = np.random.default_rng(seed)
    paths = []
    written = 0
    i = 0
    while written < n_rows:
        n = min(rows_per_file, n_rows - written)
        p = os.path.join(out_dir, f"part-{seed:03d}-{i:05d}.parquet")
        write_parquet_native({
            "key": rng.integers(0, key_hi, n, dtype=np.int64),
            "val": rng.random(n),
        }, p)
        paths.append(p)
        written += n
        i += 1
    return paths


def generate_dim_parquet(out_dir: str, n_rows: int, seed: int = 0
                         ) -> List[str]:
    """Dimension table: (key int64 unique, status int64)."""
    from .sources.native_parquet import write_parquet_native
    os.makedirs(out_dir, exist_ok=True)
    rng = np.random.default_rng(seed + 7)
    p = os.path.join(out_dir, f"dim-{seed:03d}.parquet")
    write_parquet_native({
        "key": np.arange(n_rows, dtype=np.int64),
        "status": rng.integers(0, 5, n_rows, dtype=np.int64),
    }, p)
    return [p]
