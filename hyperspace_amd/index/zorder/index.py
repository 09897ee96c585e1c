"""ZOrderCoveringIndex: the covering slice sorted by interleaved-bit
z-address.

Reference: index/zordercovering/ZOrderCoveringIndex.scala —
min/max stats (:50-95, K11 device reductions), per-row z-address
(:97-154, K10 bit-interleave kernel), range partition + sort.  Multi-
column keys are min/max scaled into equal bit budgets before
interleaving (the reference's ZOrderField per-type encodings,
ZOrderField.scala:26-569, reduce to the same normalize-then-scale on our
columnar types).

Query-time file skipping uses the Parquet column statistics written with
every index file (the reference relies on Spark's row-group stats
pushdown the same way).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Optional, Tuple

import torch

from ..base import Index, IndexerContext
from ...config import IndexConstants
from ... import ops
from ...execution.columnar import ColumnBatch
from ...log.entry import Schema, register_derived_dataset, ZORDER_INDEX_TYPE
from ...sources.parquet_io import bucket_file_name, write_batch_parquet


class ZOrderCoveringIndex(Index):
    def __init__(self, indexed_columns: List[str],
                 included_columns: List[str], schema: Schema,
                 properties: Dict[str, str]):
        self.indexed_columns = list(indexed_columns)
        self.included_columns = list(included_columns)
        self.schema = schema
        self._properties = dict(properties)

    @property
    def kind(self) -> str:
        return "ZOrderCoveringIndex"

    @property
    def kind_abbr(self) -> str:
        return "ZCI"

    def indexed_columns_list(self):
        return list(self.indexed_columns)

    def referenced_columns(self):
        return self.indexed_columns + self.included_columns

    def with_new_properties(self, props):
        return ZOrderCoveringIndex(self.indexed_columns,
                                   self.included_columns, self.schema,
                                   props)

    @property
    def can_handle_deleted_files(self) -> bool:
        return (self._properties.get(IndexConstants.LINEAGE_PROPERTY,
                                     "false").lower() == "true")

    def to_json(self):
        return {"type": ZORDER_INDEX_TYPE,
                "indexedColumns": self.indexed_columns,
                "includedColumns": self.included_columns,
                "schema": self.schema.to_json(),
                "properties": self._properties}

    @staticmethod
    def from_json(d):
        return ZOrderCoveringIndex(
            d["indexedColumns"], d["includedColumns"],
            Schema.from_json(d["schema"]), d.get("properties", {}))

    # -- build -------------------------------------------------------------
    def write(self, ctx: IndexerContext, index_data: ColumnBatch
              ) -> List[str]:
        os.makedirs(ctx.index_data_path, exist_ok=True)
        batch = index_data
        n = batch.num_rows
        if n == 0:
            return []

        z = self._zaddress(ctx, batch)
        perm = ops.sort_perm(z)
        batch = batch.gather(perm)

        # split into target-size files (zorder.targetSourceBytesPerPartition)
        target = ctx.session.conf.get(
            IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION)
        row_bytes = max(1, batch.nbytes() // n)
        rows_per_file = max(1, int(target) // row_bytes)
        from ...parallel import dist_context as dc
        task_id = dc.get_rank()
        host = batch.to("cpu") if batch.device.type == "cuda" else batch
        written = []
        chunk = 0
        for start in range(0, n, rows_per_file):
            out = os.path.join(ctx.index_data_path,
                               bucket_file_name(task_id, chunk))
            write_batch_parquet(host.slice(start,
                                           min(n, start + rows_per_file)),
                                out)
            written.append(out)
            chunk += 1
        return written

    def _zaddress(self, ctx: IndexerContext,
                  batch: ColumnBatch) -> torch.Tensor:
        """Scale each indexed column into its bit budget — min/max by
        default, sampled quantile ranks when zorder.quantile.enabled (the
        reference's percentile-bucketed ZOrderField for skewed data,
        ZOrderField.scala) — then interleave (K11 + K10)."""
        from ...config import IndexConstants
        n_cols = len(self.indexed_columns)
        bits = 64 // n_cols
        quantile = bool(ctx.session.conf.get(
            IndexConstants.ZORDER_QUANTILE_ENABLED))
        cols_u64 = []
        for c in self.indexed_columns:
            norm = ops.normalize_key(batch.tensor(c))
            s = ops.cpu_ref._as_unsigned_sortable(norm)
            if quantile:
                scaled = _quantile_rank(s, bits)
            else:
                lo64, hi64 = int(s.min()), int(s.max())
                from ...parallel import dist_context as dc
                if dc.is_distributed() and dc.get_world_size() > 1:
                    import torch.distributed as dist
                    t = dc.collective_tensor([lo64, -hi64])
                    dist.all_reduce(t, op=dist.ReduceOp.MIN)
                    lo64, hi64 = int(t[0]), -int(t[1])
                span = max(1, hi64 - lo64)
                scaled = ((s - lo64).to(torch.float64) / span
                          * float((1 << bits) - 1)).to(torch.int64)
            cols_u64.append(scaled << (64 - bits))
        return ops.zorder_key(cols_u64, bits)

    def optimize(self, ctx: IndexerContext,
                 files_to_optimize: List[str]) -> List[str]:
        """Re-sort + rewrite the given files as one z-ordered chunk set."""
        from ...sources.parquet_io import read_files_batch
        os.makedirs(ctx.index_data_path, exist_ok=True)
        batch, _ = read_files_batch(sorted(files_to_optimize))
        if ctx.session.device.type == "cuda":
            batch = batch.to(ctx.session.device)
        return self.write(ctx, batch)

    def refresh_incremental(self, ctx, appended_batch, deleted_file_ids,
                            previous_files):
        written: List[str] = []
        if appended_batch is not None and appended_batch.num_rows:
            written.extend(self.write(ctx, appended_batch))
        kept = list(previous_files)
        if deleted_file_ids:
            from ...sources.parquet_io import read_files_batch
            from ...exceptions import HyperspaceException
            if not self.can_handle_deleted_files:
                raise HyperspaceException("Index lacks lineage")
            ids = torch.tensor(sorted(deleted_file_ids), dtype=torch.int64)
            lineage_col = IndexConstants.DATA_FILE_NAME_ID_COLUMN
            kept = []
            for p in previous_files:
                sub, _ = read_files_batch([p])
                keep_mask = ~ops.isin_sorted(sub.tensor(lineage_col), ids)
                if bool(keep_mask.all()):
                    kept.append(p)
                    continue
                sub = sub.gather(
                    torch.nonzero(keep_mask, as_tuple=False).flatten())
                out = os.path.join(
                    ctx.index_data_path,
                    bucket_file_name(9, len(written)))
                write_batch_parquet(sub, out)
                written.append(out)
        return written, kept

    # -- query: stats-based file skipping ---------------------------------
    @staticmethod
    def prune_files_by_stats(files: List[str], column: str, op: str,
                             value) -> Tuple[List[str], int]:
        """Keep files whose Parquet column stats admit the predicate."""
        import pyarrow.parquet as pq
        kept, skipped = [], 0
        for p in files:
            try:
                md = pq.ParquetFile(p).metadata
                idx = md.schema.to_arrow_schema().get_field_index(column)
                lo = hi = None
                for rg in range(md.num_row_groups):
                    st = md.row_group(rg).column(idx).statistics
                    if st is None or not st.has_min_max:
                        lo = hi = None
                        break
                    lo = st.min if lo is None else min(lo, st.min)
                    hi = st.max if hi is None else max(hi, st.max)
                if lo is None:
                    kept.append(p)
                    continue
            except Exception:  # noqa: BLE001 - conservative on stat errors
                kept.append(p)
                continue
            admit = True
            if op == "=":
                admit = lo <= value <= hi
            elif op == "<":
                admit = lo < value
            elif op == "<=":
                admit = lo <= value
            elif op == ">":
                admit = hi > value
            elif op == ">=":
                admit = hi >= value
            if admit:
                kept.append(p)
            else:
                skipped += 1
        return kept, skipped

    def statistics(self):
        return {"indexedColumns": self.indexed_columns,
                "includedColumns": self.included_columns,
                "zorder": True}


def _quantile_rank(s: torch.Tensor, bits: int) -> torch.Tensor:
    """Sampled-quantile scaling: rank each value against equi-depth
    quantile boundaries (z-cells for skewed columns).  Boundary
    resolution is capped at 2^12 cells — ranks are shifted up into the
    full per-column bit budget.  Sampling + sort run on the device via
    the radix kernel."""
    n = s.numel()
    cell_bits = min(bits, 12)
    n_bounds = (1 << cell_bits) - 1
    sample_n = min(n, max(4096, n_bounds * 8))
    if n > sample_n:
        step = max(1, n // sample_n)
        sample = s[::step].contiguous()
    else:
        sample = s.contiguous()
    norm_sample = sample ^ (-0x8000000000000000)
    perm = ops.sort_perm(norm_sample)
    sample_sorted = ops.gather_rows(sample, perm)
    m = sample_sorted.numel()
    idx = (torch.arange(1, n_bounds + 1, dtype=torch.float64)
           * m / (n_bounds + 1)).to(torch.int64).clamp(max=m - 1)
    bounds = sample_sorted[idx.to(sample_sorted.device)]
    rank = torch.searchsorted(bounds.contiguous(), s)
    return rank << (bits - cell_bits)


register_derived_dataset(ZORDER_INDEX_TYPE, ZOrderCoveringIndex)
