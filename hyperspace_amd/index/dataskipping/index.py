"""DataSkippingIndex: one row per source file of sketch aggregates.

Reference: index/dataskipping/DataSkippingIndex.scala —
createIndexData (:291-317, groupBy input_file_name -> per-file sketch
aggregation = K8 segmented reductions + device bloom build),
translateFilterCondition (:143-185), DataSkippingFileIndex
(execution/DataSkippingFileIndex.scala:32-74: prunes the source file
list with the translated predicate; unconvertible files are KEPT).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Optional, Tuple

import torch

from ..base import Index, IndexerContext
from ...exceptions import HyperspaceException
from ...log.entry import (Schema, register_derived_dataset,
                          DATASKIPPING_INDEX_TYPE)
from ...plan.expr import And, Expr, Not, Or
from .sketches import Sketch, sketch_from_json


class DataSkippingIndex(Index):
    def __init__(self, sketches: List[Sketch], schema: Schema,
                 properties: Dict[str, str]):
        self.sketches = list(sketches)
        self.schema = schema  # source column schema for sketched columns
        self._properties = dict(properties)

    @property
    def kind(self) -> str:
        return "DataSkippingIndex"

    @property
    def kind_abbr(self) -> str:
        return "DS"

    def indexed_columns_list(self) -> List[str]:
        return sorted({s.base_column for s in self.sketches})

    def referenced_columns(self) -> List[str]:
        return self.indexed_columns_list()

    def with_new_properties(self, props):
        return DataSkippingIndex(self.sketches, self.schema, props)

    @property
    def can_handle_deleted_files(self) -> bool:
        # per-file rows: deleted files simply drop out of the join
        return True

    def to_json(self) -> Dict[str, Any]:
        return {"type": DATASKIPPING_INDEX_TYPE,
                "sketches": [s.to_json() for s in self.sketches],
                "schema": self.schema.to_json(),
                "properties": self._properties}

    @staticmethod
    def from_json(d) -> "DataSkippingIndex":
        return DataSkippingIndex(
            [sketch_from_json(s) for s in d.get("sketches", [])],
            Schema.from_json(d.get("schema", {"fields": []})),
            d.get("properties", {}))

    # -- build -------------------------------------------------------------
    def write(self, ctx: IndexerContext, index_data) -> List[str]:
        """index_data: dict with '_data_file_id' per-file ids and the
        sketch aggregate tensors (built by config.create_index).

        Output is split to the targetIndexDataFileSize (256 MiB default)
        capped at maxIndexDataFileCount files — the reference's writeImpl
        sizing, index/dataskipping/DataSkippingIndex.scala:187-206."""
        os.makedirs(ctx.index_data_path, exist_ok=True)
        import pyarrow as pa
        import pyarrow.parquet as pq
        from ...config import IndexConstants as IC
        arrays = {}
        row_bytes = 0
        n_rows = 0
        for name, t in index_data.items():
            n_rows = t.shape[0]
            row_bytes += t.element_size() * (t.shape[1] if t.dim() == 2
                                             else 1)
            if t.dim() == 2:  # bloom word matrix -> fixed-size list column
                arrays[name] = pa.array(list(t.numpy()))
            else:
                arrays[name] = pa.array(t.numpy())
        target = int(ctx.session.conf.get(
            IC.DATASKIPPING_TARGET_INDEX_DATA_FILE_SIZE))
        max_files = int(ctx.session.conf.get(
            IC.DATASKIPPING_MAX_INDEX_DATA_FILE_COUNT,
            IC.DATASKIPPING_MAX_INDEX_DATA_FILE_COUNT_DEFAULT))
        per_file = max(1, target // max(1, row_bytes))
        n_files = min(max_files,
                      max(1, -(-n_rows // per_file))) if n_rows else 1
        per_file = -(-n_rows // n_files) if n_rows else 1
        table = pa.table(arrays)
        written = []
        for i in range(n_files):
            out = os.path.join(
                ctx.index_data_path,
                f"part-{i:05d}-sketches_00000.c000.parquet")
            pq.write_table(table.slice(i * per_file, per_file), out,
                           compression="NONE", use_dictionary=False)
            written.append(out)
        return written

    def refresh_full(self, ctx, df):
        raise NotImplementedError

    def refresh_incremental_files(self, ctx, appended_files: List[str],
                                  deleted_file_ids: List[int],
                                  previous_files: List[str]):
        """Sketch the appended files into a new index data file; rewrite
        old sketch files dropping rows of deleted source files
        (the per-file-row analog of the covering incremental refresh)."""
        import uuid as _uuid
        from ...sources.parquet_io import read_files_batch
        from ...utils.resolver import resolve_all
        os.makedirs(ctx.index_data_path, exist_ok=True)
        written: List[str] = []
        if appended_files:
            batch, row_counts = read_files_batch(
                sorted(appended_files), columns=self.referenced_columns())
            if ctx.session.device.type == "cuda":
                batch = batch.to(ctx.session.device)
            seg = torch.zeros(len(appended_files) + 1, dtype=torch.int64)
            seg[1:] = torch.cumsum(
                torch.tensor(row_counts, dtype=torch.int64), 0)
            file_ids = torch.tensor(
                [ctx.file_id_tracker.add_file(
                    p, os.path.getsize(p),
                    int(os.stat(p).st_mtime * 1000))
                 for p in sorted(appended_files)], dtype=torch.int64)
            from ...config import IndexConstants as IC
            data: Dict[str, torch.Tensor] = {
                IC.DATA_FILE_NAME_ID_COLUMN: file_ids}
            for sketch in self.sketches:
                base = sketch.base_column
                values = sketch.compute_values(batch.tensor(base))
                dtype_name = sketch.value_type(
                    self.schema.field_type(base) or "long")
                data.update(sketch.aggregate(values, seg, dtype_name,
                                             batch.mask(base)))
            out = os.path.join(
                ctx.index_data_path,
                f"part-00001-{_uuid.uuid4().hex[:8]}_00000.c000.parquet")
            import pyarrow as pa
            import pyarrow.parquet as pq
            arrays = {}
            for name, t in data.items():
                arrays[name] = (pa.array(list(t.numpy())) if t.dim() == 2
                                else pa.array(t.numpy()))
            pq.write_table(pa.table(arrays), out, compression="NONE",
                           use_dictionary=False)
            written.append(out)

        kept = list(previous_files)
        if deleted_file_ids:
            import pyarrow.parquet as pq
            import pyarrow as pa
            gone = set(deleted_file_ids)
            kept = []
            for p in previous_files:
                t = pq.read_table(p)
                ids = t.column("_data_file_id").to_pylist()
                keep_rows = [i for i, fid in enumerate(ids)
                             if fid not in gone]
                if len(keep_rows) == len(ids):
                    kept.append(p)
                    continue
                if keep_rows:
                    out = os.path.join(
                        ctx.index_data_path,
                        f"part-00002-{_uuid.uuid4().hex[:8]}_00000"
                        ".c000.parquet")
                    pq.write_table(t.take(keep_rows), out,
                                   compression="NONE",
                                   use_dictionary=False)
                    written.append(out)
        return written, kept

    # -- query -------------------------------------------------------------
    def load_sketch_data(self, entry, device=None):
        import pyarrow.parquet as pq
        import numpy as np
        from ...execution.columnar import ColumnBatch
        paths = [p for p in entry.content.os_files()
                 if p.endswith(".parquet")]
        t = pq.read_table(paths)
        cols = {}
        for name, col in zip(t.column_names, t.columns):
            arr = col.to_numpy(zero_copy_only=False)
            if arr.dtype == object:  # list column (bloom words)
                arr = np.stack([np.asarray(x, dtype=np.int64)
                                for x in arr])
                cols[name] = torch.from_numpy(arr)
            else:
                arr = np.ascontiguousarray(arr)
                if not arr.flags.writeable:
                    arr = arr.copy()
                cols[name] = torch.from_numpy(arr)
        if device is not None and str(device) != "cpu":
            cols = {k: v.to(device) for k, v in cols.items()}
        return _SketchData(cols)

    def translate_filter(self, pred: Expr, sketch_data
                         ) -> Optional[torch.Tensor]:
        """Filter predicate -> per-file "may contain" mask, or None when
        nothing is convertible (reference translateFilterCondition:
        AND: convertible side suffices; OR: both sides must convert;
        NOT: not convertible — conservative)."""
        if isinstance(pred, And):
            left = self.translate_filter(pred.left, sketch_data)
            right = self.translate_filter(pred.right, sketch_data)
            if left is None:
                return right
            if right is None:
                return left
            return left & right
        if isinstance(pred, Or):
            left = self.translate_filter(pred.left, sketch_data)
            right = self.translate_filter(pred.right, sketch_data)
            if left is None or right is None:
                return None
            return left | right
        if isinstance(pred, Not):
            return None
        refs = pred.references()
        for sketch in self.sketches:
            if {r.lower() for r in refs} == \
                    {sketch.base_column.lower()}:
                dtype_name = self.schema.field_type(
                    sketch.base_column) or "long"
                mask = sketch.convert_predicate(pred, sketch_data,
                                                dtype_name)
                if mask is not None:
                    return mask
        return None

    def prune_files(self, entry, pred: Expr, source_files: List[str],
                    device=None) -> Tuple[List[str], int]:
        """Returns (files to scan, number skipped).  Files absent from the
        sketch table (e.g. appended after build) are kept.  With a CUDA
        ``device`` the sketch predicate evaluates on device (K9) and only
        the per-file boolean mask crosses back to the host."""
        data = self.load_sketch_data(entry, device)
        mask = self.translate_filter(pred, data)
        if mask is None:
            return source_files, 0
        mask = mask.cpu()
        file_ids = data.tensor("_data_file_id").cpu()
        id_to_keep = {int(fid): bool(m)
                      for fid, m in zip(file_ids, mask)}
        # map file path -> logged id
        path_to_id = {f.name: f.id for f in entry.source_file_infos()}
        kept, skipped = [], 0
        for p in source_files:
            fid = path_to_id.get(p)
            if fid is None or id_to_keep.get(fid, True):
                kept.append(p)
            else:
                skipped += 1
        return kept, skipped

    def statistics(self) -> Dict[str, Any]:
        return {"sketches": [f"{s.kind}({s.expr})" for s in self.sketches]}


class _SketchData:
    def __init__(self, cols: Dict[str, torch.Tensor]):
        self.cols = cols

    def tensor(self, name: str) -> torch.Tensor:
        for k, v in self.cols.items():
            if k.lower() == name.lower():
                return v
        raise HyperspaceException(f"No sketch column {name}")


register_derived_dataset(DATASKIPPING_INDEX_TYPE, DataSkippingIndex)
