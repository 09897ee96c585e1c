"""DataSkippingIndexConfig.

Reference: index/dataskipping/DataSkippingIndexConfig.scala —
(name, sketches…); rejects duplicate sketches (:86-94); auto-adds a
PartitionSketch when autoPartitionSketch (not applicable here: the
default parquet source has no hive partitioning in v0).
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from ..base import IndexConfigTrait, IndexerContext
from .index import DataSkippingIndex
from .sketches import Sketch
from ...exceptions import HyperspaceException
from ...log.entry import Schema
from ...utils.resolver import resolve_all
from ...config import IndexConstants


class DataSkippingIndexConfig(IndexConfigTrait):
    def __init__(self, index_name: str, *sketches: Sketch):
        if not index_name:
            raise HyperspaceException("Index name cannot be empty")
        if not sketches:
            raise HyperspaceException("At least one sketch is required")
        if len(set(sketches)) != len(sketches):
            raise HyperspaceException("Duplicate sketches")
        self._name = index_name
        self.sketches = list(sketches)

    @property
    def index_name(self) -> str:
        return self._name

    def referenced_columns(self) -> List[str]:
        return sorted({s.expr for s in self.sketches})

    def create_index(self, ctx: IndexerContext, df,
                     properties: Dict[str, str]
                     ) -> Tuple[DataSkippingIndex, Dict[str, torch.Tensor]]:
        from ...execution.executor import Executor
        from ...plan.nodes import Scan

        leaves = df.plan.collect_leaves()
        if len(leaves) != 1 or not isinstance(leaves[0], Scan):
            raise HyperspaceException(
                "createIndex requires a single file-source relation plan")
        scan: Scan = leaves[0]
        schema = scan.relation.schema
        cols = resolve_all(schema.field_names(), self.referenced_columns())

        # pre-assign ids deterministically (shared with covering build)
        files = sorted(scan.relation.all_files(), key=lambda f: f.name)
        for f in files:
            ctx.file_id_tracker.add_file(f.name, f.size, f.modifiedTime)

        ex = Executor(ctx.session)
        # per-file segmentation: read each column + per-file row counts
        from ...sources.parquet_io import read_files_batch
        paths = [f.name for f in files]
        batch, row_counts = read_files_batch(paths, columns=cols)
        if ctx.session.device.type == "cuda":
            batch = batch.to(ctx.session.device)
        seg = torch.zeros(len(paths) + 1, dtype=torch.int64)
        seg[1:] = torch.cumsum(torch.tensor(row_counts,
                                            dtype=torch.int64), 0)

        file_ids = torch.tensor(
            [ctx.file_id_tracker.get_file_id(f.name, f.size,
                                             f.modifiedTime)
             for f in files], dtype=torch.int64)
        index_data: Dict[str, torch.Tensor] = {
            IndexConstants.DATA_FILE_NAME_ID_COLUMN: file_ids}
        for sketch in self.sketches:
            resolved = resolve_all(schema.field_names(), [sketch.expr])[0]
            sketch.expr = resolved
            values = batch.tensor(resolved)
            dtype_name = schema.field_type(resolved) or "long"
            index_data.update(sketch.aggregate(values, seg, dtype_name,
                                               batch.mask(resolved)))

        index_schema = Schema([f for f in schema.fields
                               if f.name.lower() in
                               {c.lower() for c in cols}])
        index = DataSkippingIndex(self.sketches, index_schema,
                                  dict(properties))
        return index, index_data

    def placeholder_index(self, relation, conf):
        cols = resolve_all(relation.schema.field_names(),
                           self.referenced_columns())
        return DataSkippingIndex(
            self.sketches, relation.schema.select(cols), {})
