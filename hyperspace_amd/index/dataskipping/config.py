"""DataSkippingIndexConfig.

Reference: index/dataskipping/DataSkippingIndexConfig.scala —
(name, sketches…); rejects duplicate sketches (:86-94); auto-adds a
PartitionSketch over each hive partition column when
autoPartitionSketch is on (:56-84, default true) so disjunctions like
``A = 1 OR part = 1`` stay convertible.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from ..base import IndexConfigTrait, IndexerContext
from .index import DataSkippingIndex
from .sketches import PartitionSketch, Sketch
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
        return sorted({s.base_column for s in self.sketches})

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

        # autoPartitionSketch: cover every hive partition column
        sketches = list(self.sketches)
        auto = str(ctx.session.conf.get(
            "spark.hyperspace.index.dataSkipping.autoPartitionSketch",
            True)).lower() != "false"
        pschema_fn = getattr(scan.relation, "partition_schema", None)
        if auto and pschema_fn is not None:
            covered = {s.expr.lower() for s in sketches
                       if isinstance(s, PartitionSketch)}
            for f in pschema_fn().fields:
                # numeric partition columns only: the first-value sketch
                # compares tensors at query time (string partitions still
                # prune via the executor's metadata partition pruning)
                if f.name.lower() not in covered and \
                        f.type in ("long", "double"):
                    sketches.append(PartitionSketch(f.name))
        cols = resolve_all(schema.field_names(),
                           sorted({s.base_column for s in sketches}))

        # pre-assign ids deterministically (shared with covering build)
        files = sorted(scan.relation.all_files(), key=lambda f: f.name)
        for f in files:
            ctx.file_id_tracker.add_file(f.name, f.size, f.modifiedTime)

        ex = Executor(ctx.session)
        # per-file segmentation: read each column + per-file row counts
        # through the relation seam (partition columns materialize there)
        paths = [f.name for f in files]
        batch, row_counts = scan.relation.read_files(
            paths, cols, ctx.session.device)
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
        for sketch in sketches:
            resolved = resolve_all(schema.field_names(),
                                   [sketch.base_column])[0]
            sketch.rebind(resolved)
            values = sketch.compute_values(batch.tensor(resolved))
            dtype_name = sketch.value_type(
                schema.field_type(resolved) or "long")
            index_data.update(sketch.aggregate(values, seg, dtype_name,
                                               batch.mask(resolved)))

        index_schema = Schema([f for f in schema.fields
                               if f.name.lower() in
                               {c.lower() for c in cols}])
        index = DataSkippingIndex(sketches, index_schema,
                                  dict(properties))
        return index, index_data

    def placeholder_index(self, relation, conf):
        cols = resolve_all(relation.schema.field_names(),
                           self.referenced_columns())
        return DataSkippingIndex(
            self.sketches, relation.schema.select(cols), {})
