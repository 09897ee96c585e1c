"""ZOrderCoveringIndexConfig (reference:
index/zordercovering/ZOrderCoveringIndexConfig — same validation as the
covering config)."""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..base import IndexerContext
from ..covering.config import CoveringIndexConfig
from .index import ZOrderCoveringIndex
from ...config import IndexConstants
from ...log.entry import Schema, SchemaField
from ...utils.resolver import resolve_all


class ZOrderCoveringIndexConfig(CoveringIndexConfig):
    def create_index(self, ctx: IndexerContext, df,
                     properties: Dict[str, str]
                     ) -> Tuple[ZOrderCoveringIndex, object]:
        # reuse the covering projection/lineage pipeline, then wrap the
        # result in a z-order index; the global z-sort needs the whole
        # batch, so a streaming scan is materialized here
        cov_index, batch = super().create_index(ctx, df, properties)
        from ...execution.scan_stream import ScanStream
        if isinstance(batch, ScanStream):
            batch = batch.materialize()
        index = ZOrderCoveringIndex(
            cov_index.indexed_columns, cov_index.included_columns,
            cov_index.schema, dict(properties))
        return index, batch

    def placeholder_index(self, relation, conf):
        from ...utils.resolver import resolve_all
        schema = relation.schema
        indexed = resolve_all(schema.field_names(), self.indexed_columns)
        included = resolve_all(schema.field_names(), self.included_columns)
        return ZOrderCoveringIndex(indexed, included,
                                   schema.select(indexed + included), {})
