"""CoveringIndexConfig: user-facing config + index-data creation.

Reference: index/covering/CoveringIndexConfig.scala:38-62 (+ Builder),
validation in CoveringIndexConfigTrait.scala:23-52, data creation in
CoveringIndex.createIndexData (index/covering/CoveringIndex.scala:140-192:
projection + lineage column via input_file_name() join against the
broadcast file-id map — here the lineage ids are attached at scan time,
K12 folded into the scan).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..base import IndexConfigTrait, IndexerContext
from .index import CoveringIndex
from ...config import IndexConstants
from ...exceptions import HyperspaceException
from ...log.entry import Schema, SchemaField
from ...utils.resolver import resolve_all


class CoveringIndexConfig(IndexConfigTrait):
    def __init__(self, index_name: str, indexed_columns: List[str],
                 included_columns: Optional[List[str]] = None):
        self._name = index_name
        self.indexed_columns = list(indexed_columns)
        self.included_columns = list(included_columns or [])
        self._validate()

    def _validate(self):
        if not self._name:
            raise HyperspaceException("Index name cannot be empty")
        if not self.indexed_columns:
            raise HyperspaceException("Indexed columns cannot be empty")
        low_idx = [c.lower() for c in self.indexed_columns]
        low_inc = [c.lower() for c in self.included_columns]
        if len(set(low_idx)) != len(low_idx) or \
                len(set(low_inc)) != len(low_inc):
            raise HyperspaceException("Duplicate columns in index config")
        if set(low_idx) & set(low_inc):
            raise HyperspaceException(
                "Indexed and included columns must be disjoint")

    @property
    def index_name(self) -> str:
        return self._name

    class Builder:
        """Builder pattern (reference CoveringIndexConfig.Builder,
        index/covering/CoveringIndexConfig.scala:60-151): each setter
        may be called once; create() validates like the constructor."""

        def __init__(self):
            self._name = ""
            self._indexed: List[str] = []
            self._included: List[str] = []

        def index_name(self, name: str) -> "CoveringIndexConfig.Builder":
            if self._name:
                raise HyperspaceException("Index name is already set.")
            if not name:
                raise HyperspaceException(
                    "Empty index name is not allowed.")
            self._name = name
            return self

        def index_by(self, column: str, *more: str
                     ) -> "CoveringIndexConfig.Builder":
            if self._indexed:
                raise HyperspaceException(
                    "Indexed columns are already set.")
            self._indexed = [column, *more]
            return self

        def include(self, column: str, *more: str
                    ) -> "CoveringIndexConfig.Builder":
            if self._included:
                raise HyperspaceException(
                    "Included columns are already set.")
            self._included = [column, *more]
            return self

        # camelCase aliases mirroring the reference API
        indexName = index_name
        indexBy = index_by

        def create(self) -> "CoveringIndexConfig":
            return CoveringIndexConfig(self._name, self._indexed,
                                       self._included)

    @staticmethod
    def builder() -> "CoveringIndexConfig.Builder":
        return CoveringIndexConfig.Builder()

    def referenced_columns(self) -> List[str]:
        return self.indexed_columns + self.included_columns

    def create_index(self, ctx: IndexerContext, df,
                     properties: Dict[str, str]
                     ) -> Tuple[CoveringIndex, object]:
        from ...execution.executor import Executor
        from ...plan.nodes import Scan

        leaves = df.plan.collect_leaves()
        if len(leaves) != 1 or not isinstance(leaves[0], Scan):
            raise HyperspaceException(
                "createIndex requires a single file-source relation plan")
        scan: Scan = leaves[0]
        source_schema = scan.relation.schema
        indexed = resolve_all(source_schema.field_names(),
                              self.indexed_columns)
        included = resolve_all(source_schema.field_names(),
                               self.included_columns)
        # hive partition columns not named in the config join the
        # included set, so queries projecting them stay covered
        # (reference CreateActionBase adds missing partition columns to
        # the covering slice)
        pschema_fn = getattr(scan.relation, "partition_schema", None)
        if pschema_fn is not None:
            named = {c.lower() for c in indexed + included}
            included = included + [
                f.name for f in pschema_fn().fields
                if f.name.lower() not in named]

        lineage = (properties.get(IndexConstants.LINEAGE_PROPERTY, "false")
                   .lower() == "true")

        # pre-assign file ids deterministically across ranks (sorted file
        # order) so distributed shards agree on the lineage id space
        all_files = sorted(scan.relation.all_files(), key=lambda f: f.name)
        for f in all_files:
            ctx.file_id_tracker.add_file(f.name, f.size, f.modifiedTime)

        from ...parallel import dist_context as dc
        files = all_files
        if dc.is_distributed() and dc.get_world_size() > 1:
            files = files[dc.get_rank()::dc.get_world_size()]

        # the build consumes a ScanStream: byte-bounded file groups with
        # prefetch so disk/PCIe overlap the bucketize+sort+write pipeline
        # (group size 0 disables streaming -> one materialized batch)
        from ...execution.scan_stream import ScanStream
        group_bytes = int(ctx.session.conf.get(
            "spark.hyperspace.index.build.groupBytes", 32 << 30))
        cols = indexed + included
        stream = ScanStream(
            files, cols, ctx.session.device,
            lineage_tracker=ctx.file_id_tracker if lineage else None,
            group_bytes=group_bytes if group_bytes > 0 else (1 << 62),
            reader=scan.relation.read_files)
        batch = stream

        index_schema = Schema(
            [f for f in source_schema.fields
             if f.name.lower() in {c.lower() for c in indexed + included}])
        if lineage:
            index_schema = Schema(
                index_schema.fields
                + [SchemaField(IndexConstants.DATA_FILE_NAME_ID_COLUMN,
                               "long", False)])

        num_buckets = ctx.session.conf.num_buckets
        index = CoveringIndex(indexed, included, index_schema, num_buckets,
                              dict(properties))
        return index, batch

    def placeholder_index(self, relation, conf):
        from ...utils.resolver import resolve_all
        schema = relation.schema
        indexed = resolve_all(schema.field_names(), self.indexed_columns)
        included = resolve_all(schema.field_names(), self.included_columns)
        return CoveringIndex(indexed, included,
                             schema.select(indexed + included),
                             conf.num_buckets, {})
