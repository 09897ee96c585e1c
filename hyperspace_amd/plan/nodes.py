"""Logical plan nodes.

The engine's analog of Spark's LogicalPlan for the subset Hyperspace
rewrites: file-source scans, filter, project, equi-join — plus the two
rewrite-target nodes: IndexScan (reference IndexHadoopFsRelation,
index/plans/logical/IndexHadoopFsRelation.scala:29) and BucketUnionNode
(reference BucketUnion, index/plans/logical/BucketUnion.scala:31).

Plans are immutable; rules build rewritten copies.  Each node carries a
``tags`` dict keyed by (index-name, tag-name) used by the whyNot analysis
(reference: IndexLogEntry rule-time tag map).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

from .expr import Expr
from ..exceptions import HyperspaceException


class LogicalPlan:
    def __init__(self, children: List["LogicalPlan"]):
        self.children = children

    def output_columns(self) -> List[str]:
        raise NotImplementedError

    def transform_up(self, fn) -> "LogicalPlan":
        new_children = [c.transform_up(fn) for c in self.children]
        node = self.with_children(new_children)
        return fn(node)

    def with_children(self, children: List["LogicalPlan"]) -> "LogicalPlan":
        raise NotImplementedError

    def collect_leaves(self) -> List["LogicalPlan"]:
        if not self.children:
            return [self]
        out: List[LogicalPlan] = []
        for c in self.children:
            out.extend(c.collect_leaves())
        return out

    def pretty(self, indent: int = 0) -> str:
        s = "  " * indent + self._node_str()
        for c in self.children:
            s += "\n" + c.pretty(indent + 1)
        return s

    def _node_str(self) -> str:
        return type(self).__name__

    def __repr__(self):
        return self.pretty()


class Scan(LogicalPlan):
    """Leaf: scan of a file-based source relation.

    ``file_subset`` restricts the scan to specific files — the rewrite
    target of ApplyDataSkippingIndex (the reference swaps the FileIndex
    for a DataSkippingFileIndex; here the pruned list is explicit).
    """

    def __init__(self, relation, options: Optional[Dict[str, Any]] = None,
                 file_subset: Optional[List[str]] = None,
                 skipped_files: int = 0):
        super().__init__([])
        self.relation = relation  # sources.FileBasedRelation
        self.options = options or {}
        self.file_subset = file_subset
        self.skipped_files = skipped_files

    def output_columns(self):
        return self.relation.schema.field_names()

    def with_children(self, children):
        assert not children
        return self

    def _node_str(self):
        if self.file_subset is None:
            extra = ""
        elif self.skipped_files:
            extra = f", dataskipping:-{self.skipped_files}files"
        else:
            extra = f", files:{len(self.file_subset)}"
        return f"Scan({self.relation.describe()}{extra})"


class Filter(LogicalPlan):
    def __init__(self, condition: Expr, child: LogicalPlan):
        super().__init__([child])
        self.condition = condition

    @property
    def child(self):
        return self.children[0]

    def output_columns(self):
        return self.child.output_columns()

    def with_children(self, children):
        return Filter(self.condition, children[0])

    def _node_str(self):
        return f"Filter({self.condition!r})"


class Project(LogicalPlan):
    def __init__(self, columns: List[str], child: LogicalPlan):
        super().__init__([child])
        self.columns = columns

    @property
    def child(self):
        return self.children[0]

    def output_columns(self):
        return list(self.columns)

    def with_children(self, children):
        return Project(self.columns, children[0])

    def _node_str(self):
        return f"Project({self.columns})"


class Join(LogicalPlan):
    def __init__(self, left: LogicalPlan, right: LogicalPlan,
                 condition: Expr, join_type: str = "inner"):
        super().__init__([left, right])
        if join_type != "inner":
            raise HyperspaceException("Only inner joins supported in v0")
        self.condition = condition
        self.join_type = join_type

    @property
    def left(self):
        return self.children[0]

    @property
    def right(self):
        return self.children[1]

    def output_columns(self):
        return self.left.output_columns() + self.right.output_columns()

    def with_children(self, children):
        return Join(children[0], children[1], self.condition, self.join_type)

    def _node_str(self):
        return f"Join({self.condition!r})"


class IndexScan(LogicalPlan):
    """Leaf: scan of covering-index data instead of source data.

    ``use_bucket_spec`` — expose the index's hash-bucket layout to the
    executor (bucket pruning for filters; co-partitioned zero-shuffle
    merge join for joins).
    ``excluded_source_file_ids`` — lineage ids whose rows must be dropped
    (Hybrid Scan deletes, K7).
    """

    def __init__(self, entry, columns: List[str], use_bucket_spec: bool,
                 excluded_source_file_ids: Optional[List[int]] = None,
                 version_files: Optional[List[str]] = None):
        super().__init__([])
        self.entry = entry  # IndexLogEntry
        self.columns = columns
        self.use_bucket_spec = use_bucket_spec
        self.excluded_source_file_ids = excluded_source_file_ids or []
        # explicit file list (subset of entry content) or None = all
        self.version_files = version_files

    def output_columns(self):
        return list(self.columns)

    def with_children(self, children):
        assert not children
        return self

    def _node_str(self):
        flags = []
        if self.use_bucket_spec:
            flags.append("bucketed")
        if self.excluded_source_file_ids:
            flags.append(f"-{len(self.excluded_source_file_ids)}files")
        return (f"IndexScan({self.entry.name}"
                + (", " + ",".join(flags) if flags else "") + ")")


class UnionNode(LogicalPlan):
    """Plain union (Hybrid Scan merge for the filter rule, where bucket
    alignment is unnecessary — reference CoveringIndexRuleUtils uses
    Union for FilterIndexRule, BucketUnion for JoinIndexRule)."""

    def __init__(self, children: List[LogicalPlan]):
        super().__init__(children)

    def output_columns(self):
        return self.children[0].output_columns()

    def with_children(self, children):
        return UnionNode(children)

    def _node_str(self):
        return "Union"


class BucketUnionNode(LogicalPlan):
    """Partition-aligned union of index data and on-the-fly-bucketed
    appended source data (Hybrid Scan merge, K5/K6)."""

    def __init__(self, children: List[LogicalPlan], num_buckets: int,
                 bucket_columns: List[str]):
        super().__init__(children)
        self.num_buckets = num_buckets
        self.bucket_columns = bucket_columns

    def output_columns(self):
        return self.children[0].output_columns()

    def with_children(self, children):
        return BucketUnionNode(children, self.num_buckets,
                               self.bucket_columns)

    def _node_str(self):
        return f"BucketUnion(n={self.num_buckets}, by={self.bucket_columns})"
