"""whyNot reason taxonomy.

Reference: index/plananalysis/FilterReason.scala:19-158 — typed reason
codes recorded per (index, subplan) when plan analysis is enabled.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Tuple


class FilterReasons:
    COL_SCHEMA_MISMATCH = "COL_SCHEMA_MISMATCH"
    SOURCE_DATA_CHANGED = "SOURCE_DATA_CHANGED"
    NO_DELETE_SUPPORT = "NO_DELETE_SUPPORT"
    NO_COMMON_FILES = "NO_COMMON_FILES"
    TOO_MUCH_APPENDED = "TOO_MUCH_APPENDED"
    TOO_MUCH_DELETED = "TOO_MUCH_DELETED"
    NO_FIRST_INDEXED_COL_COND = "NO_FIRST_INDEXED_COL_COND"
    MISSING_REQUIRED_COL = "MISSING_REQUIRED_COL"
    NOT_ELIGIBLE_JOIN = "NOT_ELIGIBLE_JOIN"
    NO_AVAIL_JOIN_INDEX_PAIR = "NO_AVAIL_JOIN_INDEX_PAIR"
    NOT_ALL_JOIN_COL_INDEXED = "NOT_ALL_JOIN_COL_INDEXED"
    MISSING_INDEXED_COL = "MISSING_INDEXED_COL"
    ANOTHER_INDEX_APPLIED = "ANOTHER_INDEX_APPLIED"
    NO_FILTER_ON_INDEXED_COL = "NO_FILTER_ON_INDEXED_COL"


@dataclass
class FilterReason:
    code: str
    args: Dict[str, str]
    verbose: str = ""

    def __str__(self):
        extra = f" ({self.verbose})" if self.verbose else ""
        return f"{self.code}{extra}"


class ReasonCollector:
    """Collects (index name, plan node) -> reasons during rule analysis
    (the reference's FILTER_REASONS tag map, index/rules/IndexFilter.scala)."""

    def __init__(self, enabled: bool = False):
        self.enabled = enabled
        self.reasons: Dict[Tuple[str, int], List[FilterReason]] = {}
        self.applied: Dict[str, List[str]] = {}
        self.nodes: Dict[int, object] = {}  # id -> plan node (matrix)

    def add(self, index_name: str, plan_node, reason: FilterReason):
        if not self.enabled:
            return
        key = (index_name, id(plan_node))
        self.nodes[id(plan_node)] = plan_node
        self.reasons.setdefault(key, []).append(reason)

    def all_for_index(self, index_name: str) -> List[FilterReason]:
        out = []
        for (name, _), rs in self.reasons.items():
            if name == index_name:
                out.extend(rs)
        return out

    def by_subplan(self) -> "Dict[int, Dict[str, List[FilterReason]]]":
        """plan-node id -> index name -> reasons (the reference's
        per-(index, subplan) whyNot matrix)."""
        out: Dict[int, Dict[str, List[FilterReason]]] = {}
        for (name, nid), rs in self.reasons.items():
            out.setdefault(nid, {}).setdefault(name, []).extend(rs)
        return out
