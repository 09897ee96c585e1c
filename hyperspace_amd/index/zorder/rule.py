"""ZOrderFilterIndexRule.

Reference: index/zordercovering/ZOrderFilterIndexRule.scala:36-153 —
like FilterIndexRule but ANY indexed column may appear in the filter;
rank = fewest indexed columns; score = 60 × coverage (between the
covering filter rule's 50 and the join rule's 70).
"""

from __future__ import annotations

from typing import Dict, List, Tuple

from .index import ZOrderCoveringIndex
from ...plan.nodes import Filter, LogicalPlan, Project
from ...rules.candidate_collector import Candidate
from ...rules.filter_reason import FilterReason, FilterReasons
from ...rules.hyperspace_rules import (HyperspaceRule, _coverage,
                                       _decompose_linear, _index_scan_for,
                                       _needed_columns)


class ZOrderFilterIndexRule(HyperspaceRule):
    name = "ZOrderFilterIndexRule"
    SCORE = 60.0

    def apply(self, plan: LogicalPlan,
              candidates: Dict[int, List[Candidate]]
              ) -> Tuple[LogicalPlan, float]:
        shape = _decompose_linear(plan)
        if shape is None:
            return plan, 0.0
        project, filt, scan = shape
        if filt is None:
            return plan, 0.0
        cands = candidates.get(id(scan), [])
        filter_refs = {r.lower() for r in filt.condition.references()}
        needed = _needed_columns(project, filt, scan)

        eligible = []
        for cand in cands:
            index = cand.index
            if not isinstance(index, ZOrderCoveringIndex):
                continue
            if not filter_refs & {c.lower() for c in index.indexed_columns}:
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.MISSING_INDEXED_COL,
                    {"indexedCols": str(index.indexed_columns)}))
                continue
            covered = {c.lower() for c in index.referenced_columns()}
            if index.can_handle_deleted_files:
                from ...config import IndexConstants
                covered.add(IndexConstants.DATA_FILE_NAME_ID_COLUMN.lower())
            if not {c.lower() for c in needed} <= covered:
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.MISSING_REQUIRED_COL,
                    {"needed": str(needed)}))
                continue
            eligible.append(cand)
        if not eligible:
            return plan, 0.0

        # rank: fewest indexed columns (tighter z-curve)
        best = min(eligible, key=lambda c: len(c.index.indexed_columns))
        index_scan = _index_scan_for(best, needed, use_bucket_spec=False)
        new_plan: LogicalPlan = Filter(filt.condition, index_scan)
        if project is not None:
            new_plan = Project(project.columns, new_plan)
        self.reasons.applied.setdefault(best.name, []).append(self.name)
        return new_plan, self.SCORE * _coverage(best)
