"""ApplyDataSkippingIndex rule.

Reference: index/dataskipping/rules/ApplyDataSkippingIndex.scala:33-105 —
pattern Filter-Relation; translates the filter into a sketch predicate,
swaps the FileIndex for a pruned one; score = 1 so covering indexes
always win; ranker prefers the index with the most sketches
(DataSkippingIndexRanker.scala:30-37).
"""

from __future__ import annotations

from typing import Dict, List, Tuple

from .index import DataSkippingIndex
from ...plan.nodes import Filter, LogicalPlan, Project, Scan
from ...rules.candidate_collector import Candidate
from ...rules.filter_reason import (FilterReason, FilterReasons,
                                    ReasonCollector)
from ...rules.hyperspace_rules import HyperspaceRule, _decompose_linear


class ApplyDataSkippingIndex(HyperspaceRule):
    name = "ApplyDataSkippingIndex"
    SCORE = 1.0

    def apply(self, plan: LogicalPlan,
              candidates: Dict[int, List[Candidate]]
              ) -> Tuple[LogicalPlan, float]:
        shape = _decompose_linear(plan)
        if shape is None:
            return plan, 0.0
        project, filt, scan = shape
        if filt is None or scan.file_subset is not None:
            return plan, 0.0
        cands = [c for c in candidates.get(id(scan), [])
                 if isinstance(c.index, DataSkippingIndex)
                 and not c.hybrid_required]
        if not cands:
            return plan, 0.0

        # ranker: most sketches first
        cands.sort(key=lambda c: -len(c.index.sketches))
        for cand in cands:
            kept, skipped = cand.index.prune_files(
                cand.entry, filt.condition,
                [f.name for f in scan.relation.all_files()],
                device=self.session.device)
            if skipped == 0:
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.NO_FILTER_ON_INDEXED_COL,
                    {"note": "predicate not convertible or nothing "
                             "skippable"}))
                continue
            new_scan = Scan(scan.relation, scan.options,
                            file_subset=kept, skipped_files=skipped)
            new_plan: LogicalPlan = Filter(filt.condition, new_scan)
            if project is not None:
                new_plan = Project(project.columns, new_plan)
            self.reasons.applied.setdefault(cand.name, []).append(self.name)
            return new_plan, self.SCORE
        return plan, 0.0
