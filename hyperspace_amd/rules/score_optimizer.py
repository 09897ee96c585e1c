"""Score-based index plan optimizer.

Reference: index/rules/ScoreBasedIndexPlanOptimizer.scala:31-81 —
top-down recursion with memoized (plan -> (bestPlan, score)); tries each
rule at the node, recurses into the transformed plan's children, keeps
the max total score.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

from .candidate_collector import Candidate
from .filter_reason import ReasonCollector
from .hyperspace_rules import (FilterIndexRule, HyperspaceRule,
                               JoinIndexRule, NoOpRule)
from ..plan.nodes import LogicalPlan


class ScoreBasedIndexPlanOptimizer:
    def __init__(self, session, reasons: ReasonCollector):
        self.session = session
        self.reasons = reasons
        self.rules: List[HyperspaceRule] = [
            FilterIndexRule(session, reasons),
            JoinIndexRule(session, reasons),
        ]
        # secondary index kinds participate when present
        try:
            from ..index.zorder.rule import ZOrderFilterIndexRule
            self.rules.append(ZOrderFilterIndexRule(session, reasons))
        except ImportError:
            pass
        try:
            from ..index.dataskipping.rule import ApplyDataSkippingIndex
            self.rules.append(ApplyDataSkippingIndex(session, reasons))
        except ImportError:
            pass
        self.rules.append(NoOpRule(session, reasons))

    def apply(self, plan: LogicalPlan,
              candidates: Dict[int, List[Candidate]]) -> LogicalPlan:
        best, _ = self._rec_apply(plan, candidates, {})
        return best

    def _rec_apply(self, plan: LogicalPlan, candidates,
                   memo: Dict[int, Tuple[LogicalPlan, float]]
                   ) -> Tuple[LogicalPlan, float]:
        key = id(plan)
        if key in memo:
            return memo[key]
        best_plan, best_score = plan, -1.0
        for rule in self.rules:
            transformed, score = rule.apply(plan, candidates)
            if transformed is plan and rule.name != "NoOpRule" and \
                    score == 0.0:
                # rule didn't fire; NoOpRule covers the identity case
                continue
            child_results = [self._rec_apply(c, candidates, memo)
                             for c in transformed.children]
            total = score + sum(s for _, s in child_results)
            if transformed is not plan or rule.name == "NoOpRule":
                candidate_plan = (
                    transformed.with_children([p for p, _ in child_results])
                    if child_results else transformed)
                if total > best_score:
                    best_plan, best_score = candidate_plan, total
        memo[key] = (best_plan, max(best_score, 0.0))
        return memo[key]
