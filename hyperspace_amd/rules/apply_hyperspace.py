"""Session-injected plan rewrite entry point.

Reference: index/rules/ApplyHyperspace.scala:32-76 — fetches ACTIVE
indexes, collects candidates per relation, runs the score-based
optimizer; exception-safe (falls back to the original plan); disabled
during index maintenance via the session thread-local kill-switch.
"""

from __future__ import annotations

import logging

from .candidate_collector import CandidateIndexCollector
from .filter_reason import ReasonCollector
from .score_optimizer import ScoreBasedIndexPlanOptimizer
from ..log.constants import States
from ..plan.nodes import LogicalPlan

logger = logging.getLogger(__name__)


class ApplyHyperspace:
    def __init__(self, session, reasons: ReasonCollector = None):
        self.session = session
        self.reasons = reasons or ReasonCollector(
            enabled=session.conf.plan_analysis_enabled)

    def apply(self, plan: LogicalPlan) -> LogicalPlan:
        if not self.session.is_hyperspace_enabled():
            return plan
        try:
            entries = self.session.index_manager().get_indexes(
                [States.ACTIVE])
            if not entries:
                return plan
            collector = CandidateIndexCollector(self.session, self.reasons)
            candidates = collector.collect(plan, entries)
            if not candidates:
                return plan
            optimizer = ScoreBasedIndexPlanOptimizer(self.session,
                                                     self.reasons)
            return optimizer.apply(plan, candidates)
        except Exception:  # noqa: BLE001 - must never break a query
            logger.exception("ApplyHyperspace failed; using original plan")
            return plan
