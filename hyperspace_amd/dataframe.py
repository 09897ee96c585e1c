"""User-facing DataFrame: a logical plan + session.

Mirrors the subset of the Spark DataFrame surface that Hyperspace's API
touches: filter / select / join / collect, plus plan access for
createIndex, explain and whyNot.
"""

from __future__ import annotations

from typing import List, Optional, Union

from .plan.expr import Expr, col as _col, parse_predicate
from .plan.nodes import Filter, Join, LogicalPlan, Project


class DataFrame:
    def __init__(self, session, plan: LogicalPlan):
        self.session = session
        self.plan = plan

    # -- transforms -------------------------------------------------------
    def filter(self, condition: Union[str, Expr]) -> "DataFrame":
        if isinstance(condition, str):
            condition = parse_predicate(condition)
        return DataFrame(self.session, Filter(condition, self.plan))

    where = filter

    def select(self, *columns: str) -> "DataFrame":
        cols = list(columns)
        return DataFrame(self.session, Project(cols, self.plan))

    def join(self, other: "DataFrame", on: Union[str, Expr, List[str]],
             how: str = "inner") -> "DataFrame":
        if isinstance(on, str):
            on = [on]
        if isinstance(on, list):
            expr: Optional[Expr] = None
            for name in on:
                e = _col(name) == _col(name)
                expr = e if expr is None else (expr & e)
            condition = expr
        else:
            condition = on
        assert condition is not None
        return DataFrame(self.session,
                         Join(self.plan, other.plan, condition, how))

    # -- actions ----------------------------------------------------------
    def optimized_plan(self) -> LogicalPlan:
        """Apply the Hyperspace rewrite rules if enabled."""
        if self.session.is_hyperspace_enabled():
            from .rules.apply_hyperspace import ApplyHyperspace
            return ApplyHyperspace(self.session).apply(self.plan)
        return self.plan

    def collect(self):
        from .execution.executor import Executor
        ex = Executor(self.session)
        batch = ex.execute(self.optimized_plan())
        self._last_stats = ex.stats
        return batch

    def count(self) -> int:
        return self.collect().num_rows

    def to_pandas(self):
        import pandas as pd
        return pd.DataFrame(self.collect().to_numpy())

    def explain_plan(self) -> str:
        return self.optimized_plan().pretty()

    def __repr__(self):
        return f"DataFrame:\n{self.plan.pretty()}"
