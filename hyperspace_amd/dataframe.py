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
        condition = _bind_decimal_literals(condition, self.plan)
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


def _bind_decimal_literals(cond: Expr, plan: LogicalPlan) -> Expr:
    """Scale numeric literals compared against decimal(p,s) columns to
    the column's unscaled-int64 representation (decimals ingest as
    unscaled integers with the scale recorded in the schema; Spark's
    analyzer performs the same literal cast)."""
    import decimal as _dec
    from .plan.expr import And, Arith, BinComp, Col, In, Lit, Not, Or
    scales = {}
    for leaf in plan.collect_leaves():
        rel = getattr(leaf, "relation", None)
        if rel is None:
            continue
        try:
            fields = rel.schema.fields
        except Exception:  # noqa: BLE001
            continue
        for f in fields:
            t = f.type or ""
            if t.startswith("decimal("):
                try:
                    scales[f.name.lower()] = int(
                        t[len("decimal("):-1].split(",")[1])
                except (IndexError, ValueError):
                    pass
    if not scales:
        return cond

    def col_scale(e):
        if isinstance(e, Col):
            return scales.get(e.name.lower())
        if isinstance(e, Arith):
            return col_scale(e.left) if col_scale(e.left) is not None \
                else col_scale(e.right)
        return None

    def scale_value(v, s):
        return int(_dec.Decimal(str(v)).scaleb(s))

    def walk(e):
        if isinstance(e, And):
            return And(walk(e.left), walk(e.right))
        if isinstance(e, Or):
            return Or(walk(e.left), walk(e.right))
        if isinstance(e, Not):
            return Not(walk(e.child))
        if isinstance(e, BinComp) and isinstance(e.right, Lit):
            s = col_scale(e.left)
            if s is not None and isinstance(
                    e.right.value, (int, float, _dec.Decimal)):
                return BinComp(e.op, e.left,
                               Lit(scale_value(e.right.value, s)))
        if isinstance(e, In):
            s = col_scale(e.col)
            if s is not None:
                return In(e.col, [
                    scale_value(v, s)
                    if isinstance(v, (int, float, _dec.Decimal)) else v
                    for v in e.values])
        return e

    return walk(cond)
