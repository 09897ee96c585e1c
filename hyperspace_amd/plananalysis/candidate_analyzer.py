"""whyNot(): explain why indexes were not applied to a query.

Reference: index/plananalysis/CandidateIndexAnalyzer.scala:29-346 —
re-runs the rule pipeline with reason tagging enabled; renders a reason
table per (index, subplan).
"""

from __future__ import annotations

from typing import List

from ..log.constants import States
from ..rules.apply_hyperspace import ApplyHyperspace
from ..rules.filter_reason import ReasonCollector


class CandidateIndexAnalyzer:
    def __init__(self, session):
        self.session = session

    def why_not_string(self, df, index_name: str = "",
                       extended: bool = False) -> str:
        reasons = ReasonCollector(enabled=True)
        was_enabled = self.session._hyperspace_enabled
        try:
            self.session._hyperspace_enabled = True
            ApplyHyperspace(self.session, reasons).apply(df.plan)
        finally:
            self.session._hyperspace_enabled = was_enabled

        entries = self.session.index_manager().get_indexes([States.ACTIVE])
        lines: List[str] = []
        bar = "=" * 64
        lines.append(bar)
        lines.append("whyNot report:")
        lines.append(bar)
        names = ([index_name] if index_name
                 else sorted(e.name for e in entries))
        for name in names:
            applied = reasons.applied.get(name)
            lines.append(f"Index: {name}")
            if applied:
                lines.append(f"  APPLIED via {', '.join(applied)}")
                continue
            rs = reasons.all_for_index(name) + reasons.all_for_index("")
            if not rs:
                lines.append("  Not applicable to this plan "
                             "(no matching relation or another index won)")
            for r in rs:
                detail = f" {r.args}" if (extended and r.args) else ""
                lines.append(f"  {r.code}{detail}")

        # per-(index, subplan) matrix (reference
        # CandidateIndexAnalyzer.scala:29-346 renders the reason table
        # against each sub-plan the rules considered)
        matrix = reasons.by_subplan()
        if matrix:
            lines.append(bar)
            lines.append("Reasons per sub-plan:")
            lines.append(bar)
            for sp_i, (nid, per_index) in enumerate(matrix.items(), 1):
                node = reasons.nodes.get(nid)
                lines.append(
                    f"SubPlan #{sp_i}: {_one_line(node)}")
                width = max([len(n) for n in per_index if n] + [5])
                lines.append(f"  {'index'.ljust(width)} | reason")
                lines.append(f"  {'-' * width}-+-{'-' * 30}")
                for iname in sorted(per_index):
                    if index_name and iname and iname != index_name:
                        continue
                    for r in per_index[iname]:
                        detail = f" {r.args}" if (extended and r.args) \
                            else ""
                        lines.append(
                            f"  {(iname or '*').ljust(width)} | "
                            f"{r.code}{detail}")
        return "\n".join(lines)


def _one_line(node) -> str:
    """Compress a plan subtree into one line for the matrix header."""
    if node is None:
        return "<plan>"
    try:
        flat = " / ".join(part.strip()
                          for part in node.pretty().splitlines())
    except Exception:  # noqa: BLE001
        flat = repr(node)
    return flat[:110] + ("..." if len(flat) > 110 else "")
