"""explain(): plan-with vs plan-without Hyperspace.

Reference: index/plananalysis/PlanAnalyzer.scala:37-418 — plans the query
with and without the rules, walks both trees highlighting differing
subtrees, and prints indexes used + physical operator diff.
"""

from __future__ import annotations

from typing import Dict, List

from ..plan.nodes import IndexScan, LogicalPlan
from ..rules.apply_hyperspace import ApplyHyperspace
from ..rules.filter_reason import ReasonCollector


def _collect_index_scans(plan: LogicalPlan) -> List[IndexScan]:
    out = []

    def walk(n):
        if isinstance(n, IndexScan):
            out.append(n)
        for c in n.children:
            walk(c)

    walk(plan)
    return out


def _operator_counts(plan: LogicalPlan) -> Dict[str, int]:
    counts: Dict[str, int] = {}

    def walk(n):
        name = type(n).__name__
        counts[name] = counts.get(name, 0) + 1
        for c in n.children:
            walk(c)

    walk(plan)
    return counts


class DisplayMode:
    """Rendering modes for explain output (reference:
    index/plananalysis/ BufferStream/DisplayMode — console ANSI,
    plaintext markers, HTML tags; selected by conf
    spark.hyperspace.explain.displayMode)."""

    def __init__(self, highlight_open: str, highlight_close: str,
                 newline: str = "\n"):
        self.highlight_open = highlight_open
        self.highlight_close = highlight_close
        self.newline = newline

    @staticmethod
    def for_conf(conf) -> "DisplayMode":
        from ..config import IndexConstants
        mode = (conf.get(IndexConstants.DISPLAY_MODE) or "plaintext").lower()
        if mode == "console":
            return DisplayMode("\x1b[92m", "\x1b[0m")
        if mode == "html":
            return DisplayMode("<b>", "</b>", newline="<br>")
        return DisplayMode("<----", "---->")


class PlanAnalyzer:
    def __init__(self, session):
        self.session = session

    def explain_string(self, df, verbose: bool = False) -> str:
        original = df.plan
        reasons = ReasonCollector(enabled=True)
        was_enabled = self.session._hyperspace_enabled
        try:
            self.session._hyperspace_enabled = True
            rewritten = ApplyHyperspace(self.session, reasons).apply(original)
        finally:
            self.session._hyperspace_enabled = was_enabled

        mode = DisplayMode.for_conf(self.session.conf)
        orig_lines = set(original.pretty().splitlines())
        new_lines = set(rewritten.pretty().splitlines())

        def highlight(plan_str, other):
            out = []
            for ln in plan_str.splitlines():
                if ln not in other:
                    out.append(f"{mode.highlight_open}{ln.lstrip()}"
                               f"{mode.highlight_close}".rjust(
                                   len(ln) + len(mode.highlight_open)
                                   + len(mode.highlight_close)))
                else:
                    out.append(ln)
            return "\n".join(out)

        lines: List[str] = []
        bar = "=" * 64
        lines.append(bar)
        lines.append("Plan with indexes:")
        lines.append(bar)
        lines.append(highlight(rewritten.pretty(), orig_lines))
        lines.append("")
        lines.append(bar)
        lines.append("Plan without indexes:")
        lines.append(bar)
        lines.append(highlight(original.pretty(), new_lines))
        lines.append("")
        lines.append(bar)
        lines.append("Indexes used:")
        lines.append(bar)
        for scan in _collect_index_scans(rewritten):
            entry = scan.entry
            loc = ""
            infos = entry.content.os_file_infos()
            if infos:
                import os
                loc = os.path.dirname(infos[0].name)
            lines.append(f"{entry.name}:{loc}")
        lines.append("")
        out = "\n".join(lines)
        tail: List[str] = []
        lines = tail  # verbose section appended below
        if verbose:
            lines.append(bar)
            lines.append("Physical operator stats:")
            lines.append(bar)
            with_counts = _operator_counts(rewritten)
            without_counts = _operator_counts(original)
            all_ops = sorted(set(with_counts) | set(without_counts))
            lines.append(f"{'operator':<28}{'with':>6}{'without':>9}"
                         f"{'diff':>6}")
            for op in all_ops:
                w = with_counts.get(op, 0)
                wo = without_counts.get(op, 0)
                lines.append(f"{op:<28}{w:>6}{wo:>9}{w - wo:>6}")
        result = out + ("\n" + "\n".join(tail) if tail else "")
        if mode.newline != "\n":
            result = result.replace("\n", mode.newline)
        return result
