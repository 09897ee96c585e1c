"""Min/max skipping-effectiveness analysis.

Reference: util/MinMaxAnalysisUtil.scala:624-780 — per-file min/max
histogram of value-range -> #files-to-read lookup cost; used to judge
whether a z-order or data-skipping index would help a column.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np


def analyze(df, columns: List[str], buckets: int = 20) -> str:
    """Text report: for each column, how many files a point lookup must
    read across the value range (lower = better clustered)."""
    from ..plan.nodes import Scan
    leaves = df.plan.collect_leaves()
    if len(leaves) != 1 or not isinstance(leaves[0], Scan):
        return "minmax analysis requires a single-relation plan"
    relation = leaves[0].relation
    import pyarrow.parquet as pq

    lines: List[str] = []
    for col in columns:
        per_file = []
        for f in relation.all_files():
            try:
                md = pq.ParquetFile(f.name).metadata
                idx = md.schema.to_arrow_schema().get_field_index(col)
                lo = hi = None
                for rg in range(md.num_row_groups):
                    st = md.row_group(rg).column(idx).statistics
                    if st is None or not st.has_min_max:
                        lo = None
                        break
                    lo = st.min if lo is None else min(lo, st.min)
                    hi = st.max if hi is None else max(hi, st.max)
                if lo is not None:
                    per_file.append((lo, hi))
            except Exception:  # noqa: BLE001
                continue
        lines.append(f"column: {col}")
        if not per_file:
            lines.append("  (no statistics available)")
            continue
        gmin = min(lo for lo, _ in per_file)
        gmax = max(hi for _, hi in per_file)
        span = (gmax - gmin) or 1
        n_files = len(per_file)
        lines.append(f"  files: {n_files}  range: [{gmin}, {gmax}]")
        lines.append(f"  {'value range':<28}{'#files to read':>15}")
        total = 0
        for b in range(buckets):
            b_lo = gmin + span * b / buckets
            b_hi = gmin + span * (b + 1) / buckets
            hits = sum(1 for lo, hi in per_file
                       if not (hi < b_lo or lo > b_hi))
            total += hits
            bar = "#" * max(1, int(40 * hits / n_files))
            lines.append(f"  [{b_lo:>10.4g}, {b_hi:>10.4g}]"
                         f"{hits:>10}  {bar}")
        avg = total / buckets
        lines.append(f"  avg files per point lookup: {avg:.1f} / {n_files} "
                     f"({100 * avg / n_files:.0f}% — lower is better "
                     "clustering; consider a z-order or min/max "
                     "data-skipping index if high)")
    return "\n".join(lines)


def _collect_per_file(relation, col):
    import pyarrow.parquet as pq
    per_file = []
    for f in relation.all_files():
        try:
            md = pq.ParquetFile(f.name).metadata
            idx = md.schema.to_arrow_schema().get_field_index(col)
            lo = hi = None
            for rg in range(md.num_row_groups):
                st = md.row_group(rg).column(idx).statistics
                if st is None or not st.has_min_max:
                    lo = None
                    break
                lo = st.min if lo is None else min(lo, st.min)
                hi = st.max if hi is None else max(hi, st.max)
            if lo is not None:
                per_file.append((f.name, lo, hi))
        except Exception:  # noqa: BLE001
            continue
    return per_file


def analyze_html(df, columns: List[str], buckets: int = 20) -> str:
    """Self-contained HTML report with inline SVG histograms (reference
    MinMaxAnalysisUtil's d3 output, util/MinMaxAnalysisUtil.scala —
    rendered without external scripts so it works offline)."""
    from ..plan.nodes import Scan
    leaves = df.plan.collect_leaves()
    if len(leaves) != 1 or not isinstance(leaves[0], Scan):
        return "<html><body>minmax analysis requires a " \
               "single-relation plan</body></html>"
    relation = leaves[0].relation
    parts: List[str] = [
        "<!DOCTYPE html><html><head><meta charset='utf-8'>",
        "<title>Hyperspace min/max analysis</title>",
        "<style>body{font-family:monospace}"
        ".bar{fill:#4878a8}.lbl{font-size:10px}</style></head><body>",
        "<h2>Min/max skipping-effectiveness analysis</h2>",
    ]
    for col in columns:
        per_file = _collect_per_file(relation, col)
        parts.append(f"<h3>column: {col}</h3>")
        if not per_file:
            parts.append("<p>(no statistics available)</p>")
            continue
        gmin = min(lo for _, lo, _ in per_file)
        gmax = max(hi for _, _, hi in per_file)
        span = (gmax - gmin) or 1
        n_files = len(per_file)
        hits_per_bucket = []
        for b in range(buckets):
            b_lo = gmin + span * b / buckets
            b_hi = gmin + span * (b + 1) / buckets
            hits_per_bucket.append(sum(
                1 for _, lo, hi in per_file
                if not (hi < b_lo or lo > b_hi)))
        avg = sum(hits_per_bucket) / buckets
        parts.append(
            f"<p>files: {n_files} &nbsp; range: [{gmin}, {gmax}] "
            f"&nbsp; avg files per point lookup: {avg:.1f} "
            f"({100 * avg / n_files:.0f}%)</p>")
        w, h, bw = 600, 160, 600 // buckets
        parts.append(f"<svg width='{w}' height='{h + 30}' "
                     "xmlns='http://www.w3.org/2000/svg'>")
        for b, hits in enumerate(hits_per_bucket):
            bh = int(h * hits / n_files)
            parts.append(
                f"<rect class='bar' x='{b * bw}' y='{h - bh}' "
                f"width='{bw - 2}' height='{bh}'>"
                f"<title>bucket {b}: {hits}/{n_files} files</title>"
                "</rect>")
            parts.append(
                f"<text class='lbl' x='{b * bw}' y='{h + 12}'>"
                f"{hits}</text>")
        parts.append("</svg>")
    parts.append("</body></html>")
    return "".join(parts)
