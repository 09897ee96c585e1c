"""Case-insensitive column resolution.

Reference: util/ResolverUtils.scala:44-104 (nested-column ``__hs_nested.``
prefixing is not supported in v0 — flat columns only).
"""

from __future__ import annotations

from typing import Iterable, List, Optional

from ..exceptions import HyperspaceException


def resolve(available: Iterable[str], requested: str) -> Optional[str]:
    """Return the canonical-cased name from ``available`` matching
    ``requested`` case-insensitively, or None."""
    for a in available:
        if a.lower() == requested.lower():
            return a
    return None


def resolve_all(available: Iterable[str],
                requested: Iterable[str]) -> List[str]:
    """Resolve every requested column; raise if any is missing."""
    avail = list(available)
    out = []
    missing = []
    for r in requested:
        m = resolve(avail, r)
        if m is None:
            missing.append(r)
        else:
            out.append(m)
    if missing:
        raise HyperspaceException(
            f"Columns not found: {missing} (available: {avail})")
    return out
