"""Case-insensitive column resolution.

Reference: util/ResolverUtils.scala:44-104.  Nested struct leaves are
flattened to dotted names ("a.b.c") at the Arrow boundary; the
reference's ``__hs_nested.a.b.c`` spelling resolves as an alias of the
dotted form, so configs written for the reference keep working.
"""

from __future__ import annotations

from typing import Iterable, List, Optional

from ..exceptions import HyperspaceException

NESTED_PREFIX = "__hs_nested."


def _canon(name: str) -> str:
    n = name.lower()
    return n[len(NESTED_PREFIX):] if n.startswith(NESTED_PREFIX) else n


def resolve(available: Iterable[str], requested: str) -> Optional[str]:
    """Return the canonical-cased name from ``available`` matching
    ``requested`` case-insensitively (``__hs_nested.`` prefixes are
    ignored on both sides), or None."""
    req = _canon(requested)
    for a in available:
        if _canon(a) == req:
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
