"""Expression tree for filters and join conditions.

The engine's analog of Catalyst expressions — only what the rule layer
needs: column refs, literals, comparisons, boolean connectives, IN.
"""

from __future__ import annotations

import re
from typing import Any, List, Set, Union

from ..exceptions import HyperspaceException


class Expr:
    def references(self) -> Set[str]:
        raise NotImplementedError

    # sugar so users can write col("a") == 1
    def __and__(self, other):
        return And(self, _wrap(other))

    def __or__(self, other):
        return Or(self, _wrap(other))

    def __invert__(self):
        return Not(self)

    # arithmetic sugar: (col("a") % 10) == 3 — scalar expressions the
    # data-skipping sketches can bind to (reference ExpressionUtils
    # accepts arbitrary deterministic scalar expressions)
    def __add__(self, other):
        return Arith("+", self, _wrap(other))

    def __sub__(self, other):
        return Arith("-", self, _wrap(other))

    def __mul__(self, other):
        return Arith("*", self, _wrap(other))

    def __mod__(self, other):
        return Arith("%", self, _wrap(other))

    def __truediv__(self, other):
        return Arith("/", self, _wrap(other))

    def isin(self, values):
        return In(self, list(values))


class Col(Expr):
    def __init__(self, name: str):
        self.name = name

    def references(self):
        return {self.name}

    def __eq__(self, other):  # type: ignore[override]
        return BinComp("=", self, _wrap(other))

    def __ne__(self, other):  # type: ignore[override]
        return BinComp("!=", self, _wrap(other))

    def __lt__(self, other):
        return BinComp("<", self, _wrap(other))

    def __le__(self, other):
        return BinComp("<=", self, _wrap(other))

    def __gt__(self, other):
        return BinComp(">", self, _wrap(other))

    def __ge__(self, other):
        return BinComp(">=", self, _wrap(other))

    def is_not_null(self):
        return IsNotNull(self)

    def is_null(self):
        return IsNull(self)

    def __hash__(self):
        return hash(("Col", self.name.lower()))

    def __repr__(self):
        return self.name


class Lit(Expr):
    def __init__(self, value: Any):
        self.value = value

    def references(self):
        return set()

    def __repr__(self):
        return repr(self.value)


class BinComp(Expr):
    """left <op> right with op in =, !=, <, <=, >, >=."""

    OPS = ("=", "!=", "<", "<=", ">", ">=")

    def __init__(self, op: str, left: Expr, right: Expr):
        if op not in self.OPS:
            raise HyperspaceException(f"Bad comparison op {op}")
        self.op = op
        self.left = left
        self.right = right

    def references(self):
        return self.left.references() | self.right.references()

    def __repr__(self):
        return f"({self.left!r} {self.op} {self.right!r})"


class Arith(Expr):
    """Scalar arithmetic over columns/literals: +, -, *, %, /.

    ``%`` follows Java/Spark remainder semantics (sign of the dividend).
    Comparisons on an Arith left-hand side evaluate it per row; the
    data-skipping sketches match predicates whose LHS is structurally
    equal to the sketch's expression."""

    OPS = ("+", "-", "*", "%", "/")

    def __init__(self, op: str, left: Expr, right: Expr):
        if op not in self.OPS:
            raise HyperspaceException(f"Bad arithmetic op {op}")
        self.op = op
        self.left = left
        self.right = right

    def references(self):
        return self.left.references() | self.right.references()

    def __eq__(self, other):  # type: ignore[override]
        if isinstance(other, Arith):
            # structural equality when both sides are expression trees
            # (sketch matching); comparison against anything else builds
            # a predicate like Col.__eq__ does
            return (self.op == other.op and
                    _expr_eq(self.left, other.left) and
                    _expr_eq(self.right, other.right))
        return BinComp("=", self, _wrap(other))

    def __ne__(self, other):  # type: ignore[override]
        return BinComp("!=", self, _wrap(other))

    def __lt__(self, other):
        return BinComp("<", self, _wrap(other))

    def __le__(self, other):
        return BinComp("<=", self, _wrap(other))

    def __gt__(self, other):
        return BinComp(">", self, _wrap(other))

    def __ge__(self, other):
        return BinComp(">=", self, _wrap(other))

    def __hash__(self):
        return hash(("Arith", self.op, repr(self)))

    def __repr__(self):
        return f"({self.left!r} {self.op} {self.right!r})"


def _expr_eq(a: "Expr", b: "Expr") -> bool:
    """Structural expression equality (case-insensitive column names)."""
    if isinstance(a, Col) and isinstance(b, Col):
        return a.name.lower() == b.name.lower()
    if isinstance(a, Lit) and isinstance(b, Lit):
        return a.value == b.value
    if isinstance(a, Arith) and isinstance(b, Arith):
        return (a.op == b.op and _expr_eq(a.left, b.left) and
                _expr_eq(a.right, b.right))
    return False


class And(Expr):
    def __init__(self, left: Expr, right: Expr):
        self.left, self.right = left, right

    def references(self):
        return self.left.references() | self.right.references()

    def __repr__(self):
        return f"({self.left!r} AND {self.right!r})"


class Or(Expr):
    def __init__(self, left: Expr, right: Expr):
        self.left, self.right = left, right

    def references(self):
        return self.left.references() | self.right.references()

    def __repr__(self):
        return f"({self.left!r} OR {self.right!r})"


class Not(Expr):
    def __init__(self, child: Expr):
        self.child = child

    def references(self):
        return self.child.references()

    def __repr__(self):
        return f"(NOT {self.child!r})"


class In(Expr):
    def __init__(self, col: Expr, values: List[Any]):
        self.col = col
        self.values = values

    def references(self):
        return self.col.references()

    def __repr__(self):
        return f"({self.col!r} IN {self.values!r})"


class IsNotNull(Expr):
    def __init__(self, col: Expr):
        self.col = col

    def references(self):
        return self.col.references()

    def __repr__(self):
        return f"({self.col!r} IS NOT NULL)"


class IsNull(Expr):
    def __init__(self, col: Expr):
        self.col = col

    def references(self):
        return self.col.references()

    def __repr__(self):
        return f"({self.col!r} IS NULL)"


def col(name: str) -> Col:
    return Col(name)


def lit(value: Any) -> Lit:
    return Lit(value)


def _wrap(v: Union[Expr, Any]) -> Expr:
    return v if isinstance(v, Expr) else Lit(v)


# ---------------------------------------------------------------------------
# Helpers used by the rules
# ---------------------------------------------------------------------------

def split_conjunctive(e: Expr) -> List[Expr]:
    """Flatten nested ANDs into a predicate list (CNF top level)."""
    if isinstance(e, And):
        return split_conjunctive(e.left) + split_conjunctive(e.right)
    return [e]


def extract_equi_join_keys(e: Expr) -> List[tuple]:
    """For a join condition that is a CNF of Col = Col, return
    [(left_name, right_name)] or [] if not an equi-join
    (reference: JoinIndexRule CNF check, index/covering/JoinIndexRule.scala).
    """
    pairs = []
    for p in split_conjunctive(e):
        if isinstance(p, BinComp) and p.op == "=" and \
                isinstance(p.left, Col) and isinstance(p.right, Col):
            pairs.append((p.left.name, p.right.name))
        else:
            return []
    return pairs


_PRED_RE = re.compile(
    r"^\s*([A-Za-z_][A-Za-z0-9_.]*)\s*(=|==|!=|<=|>=|<|>)\s*(.+?)\s*$")
_IN_RE = re.compile(
    r"^\s*([A-Za-z_][A-Za-z0-9_.]*)\s+(?:in|IN)\s*\(([^)]*)\)\s*$")
_NULL_RE = re.compile(
    r"^\s*([A-Za-z_][A-Za-z0-9_.]*)\s+(?:is|IS)\s+"
    r"(?:(not|NOT)\s+)?(?:null|NULL)\s*$")


def _parse_literal(val: str) -> Any:
    if val.startswith("'") or val.startswith('"'):
        return val.strip("'\"")
    try:
        return int(val)
    except ValueError:
        return float(val)


def parse_predicate(s: str) -> Expr:
    """Tiny predicate parser: 'col <op> literal', 'col IN (v, ...)',
    'col IS [NOT] NULL', joined by AND."""
    parts = re.split(r"\s+(?:AND|and)\s+(?![^(]*\))", s)
    exprs: List[Expr] = []
    for part in parts:
        m = _IN_RE.match(part)
        if m:
            name, vals = m.groups()
            exprs.append(In(Col(name),
                            [_parse_literal(v.strip())
                             for v in vals.split(",") if v.strip()]))
            continue
        m = _NULL_RE.match(part)
        if m:
            name, neg = m.groups()
            exprs.append(IsNotNull(Col(name)) if neg
                         else IsNull(Col(name)))
            continue
        m = _PRED_RE.match(part)
        if not m:
            raise HyperspaceException(f"Cannot parse predicate: {part}")
        name, op, val = m.groups()
        if op == "==":
            op = "="
        exprs.append(BinComp(op, Col(name), Lit(_parse_literal(val))))
    out = exprs[0]
    for e in exprs[1:]:
        out = And(out, e)
    return out
