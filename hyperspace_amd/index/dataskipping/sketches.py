"""Sketches: per-source-file summaries used to skip files at query time.

Reference: index/dataskipping/sketches/ — Sketch trait (Sketch.scala:36-119),
MinMaxSketch (MinMaxSketch.scala:45-100), BloomFilterSketch
(BloomFilterSketch.scala:47-87), PartitionSketch (PartitionSketch.scala:38-74).

Each sketch:
  - contributes aggregate columns to the per-file index data (build side,
    K8 segmented reductions / device bloom build), and
  - converts filter predicates on its column into a boolean "file may
    contain matches" expression over those aggregates (query side, K9).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import re

import torch

from ... import ops
from ...exceptions import HyperspaceException
from ...plan.expr import (Arith, BinComp, Col, Expr, In, IsNotNull, Lit,
                          _expr_eq)

# Sketch expressions are arbitrary composed deterministic scalar
# arithmetic over ONE column and numeric literals — e.g. "a % 10",
# "a * 2 + 1", "(a + 3) % 7" — parsed into the plan layer's Expr trees
# and matched structurally against predicate left-hand sides (the
# reference resolves arbitrary deterministic scalar expressions through
# Catalyst and rejects aggregates/windows/subqueries,
# dataskipping/expressions/ExpressionUtils.scala:38-95; this grammar has
# no aggregate forms, and multi-column / column-free expressions are
# rejected below).

_TOKEN_RE = re.compile(
    r"\s*(\d+\.\d+|\d+|[A-Za-z_][\w.]*|[()+\-*/%])")


def parse_expr_string(s: str) -> Expr:
    """Parse '+ - * / %' arithmetic with parentheses over column names
    and numeric literals into an Expr tree (standard precedence)."""
    tokens: List[str] = []
    pos = 0
    while pos < len(s):
        m = _TOKEN_RE.match(s, pos)
        if m is None:
            if s[pos:].strip():
                raise HyperspaceException(
                    f"Bad sketch expression near {s[pos:]!r}")
            break
        tokens.append(m.group(1))
        pos = m.end()
    it = {"i": 0}

    def peek():
        return tokens[it["i"]] if it["i"] < len(tokens) else None

    def take():
        t = peek()
        it["i"] += 1
        return t

    def atom() -> Expr:
        t = take()
        if t is None:
            raise HyperspaceException(f"Truncated sketch expression {s!r}")
        if t == "(":
            e = add()
            if take() != ")":
                raise HyperspaceException(f"Unbalanced parens in {s!r}")
            return e
        if t == "-":
            inner = atom()
            if isinstance(inner, Lit):
                return Lit(-inner.value)
            return Arith("-", Lit(0), inner)
        if re.fullmatch(r"\d+\.\d+", t):
            return Lit(float(t))
        if re.fullmatch(r"\d+", t):
            return Lit(int(t))
        if re.fullmatch(r"[A-Za-z_][\w.]*", t):
            return Col(t)
        raise HyperspaceException(f"Bad token {t!r} in {s!r}")

    def mul() -> Expr:
        e = atom()
        while peek() in ("*", "/", "%"):
            e = Arith(take(), e, atom())
        return e

    def add() -> Expr:
        e = mul()
        while peek() in ("+", "-"):
            e = Arith(take(), e, mul())
        return e

    out = add()
    if peek() is not None:
        raise HyperspaceException(f"Trailing tokens in {s!r}")
    return out


def expr_to_string(e: Expr) -> str:
    """Canonical form: flat for a single binary op ('a % 10' — the
    round-1 naming), parenthesized sub-expressions when nested."""
    def fmt(x: Expr, nested: bool) -> str:
        if isinstance(x, Col):
            return x.name
        if isinstance(x, Lit):
            return repr(x.value)
        if isinstance(x, Arith):
            inner = f"{fmt(x.left, True)} {x.op} {fmt(x.right, True)}"
            return f"({inner})" if nested else inner
        raise HyperspaceException(f"Not a sketch expression: {x!r}")
    return fmt(e, False)


def eval_expr_tree(e: Expr, columns: Dict[str, torch.Tensor]):
    """Evaluate an arithmetic Expr over tensors.  ``%`` uses Java/Spark
    remainder semantics (sign of the dividend: fmod); ``/`` is double
    division."""
    if isinstance(e, Col):
        for k, v in columns.items():
            if k.lower() == e.name.lower():
                return v
        raise HyperspaceException(f"No column {e.name}")
    if isinstance(e, Lit):
        return e.value
    if isinstance(e, Arith):
        lv = eval_expr_tree(e.left, columns)
        rv = eval_expr_tree(e.right, columns)
        if e.op == "+":
            return lv + rv
        if e.op == "-":
            return lv - rv
        if e.op == "*":
            return lv * rv
        if e.op == "%":
            return torch.fmod(lv, rv) if torch.is_tensor(lv) else lv % rv
        lt = lv.to(torch.float64) if torch.is_tensor(lv) else float(lv)
        return lt / rv
    raise HyperspaceException(f"Cannot evaluate {e!r}")


def _has_division(e: Expr) -> bool:
    return isinstance(e, Arith) and (
        e.op == "/" or _has_division(e.left) or _has_division(e.right))


def _rename_cols(e: Expr, new_name: str) -> Expr:
    if isinstance(e, Col):
        return Col(new_name)
    if isinstance(e, Arith):
        return Arith(e.op, _rename_cols(e.left, new_name),
                     _rename_cols(e.right, new_name))
    return e

MINMAX_SKETCH_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.sketches.MinMaxSketch")
BLOOM_SKETCH_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.sketches."
    "BloomFilterSketch")
PARTITION_SKETCH_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.sketches.PartitionSketch")


class Sketch:
    """Base sketch over a single source column expression: a column
    name or any composed arithmetic over that column and literals
    (reference SingleExprSketch + ExpressionUtils resolve/normalize)."""

    def __init__(self, expr: str):
        self.expr = expr.strip()
        self._tree: Optional[Expr] = None

    @property
    def tree(self) -> Expr:
        if self._tree is None:
            self._tree = parse_expr_string(self.expr)
        return self._tree

    @property
    def base_column(self) -> str:
        cols = sorted(self.tree.references())
        if len(cols) != 1:
            raise HyperspaceException(
                "A sketch expression must reference exactly one column; "
                f"{self.expr!r} references {cols or 'none'}")
        return cols[0]

    def rebind(self, resolved_col: str) -> None:
        """Rewrite the expression with the schema-resolved column."""
        self._tree = _rename_cols(self.tree, resolved_col)
        self.expr = expr_to_string(self._tree)

    def compute_values(self, col_values: torch.Tensor) -> torch.Tensor:
        out = eval_expr_tree(self.tree, {self.base_column: col_values})
        if not torch.is_tensor(out):
            raise HyperspaceException(
                f"Sketch expression {self.expr!r} is constant")
        return out

    def value_type(self, col_type: str) -> str:
        return "double" if _has_division(self.tree) else col_type

    def matches_lhs(self, lhs: Expr) -> bool:
        """True when a predicate's left-hand side is structurally this
        sketch's expression."""
        return _expr_eq(lhs, self.tree)

    @property
    def kind(self) -> str:
        raise NotImplementedError

    def out_columns(self) -> List[str]:
        """Names of the aggregate columns this sketch adds."""
        raise NotImplementedError

    def aggregate(self, values: torch.Tensor, seg_off: torch.Tensor,
                  dtype_name: str,
                  mask: Optional[torch.Tensor] = None
                  ) -> Dict[str, torch.Tensor]:
        """Per-file aggregates; ``values`` is the source column (device ok),
        ``seg_off`` the per-file row offsets, ``mask`` the optional
        validity mask (nulls are ignored, as in Spark's min/max/first
        aggregates)."""
        raise NotImplementedError

    def convert_predicate(self, pred: Expr, sketch_data,
                          dtype_name: str) -> Optional[torch.Tensor]:
        """Boolean tensor over files ("may contain") or None if the
        predicate is not convertible by this sketch."""
        raise NotImplementedError

    def to_json(self) -> Dict[str, Any]:
        raise NotImplementedError

    def __eq__(self, other):
        return type(self) is type(other) and \
            self.expr.lower() == other.expr.lower()

    def __hash__(self):
        return hash((type(self).__name__, self.expr.lower()))


def _norm_scalar(value, dtype_name: str) -> int:
    dt = {"long": torch.int64, "integer": torch.int32,
          "double": torch.float64, "float": torch.float32}.get(
              dtype_name, torch.int64)
    return int(ops.cpu_ref.normalize_key(
        torch.tensor([value], dtype=dt))[0])


class MinMaxSketch(Sketch):
    """Per-file min/max; converts =, <, <=, >, >=, In, IsNotNull
    (reference MinMaxSketch.scala:45-100)."""

    @property
    def kind(self):
        return "MinMax"

    def out_columns(self):
        return [f"MinMax_{self.expr}__min", f"MinMax_{self.expr}__max"]

    def aggregate(self, values, seg_off, dtype_name, mask=None):
        norm = ops.normalize_key(values)
        seg = seg_off.to(norm.device) if norm.is_cuda else seg_off
        if mask is not None and not bool(mask.all()):
            # normalized keys compare as signed int64, so the
            # null-neutral sentinels are the signed extremes
            m = mask.to(norm.device)
            mins, _ = ops.segmented_minmax(
                torch.where(m, norm,
                            torch.full_like(norm, 2**63 - 1)), seg)
            _, maxs = ops.segmented_minmax(
                torch.where(m, norm,
                            torch.full_like(norm, -2**63)), seg)
        else:
            mins, maxs = ops.segmented_minmax(norm, seg)
        return {self.out_columns()[0]: mins.cpu(),
                self.out_columns()[1]: maxs.cpu()}

    def convert_predicate(self, pred, sketch_data, dtype_name):
        mn = sketch_data.tensor(self.out_columns()[0])
        mx = sketch_data.tensor(self.out_columns()[1])
        dtype_name = self.value_type(dtype_name)
        if isinstance(pred, BinComp) and isinstance(pred.right, Lit) and \
                self.matches_lhs(pred.left):
            v = _norm_scalar(pred.right.value, dtype_name)
            if pred.op == "=":
                return (mn <= v) & (mx >= v)
            if pred.op == "<":
                return mn < v
            if pred.op == "<=":
                return mn <= v
            if pred.op == ">":
                return mx > v
            if pred.op == ">=":
                return mx >= v
            return None  # != not convertible
        if isinstance(pred, In) and self.matches_lhs(pred.col):
            out = torch.zeros(mn.numel(), dtype=torch.bool,
                              device=mn.device)
            for value in pred.values:
                v = _norm_scalar(value, dtype_name)
                out |= (mn <= v) & (mx >= v)
            return out
        if isinstance(pred, IsNotNull) and self.matches_lhs(pred.col):
            return torch.ones(mn.numel(), dtype=torch.bool,
                              device=mn.device)
        return None

    def to_json(self):
        return {"type": MINMAX_SKETCH_TYPE, "expr": self.expr,
                "dataType": None}

    @staticmethod
    def from_json(d):
        return MinMaxSketch(d["expr"])


class BloomFilterSketch(Sketch):
    """Per-file bloom filter; converts =, In
    (reference BloomFilterSketch.scala:47-87).  Parameters follow the
    reference: fpp + expected distinct count per file."""

    def __init__(self, expr: str, fpp: float = 0.01,
                 expected_distinct: int = 10000):
        super().__init__(expr)
        self.fpp = fpp
        self.expected_distinct = expected_distinct
        import math
        m = int(-expected_distinct * math.log(fpp) / (math.log(2) ** 2))
        self.m_bits = max(64, (m + 63) // 64 * 64)
        self.k = max(1, round(m / expected_distinct * math.log(2)))

    @property
    def kind(self):
        return "BloomFilter"

    def out_columns(self):
        return [f"BloomFilter_{self.expr}__bf"]

    def aggregate(self, values, seg_off, dtype_name, mask=None):
        vals = values
        if vals.dtype == torch.float64:
            vals = vals.view(torch.int64)
        elif vals.dtype == torch.float32:
            # widen to f64 bits so probe-side conversion matches exactly
            vals = vals.to(torch.float64).view(torch.int64)
        elif vals.dtype != torch.int64:
            vals = vals.to(torch.int64)
        if mask is not None:
            mask = mask.to(vals.device)
        words_per_file = []
        for s in range(seg_off.numel() - 1):
            a, b = int(seg_off[s]), int(seg_off[s + 1])
            part = vals[a:b]
            if mask is not None:
                part = part[mask[a:b]]  # nulls never enter the filter
            words = ops.bloom_build(part, self.m_bits, self.k)
            words_per_file.append(words.cpu())
        return {self.out_columns()[0]: torch.stack(words_per_file)
                if words_per_file
                else torch.zeros((0, self.m_bits // 64),
                                 dtype=torch.int64)}

    def convert_predicate(self, pred, sketch_data, dtype_name):
        dtype_name = self.value_type(dtype_name)
        values = None
        if isinstance(pred, BinComp) and pred.op == "=" and \
                isinstance(pred.right, Lit) and \
                self.matches_lhs(pred.left):
            values = [pred.right.value]
        elif isinstance(pred, In) and self.matches_lhs(pred.col):
            values = pred.values
        if values is None:
            return None
        words = sketch_data.tensor(self.out_columns()[0])  # [files, words]
        dt = {"double": torch.float64,
              "float": torch.float32}.get(dtype_name, torch.int64)
        t = torch.tensor(list(values), dtype=dt)
        if dt == torch.float64:
            t = t.view(torch.int64)
        elif dt == torch.float32:
            t = t.to(torch.float64).view(torch.int64)
        else:
            t = t.to(torch.int64)
        # K9 on device: one batched kernel probes every value against
        # every file's filter (host loop fallback on CPU)
        return ops.bloom_probe_many(t.to(words.device), words,
                                    self.m_bits, self.k)

    def to_json(self):
        return {"type": BLOOM_SKETCH_TYPE, "expr": self.expr,
                "fpp": self.fpp,
                "expectedDistinctCountPerFile": self.expected_distinct,
                "dataType": None}

    @staticmethod
    def from_json(d):
        return BloomFilterSketch(
            d["expr"], d.get("fpp", 0.01),
            d.get("expectedDistinctCountPerFile", 10000))


class PartitionSketch(Sketch):
    """First-value sketch over a partition-constant column, kept so
    disjunctions like ``A=1 OR part=1`` stay convertible
    (reference PartitionSketch.scala:38-74 with FirstNullSafe)."""

    @property
    def kind(self):
        return "Partition"

    def out_columns(self):
        return [f"Partition_{self.expr}__first"]

    def aggregate(self, values, seg_off, dtype_name, mask=None):
        firsts = []
        vals = values.cpu()
        mk = mask.cpu() if mask is not None else None
        for s in range(seg_off.numel() - 1):
            a, b = int(seg_off[s]), int(seg_off[s + 1])
            if mk is not None and b > a:
                valid_pos = torch.nonzero(mk[a:b],
                                          as_tuple=False).flatten()
                a = a + int(valid_pos[0]) if valid_pos.numel() else b
            firsts.append(vals[a] if b > a else torch.tensor(
                0, dtype=vals.dtype))
        return {self.out_columns()[0]: torch.stack(firsts) if firsts
                else torch.zeros(0, dtype=vals.dtype)}

    def convert_predicate(self, pred, sketch_data, dtype_name):
        first = sketch_data.tensor(self.out_columns()[0])
        if isinstance(pred, BinComp) and pred.op == "=" and \
                isinstance(pred.right, Lit) and \
                self.matches_lhs(pred.left):
            return first == pred.right.value  # scalar broadcast, any dev
        return None

    def to_json(self):
        return {"type": PARTITION_SKETCH_TYPE, "expr": self.expr,
                "dataType": None}

    @staticmethod
    def from_json(d):
        return PartitionSketch(d["expr"])


_SKETCH_REGISTRY = {
    MINMAX_SKETCH_TYPE: MinMaxSketch,
    BLOOM_SKETCH_TYPE: BloomFilterSketch,
    PARTITION_SKETCH_TYPE: PartitionSketch,
}


def sketch_from_json(d: Dict[str, Any]) -> Sketch:
    cls = _SKETCH_REGISTRY.get(d.get("type"))
    if cls is None:
        raise HyperspaceException(f"Unknown sketch type {d.get('type')}")
    return cls.from_json(d)
