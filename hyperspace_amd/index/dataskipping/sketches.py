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

# single binary arithmetic over a column and a numeric literal, e.g.
# "a % 10" — the supported scalar-expression subset (the reference
# accepts arbitrary deterministic scalar expressions via Catalyst;
# here the predicate language is comparison-based, so sketches bind to
# the same single-op arithmetic the Expr API can express)
_ARITH_RE = re.compile(
    r"^\s*([A-Za-z_][\w.]*)\s*([+\-*/%])\s*(-?\d+(?:\.\d+)?)\s*$")


def parse_sketch_expr(s: str):
    """'col' or 'col OP literal' -> (col_name, op|None, literal|None)."""
    m = _ARITH_RE.match(s)
    if m:
        col_s, op, lit = m.groups()
        value = float(lit) if "." in lit else int(lit)
        return col_s, op, value
    return s.strip(), None, None


def eval_sketch_values(values: torch.Tensor, op, lit) -> torch.Tensor:
    """Apply the sketch's arithmetic to a column tensor.  ``%`` uses
    Java/Spark remainder semantics (sign of the dividend: fmod)."""
    if op is None:
        return values
    if op == "+":
        return values + lit
    if op == "-":
        return values - lit
    if op == "*":
        return values * lit
    if op == "%":
        return torch.fmod(values, lit)
    return values.to(torch.float64) / lit  # '/' is double division

MINMAX_SKETCH_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.sketches.MinMaxSketch")
BLOOM_SKETCH_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.sketches."
    "BloomFilterSketch")
PARTITION_SKETCH_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.sketches.PartitionSketch")


class Sketch:
    """Base sketch over a single source column expression (a column
    name or ``col OP literal`` arithmetic)."""

    def __init__(self, expr: str):
        self.expr = expr

    @property
    def base_column(self) -> str:
        return parse_sketch_expr(self.expr)[0]

    def _parsed(self):
        return parse_sketch_expr(self.expr)

    def rebind(self, resolved_col: str) -> None:
        """Rewrite the expression with the schema-resolved column."""
        _, op, lit = self._parsed()
        self.expr = (f"{resolved_col} {op} {lit}" if op is not None
                     else resolved_col)

    def compute_values(self, col_values: torch.Tensor) -> torch.Tensor:
        _, op, lit = self._parsed()
        return eval_sketch_values(col_values, op, lit)

    def value_type(self, col_type: str) -> str:
        _, op, _ = self._parsed()
        if op == "/":
            return "double"
        return col_type

    def matches_lhs(self, lhs: Expr) -> bool:
        """True when a predicate's left-hand side is structurally this
        sketch's expression."""
        name, op, lit = self._parsed()
        if op is None:
            return isinstance(lhs, Col) and \
                lhs.name.lower() == name.lower()
        return isinstance(lhs, Arith) and _expr_eq(
            lhs, Arith(op, Col(name), Lit(lit)))

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
            out = torch.zeros(mn.numel(), dtype=torch.bool)
            for value in pred.values:
                v = _norm_scalar(value, dtype_name)
                out |= (mn <= v) & (mx >= v)
            return out
        if isinstance(pred, IsNotNull) and self.matches_lhs(pred.col):
            return torch.ones(mn.numel(), dtype=torch.bool)
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
        n_files = words.shape[0]
        out = torch.zeros(n_files, dtype=torch.bool)
        for v in values:
            dt = {"double": torch.float64,
                  "float": torch.float32}.get(dtype_name, torch.int64)
            t = torch.tensor([v], dtype=dt)
            if dt == torch.float64:
                t = t.view(torch.int64)
            elif dt == torch.float32:
                t = t.to(torch.float64).view(torch.int64)
            else:
                t = t.to(torch.int64)
            for f in range(n_files):
                hit = ops.cpu_ref.bloom_probe(t, words[f], self.m_bits,
                                              self.k)
                out[f] |= bool(hit[0])
        return out

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
            v = torch.tensor(pred.right.value, dtype=first.dtype)
            return first == v
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
