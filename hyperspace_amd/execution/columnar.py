"""Device-resident columnar batch.

The engine's unit of data is a ``ColumnBatch``: an ordered map of column name
-> torch tensor, optionally resident on a GPU.  This replaces the reference's
Spark DataFrame partitions (the reference's data plane is Spark row
iterators; ours is columnar tensors sized for 288 GB HBM3E per GPU).

Strings are dictionary-encoded: a ``StringColumn`` holds int32 codes plus a
sorted value dictionary, so comparisons and sorts on codes are
order-correct.

Nullable columns carry a per-column validity mask (bool tensor, True =
valid) in ``ColumnBatch.masks``; a column with no mask entry is entirely
non-null, so null-free data pays nothing.  Null semantics follow Spark:
comparisons/IN never match null, inner joins drop null keys, sorts are
NULLS FIRST, hash partitioning passes the seed through null keys.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Sequence, Union

import numpy as np
import torch

from ..exceptions import HyperspaceException

_SPARK_TO_TORCH = {
    "long": torch.int64, "integer": torch.int32, "short": torch.int16,
    "byte": torch.int8, "double": torch.float64, "float": torch.float32,
    "boolean": torch.bool,
}


class StringColumn:
    """Dictionary-encoded string column.

    ``codes[i]`` indexes into ``values`` (a sorted list of unique strings),
    so ordering by code == lexicographic ordering by value.
    """

    __slots__ = ("codes", "values")

    def __init__(self, codes: torch.Tensor, values: List[str]):
        assert codes.dtype == torch.int32
        self.codes = codes
        self.values = values

    @staticmethod
    def from_strings(strings: Sequence[str]) -> "StringColumn":
        values, inverse = np.unique(np.asarray(strings, dtype=object),
                                    return_inverse=True)
        return StringColumn(
            torch.from_numpy(inverse.astype(np.int32)), list(values))

    def code_of(self, s: str) -> int:
        """Code for value ``s`` or -1 if absent."""
        import bisect
        i = bisect.bisect_left(self.values, s)
        if i < len(self.values) and self.values[i] == s:
            return i
        return -1

    def searchsorted(self, s: str) -> int:
        import bisect
        return bisect.bisect_left(self.values, s)

    def to(self, device) -> "StringColumn":
        return StringColumn(self.codes.to(device), self.values)

    def gather(self, idx: torch.Tensor) -> "StringColumn":
        return StringColumn(self.codes[idx], self.values)

    def to_numpy(self) -> np.ndarray:
        vals = np.asarray(self.values, dtype=object)
        return vals[self.codes.cpu().numpy()]

    def __len__(self):
        return self.codes.numel()


Column = Union[torch.Tensor, StringColumn]


class ColumnBatch:
    """Ordered name -> column map; all columns share the same row count.

    ``masks`` maps column name -> bool validity tensor (True = valid);
    columns absent from ``masks`` have no nulls.
    """

    def __init__(self, columns: "Dict[str, Column]",
                 masks: "Optional[Dict[str, torch.Tensor]]" = None):
        self.columns: Dict[str, Column] = dict(columns)
        self.masks: Dict[str, torch.Tensor] = dict(masks or {})
        n = None
        for name, col in self.columns.items():
            cn = len(col) if isinstance(col, StringColumn) else col.numel()
            if n is None:
                n = cn
            elif n != cn:
                raise HyperspaceException(
                    f"Column {name} length {cn} != {n}")
        self._num_rows = n or 0
        for name, m in self.masks.items():
            if m.numel() != self._num_rows:
                raise HyperspaceException(
                    f"Mask {name} length {m.numel()} != {self._num_rows}")

    # -- basic accessors --------------------------------------------------
    @property
    def num_rows(self) -> int:
        return self._num_rows

    @property
    def names(self) -> List[str]:
        return list(self.columns.keys())

    def column(self, name: str) -> Column:
        for k, v in self.columns.items():
            if k.lower() == name.lower():
                return v
        raise HyperspaceException(
            f"No column {name}; have {self.names}")

    def tensor(self, name: str) -> torch.Tensor:
        col = self.column(name)
        return col.codes if isinstance(col, StringColumn) else col

    def has_column(self, name: str) -> bool:
        return any(k.lower() == name.lower() for k in self.columns)

    def mask(self, name: str) -> Optional[torch.Tensor]:
        """Validity mask (True = valid) for a column, or None if the
        column has no nulls."""
        for k, v in self.masks.items():
            if k.lower() == name.lower():
                return v
        return None

    def has_nulls(self, name: str) -> bool:
        m = self.mask(name)
        return m is not None and not bool(m.all())

    def _stored_key(self, name: str) -> str:
        for k in self.columns:
            if k.lower() == name.lower():
                return k
        raise HyperspaceException(f"No column {name}")

    @property
    def device(self) -> torch.device:
        for col in self.columns.values():
            t = col.codes if isinstance(col, StringColumn) else col
            return t.device
        return torch.device("cpu")

    def nbytes(self) -> int:
        total = 0
        for col in self.columns.values():
            t = col.codes if isinstance(col, StringColumn) else col
            total += t.numel() * t.element_size()
            if isinstance(col, StringColumn):
                # dictionary payload estimate (host-side, but budgeted so
                # string-heavy cached batches don't under-account)
                total += sum(len(v) for v in col.values)
        for m in self.masks.values():
            total += m.numel() * m.element_size()
        return total

    # -- transforms -------------------------------------------------------
    def select(self, names: Iterable[str]) -> "ColumnBatch":
        out = {}
        for n in names:
            for k, v in self.columns.items():
                if k.lower() == n.lower():
                    out[k] = v
                    break
            else:
                raise HyperspaceException(f"No column {n}")
        masks = {k: m for k, m in self.masks.items() if k in out}
        return ColumnBatch(out, masks)

    def with_column(self, name: str, col: Column,
                    mask: Optional[torch.Tensor] = None) -> "ColumnBatch":
        out = dict(self.columns)
        out[name] = col
        masks = {k: m for k, m in self.masks.items()
                 if k.lower() != name.lower()}
        if mask is not None:
            masks[name] = mask
        return ColumnBatch(out, masks)

    def drop(self, name: str) -> "ColumnBatch":
        out = {k: v for k, v in self.columns.items()
               if k.lower() != name.lower()}
        masks = {k: m for k, m in self.masks.items()
                 if k.lower() != name.lower()}
        return ColumnBatch(out, masks)

    def gather(self, idx: torch.Tensor) -> "ColumnBatch":
        from .. import ops
        return ColumnBatch(
            {k: (StringColumn(ops.gather_rows(v.codes, idx.to(
                     v.codes.device)), v.values)
                 if isinstance(v, StringColumn)
                 else ops.gather_rows(v, idx.to(v.device)))
             for k, v in self.columns.items()},
            {k: m[idx.to(m.device)] for k, m in self.masks.items()})

    def slice(self, start: int, end: int) -> "ColumnBatch":
        return ColumnBatch(
            {k: (StringColumn(v.codes[start:end], v.values)
                 if isinstance(v, StringColumn) else v[start:end])
             for k, v in self.columns.items()},
            {k: m[start:end] for k, m in self.masks.items()})

    def to(self, device) -> "ColumnBatch":
        return ColumnBatch(
            {k: v.to(device) for k, v in self.columns.items()},
            {k: m.to(device) for k, m in self.masks.items()})

    @staticmethod
    def concat(batches: "List[ColumnBatch]") -> "ColumnBatch":
        batches = [b for b in batches if b.num_rows >= 0]
        if not batches:
            return ColumnBatch({})
        names = batches[0].names
        out: Dict[str, Column] = {}
        masks: Dict[str, torch.Tensor] = {}
        for n in names:
            cols = [b.column(n) for b in batches]
            if isinstance(cols[0], StringColumn):
                # merge dictionaries
                merged = sorted(set().union(*[set(c.values) for c in cols]))
                remap_codes = []
                val_index = {v: i for i, v in enumerate(merged)}
                for c in cols:
                    lut = torch.tensor(
                        [val_index[v] for v in c.values], dtype=torch.int32,
                        device=c.codes.device)
                    remap_codes.append(lut[c.codes.long()])
                out[n] = StringColumn(torch.cat(remap_codes), merged)
            else:
                out[n] = torch.cat(cols)
            if any(b.mask(n) is not None for b in batches):
                parts = []
                for b in batches:
                    m = b.mask(n)
                    if m is None:
                        m = torch.ones(b.num_rows, dtype=torch.bool,
                                       device=b.device)
                    parts.append(m)
                masks[n] = torch.cat(parts)
        return ColumnBatch(out, masks)

    # -- conversion -------------------------------------------------------
    @staticmethod
    def from_arrow(table) -> "ColumnBatch":
        """pyarrow.Table -> ColumnBatch (host tensors).  Null slots become
        validity-mask entries; the value buffer holds a fill (0 / "").
        Struct columns flatten into dotted leaf columns ("a.b.c") with
        parent-null propagation — the engine's analog of the reference's
        ``__hs_nested.`` flat aliases (util/ResolverUtils.scala)."""
        import pyarrow as pa
        import pyarrow.compute as pc
        cols: Dict[str, Column] = {}
        masks: Dict[str, torch.Tensor] = {}
        pairs = []
        for name, col in zip(table.column_names, table.columns):
            col = col.combine_chunks()
            if pa.types.is_struct(col.type):
                stack = [(name, col)]
                while stack:
                    pname, pcol = stack.pop()
                    for i, f in enumerate(pcol.type):
                        child = pc.struct_field(pcol, [i])
                        leaf = f"{pname}.{f.name}"
                        if pa.types.is_struct(child.type):
                            stack.append((leaf, child))
                        else:
                            pairs.append((leaf, child))
            else:
                pairs.append((name, col))
        for name, col in pairs:
            if col.null_count:
                masks[name] = torch.from_numpy(
                    np.ascontiguousarray(col.is_valid().to_numpy(
                        zero_copy_only=False)))
                if pa.types.is_string(col.type) or \
                        pa.types.is_large_string(col.type):
                    col = col.fill_null("")
                else:
                    col = col.fill_null(0)
            if pa.types.is_string(col.type) or pa.types.is_large_string(
                    col.type):
                cols[name] = StringColumn.from_strings(col.to_pylist())
            elif pa.types.is_decimal(col.type):
                # decimal(p<=18, s): unscaled int64 representation —
                # order-correct for a fixed scale; the scale rides in
                # the schema type string ("decimal(p,s)") and literals
                # are scaled at predicate binding (dataframe.py)
                if col.type.precision > 18:
                    raise HyperspaceException(
                        f"decimal precision > 18 unsupported: {name}")
                s = col.type.scale
                vals = [0 if d is None else int(d.scaleb(s))
                        for d in col.to_pylist()]
                cols[name] = torch.tensor(vals, dtype=torch.int64)
            else:
                np_arr = col.to_numpy(zero_copy_only=False)
                if np_arr.dtype == np.dtype("datetime64[us]") or \
                        np_arr.dtype.kind == "M":
                    np_arr = np_arr.astype("int64")
                np_arr = np.ascontiguousarray(np_arr)
                if not np_arr.flags.writeable:
                    np_arr = np_arr.copy()
                cols[name] = torch.from_numpy(np_arr)
        return ColumnBatch(cols, masks)

    def to_arrow(self):
        import pyarrow as pa
        arrays = {}
        for k, v in self.columns.items():
            m = self.mask(k)
            null_mask = (~m.cpu()).numpy() if m is not None else None
            if isinstance(v, StringColumn):
                arrays[k] = pa.array(v.to_numpy(), mask=null_mask)
            else:
                arrays[k] = pa.array(v.cpu().numpy(), mask=null_mask)
        return pa.table(arrays)

    def to_numpy(self) -> Dict[str, np.ndarray]:
        return {k: (v.to_numpy() if isinstance(v, StringColumn)
                    else v.cpu().numpy())
                for k, v in self.columns.items()}

    def __repr__(self):
        parts = []
        for k, v in self.columns.items():
            t = v.codes if isinstance(v, StringColumn) else v
            kind = "str" if isinstance(v, StringColumn) else str(t.dtype)
            parts.append(f"{k}:{kind}")
        return (f"ColumnBatch[{self._num_rows} rows, "
                f"{', '.join(parts)}, dev={self.device}]")
