"""Device-resident columnar batch.

The engine's unit of data is a ``ColumnBatch``: an ordered map of column name
-> torch tensor, optionally resident on a GPU.  This replaces the reference's
Spark DataFrame partitions (the reference's data plane is Spark row
iterators; ours is columnar tensors sized for 288 GB HBM3E per GPU).

Strings are dictionary-encoded: a ``StringColumn`` holds int32 codes plus a
sorted value dictionary, so comparisons and sorts on codes are
order-correct.  Nulls are not supported in v0 (synthetic + TPC-H-shaped data
is non-null); validity masks are a planned extension.
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
    """Ordered name -> column map; all columns share the same row count."""

    def __init__(self, columns: "Dict[str, Column]"):
        self.columns: Dict[str, Column] = dict(columns)
        n = None
        for name, col in self.columns.items():
            cn = len(col) if isinstance(col, StringColumn) else col.numel()
            if n is None:
                n = cn
            elif n != cn:
                raise HyperspaceException(
                    f"Column {name} length {cn} != {n}")
        self._num_rows = n or 0

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
        return ColumnBatch(out)

    def with_column(self, name: str, col: Column) -> "ColumnBatch":
        out = dict(self.columns)
        out[name] = col
        return ColumnBatch(out)

    def drop(self, name: str) -> "ColumnBatch":
        out = {k: v for k, v in self.columns.items()
               if k.lower() != name.lower()}
        return ColumnBatch(out)

    def gather(self, idx: torch.Tensor) -> "ColumnBatch":
        from .. import ops
        return ColumnBatch({
            k: (StringColumn(ops.gather_rows(v.codes, idx.to(
                    v.codes.device)), v.values)
                if isinstance(v, StringColumn)
                else ops.gather_rows(v, idx.to(v.device)))
            for k, v in self.columns.items()})

    def slice(self, start: int, end: int) -> "ColumnBatch":
        return ColumnBatch({
            k: (StringColumn(v.codes[start:end], v.values)
                if isinstance(v, StringColumn) else v[start:end])
            for k, v in self.columns.items()})

    def to(self, device) -> "ColumnBatch":
        return ColumnBatch({
            k: v.to(device) for k, v in self.columns.items()})

    @staticmethod
    def concat(batches: "List[ColumnBatch]") -> "ColumnBatch":
        batches = [b for b in batches if b.num_rows >= 0]
        if not batches:
            return ColumnBatch({})
        names = batches[0].names
        out: Dict[str, Column] = {}
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
        return ColumnBatch(out)

    # -- conversion -------------------------------------------------------
    @staticmethod
    def from_arrow(table) -> "ColumnBatch":
        """pyarrow.Table -> ColumnBatch (host tensors)."""
        import pyarrow as pa
        cols: Dict[str, Column] = {}
        for name, col in zip(table.column_names, table.columns):
            col = col.combine_chunks()
            if pa.types.is_string(col.type) or pa.types.is_large_string(
                    col.type):
                cols[name] = StringColumn.from_strings(col.to_pylist())
            else:
                np_arr = col.to_numpy(zero_copy_only=False)
                if np_arr.dtype == np.dtype("datetime64[us]") or \
                        np_arr.dtype.kind == "M":
                    np_arr = np_arr.astype("int64")
                np_arr = np.ascontiguousarray(np_arr)
                if not np_arr.flags.writeable:
                    np_arr = np_arr.copy()
                cols[name] = torch.from_numpy(np_arr)
        return ColumnBatch(cols)

    def to_arrow(self):
        import pyarrow as pa
        arrays = {}
        for k, v in self.columns.items():
            if isinstance(v, StringColumn):
                arrays[k] = pa.array(v.to_numpy())
            else:
                arrays[k] = pa.array(v.cpu().numpy())
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
