"""Default Parquet source provider.

Reference: index/sources/default/ (DefaultFileBasedRelation: signature =
md5 chain over sorted (path,size,mtime), default/DefaultFileBasedRelation.scala:45-53;
DataPathFilter excludes files starting with '_' or '.', util/PathUtils.scala;
hive-style partition discovery + partition base path,
default/DefaultFileBasedRelation.scala:75-89).

Hive partitioning: ``root/key=value/...`` directory segments become
table columns (appended after the file schema).  Partition values
attach as constant per-file columns at read time, and partition-only
predicates prune files before any IO (Spark's partition pruning).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Optional

from .interfaces import FileBasedRelation, FileBasedSourceProvider
from ..log.entry import FileInfo, Relation, Schema, SchemaField
from ..utils.hashing import md5_hex
from ..exceptions import HyperspaceException


def partition_values_of(root_paths: List[str], path: str
                        ) -> "Dict[str, str]":
    """key=value directory segments between the matching root and the
    file, in order."""
    for root in root_paths:
        r = os.path.abspath(root)
        if path.startswith(r + os.sep):
            rel = os.path.relpath(os.path.dirname(path), r)
            out: Dict[str, str] = {}
            for seg in rel.split(os.sep):
                if "=" in seg:
                    k, v = seg.split("=", 1)
                    out[k] = v
            return out
    return {}


def _infer_partition_type(values: List[str]) -> str:
    try:
        for v in values:
            int(v)
        return "long"
    except ValueError:
        pass
    try:
        for v in values:
            float(v)
        return "double"
    except ValueError:
        return "string"


def _cast_partition_value(v: str, spark_type: str) -> Any:
    if spark_type == "long":
        return int(v)
    if spark_type == "double":
        return float(v)
    return v


def _is_data_file(name: str) -> bool:
    base = os.path.basename(name)
    return not (base.startswith("_") or base.startswith("."))


def list_data_files(root_paths: List[str], suffix: str = "") -> List[str]:
    # globbing paths (reference: GLOBBING_PATTERN_KEY /
    # SparkHadoopUtil.globPathIfNecessary): patterns expand before the
    # walk and must match at least one path
    import glob as _glob
    expanded: List[str] = []
    for root in root_paths:
        if any(ch in root for ch in "*?["):
            matches = sorted(_glob.glob(root))
            if not matches:
                raise HyperspaceException(
                    f"Glob pattern matched nothing: {root}")
            expanded.extend(matches)
        else:
            expanded.append(root)
    out: List[str] = []
    for root in expanded:
        if os.path.isfile(root):
            if _is_data_file(root):
                out.append(os.path.abspath(root))
            continue
        if not os.path.isdir(root):
            raise HyperspaceException(f"No such path: {root}")
        for dirpath, dirnames, filenames in os.walk(root):
            dirnames[:] = [d for d in dirnames if _is_data_file(d)]
            for fn in filenames:
                if _is_data_file(fn) and fn.endswith(suffix):
                    out.append(os.path.abspath(os.path.join(dirpath, fn)))
    return sorted(out)


class ParquetRelation(FileBasedRelation):
    def __init__(self, root_paths: List[str],
                 options: Optional[Dict[str, str]] = None,
                 schema: Optional[Schema] = None):
        self._root_paths = [os.path.abspath(p) for p in root_paths]
        self._options = options or {}
        self._schema = schema
        self._files: Optional[List[FileInfo]] = None

    @property
    def root_paths(self):
        return self._root_paths

    @property
    def file_format(self):
        return "parquet"

    @property
    def options(self):
        return self._options

    @property
    def schema(self) -> Schema:
        if self._schema is None:
            files = self.all_files()
            if not files:
                raise HyperspaceException(
                    f"No parquet files under {self._root_paths}")
            import pyarrow.parquet as pq
            sch = Schema.from_arrow(pq.read_schema(files[0].name))
            pschema = self.partition_schema()
            existing = {f.name.lower() for f in sch.fields}
            sch.fields.extend(f for f in pschema.fields
                              if f.name.lower() not in existing)
            self._schema = sch
        return self._schema

    # -- hive partitioning -------------------------------------------------
    def partition_schema(self) -> Schema:
        """Partition columns inferred from key=value path segments."""
        if getattr(self, "_pschema", None) is None:
            per_key: Dict[str, List[str]] = {}
            order: List[str] = []
            for f in self.all_files():
                for k, v in partition_values_of(self._root_paths,
                                                f.name).items():
                    if k not in per_key:
                        per_key[k] = []
                        order.append(k)
                    per_key[k].append(v)
            self._pschema = Schema([
                SchemaField(k, _infer_partition_type(per_key[k]), False)
                for k in order])
        return self._pschema

    def partition_values(self, path: str) -> Dict[str, Any]:
        """Typed partition values for one data file."""
        pschema = self.partition_schema()
        raw = partition_values_of(self._root_paths, path)
        return {f.name: _cast_partition_value(raw[f.name], f.type)
                for f in pschema.fields if f.name in raw}

    def read_files(self, paths: List[str], columns, device):
        return partitioned_read_files(
            self, paths, columns, device,
            lambda p, c, d: super(ParquetRelation, self).read_files(
                p, c, d))

    def prune_partitions(self, cond) -> Optional[List[str]]:
        """Files surviving the partition-column conjuncts of ``cond``,
        or None when the predicate has no partition conjunct (Spark's
        partition pruning: evaluated on metadata, before any IO)."""
        return prune_partitions_generic(self, cond)

    def all_files(self) -> List[FileInfo]:
        # re-listed on every call (like Spark's InMemoryFileIndex refresh
        # semantics across queries): a stale cache would let a changed
        # source pass the signature check
        infos = []
        for p in list_data_files(self._root_paths, suffix=".parquet"):
            st = os.stat(p)
            infos.append(FileInfo(p, st.st_size, int(st.st_mtime * 1000)))
        return infos

    def signature(self) -> str:
        """md5 over sorted (path,size,mtime) — matches the reference's
        default FileBasedSignatureProvider semantics."""
        parts = [f"{f.name},{f.size},{f.modifiedTime}"
                 for f in sorted(self.all_files(), key=lambda f: f.name)]
        return md5_hex("\n".join(parts))

    def refreshed(self) -> "ParquetRelation":
        return ParquetRelation(self._root_paths, self._options, None)


def partitioned_read_files(relation, paths: List[str], columns, device,
                           base_read):
    """Generic partitioned read: file columns come from ``base_read``,
    partition columns materialize as per-file constants (Spark attaches
    partition values the same way).  Shared by the hive-style default
    source and the Delta source (partitionValues from the action log)."""
    pschema = relation.partition_schema()
    if not pschema.fields:
        return base_read(paths, columns, device)
    import torch
    from ..execution.columnar import ColumnBatch, StringColumn
    pnames = {f.name.lower(): f for f in pschema.fields}
    file_cols = None
    part_wanted = [f for f in pschema.fields]
    if columns is not None:
        file_cols = [c for c in columns if c.lower() not in pnames]
        part_wanted = [pnames[c.lower()] for c in columns
                       if c.lower() in pnames]
    if file_cols == []:
        # partition-only projection: row counts from footers, no IO
        import pyarrow.parquet as pq
        row_counts = [pq.ParquetFile(p).metadata.num_rows
                      for p in paths]
        batch = ColumnBatch({})
    else:
        batch, row_counts = base_read(paths, file_cols, device)
    dev = batch.device if batch.columns else (
        device if getattr(device, "type", "cpu") == "cuda"
        else torch.device("cpu"))
    cols = dict(batch.columns)
    for f in part_wanted:
        vals = [relation.partition_values(p).get(f.name) for p in paths]
        if f.type == "string":
            uniq = sorted(set(vals))
            code_of = {v: i for i, v in enumerate(uniq)}
            codes = torch.cat([
                torch.full((rc,), code_of[v], dtype=torch.int32)
                for v, rc in zip(vals, row_counts)]) \
                if row_counts else torch.empty(0, dtype=torch.int32)
            cols[f.name] = StringColumn(codes.to(dev), uniq)
        else:
            dt = torch.int64 if f.type in ("long", "integer", "date") \
                else torch.float64
            cols[f.name] = torch.cat([
                torch.full((rc,), v, dtype=dt)
                for v, rc in zip(vals, row_counts)]).to(dev) \
                if row_counts else torch.empty(0, dtype=dt)
    out = ColumnBatch(cols, dict(batch.masks))
    if columns is not None:
        out = out.select(columns)
    return out, row_counts


def prune_partitions_generic(relation, cond) -> Optional[List[str]]:
    """Partition pruning over any relation exposing partition_schema()
    and partition_values(path)."""
    pschema = relation.partition_schema()
    if not pschema.fields:
        return None
    from ..plan.expr import split_conjunctive
    pnames = {f.name.lower() for f in pschema.fields}
    conjuncts = [c for c in split_conjunctive(cond)
                 if c.references() and
                 {r.lower() for r in c.references()} <= pnames]
    if not conjuncts:
        return None
    kept = []
    for f in relation.all_files():
        vals = {k.lower(): v
                for k, v in relation.partition_values(f.name).items()}
        if all(_eval_on_values(c, vals) for c in conjuncts):
            kept.append(f.name)
    return kept


def _eval_on_values(e, vals: Dict[str, Any]) -> bool:
    """Evaluate a predicate over one file's partition values (SQL null
    semantics: a missing key never matches a comparison)."""
    from ..plan.expr import (And, BinComp, Col, In, IsNotNull, IsNull,
                             Lit, Not, Or)
    if isinstance(e, And):
        return _eval_on_values(e.left, vals) and \
            _eval_on_values(e.right, vals)
    if isinstance(e, Or):
        return _eval_on_values(e.left, vals) or \
            _eval_on_values(e.right, vals)
    if isinstance(e, Not):
        return not _eval_on_values(e.child, vals)
    if isinstance(e, IsNull):
        return vals.get(e.col.name.lower()) is None
    if isinstance(e, IsNotNull):
        return vals.get(e.col.name.lower()) is not None
    if isinstance(e, In):
        v = vals.get(e.col.name.lower())
        return v is not None and v in set(e.values)
    if isinstance(e, BinComp) and isinstance(e.left, Col) and \
            isinstance(e.right, Lit):
        v = vals.get(e.left.name.lower())
        if v is None:
            return False
        r = e.right.value
        if isinstance(v, (int, float)) and isinstance(r, str):
            return False
        if isinstance(v, str) and isinstance(r, (int, float)):
            return False
        return {"=": v == r, "!=": v != r, "<": v < r, "<=": v <= r,
                ">": v > r, ">=": v >= r}[e.op]
    return True  # unknown form: keep the file (safe)


class ParquetSourceProvider(FileBasedSourceProvider):
    def supports(self, relation) -> bool:
        return isinstance(relation, ParquetRelation)

    def from_metadata(self, metadata: Relation
                      ) -> Optional[FileBasedRelation]:
        if metadata.fileFormat != "parquet":
            return None
        return ParquetRelation(metadata.rootPaths, metadata.options)
