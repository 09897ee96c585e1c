"""CSV / JSON-lines / ORC / Avro file sources.

Reference: the default source supports avro,csv,json,orc,parquet,text
(util/HyperspaceConf.scala:110-115).  These formats have no device
decode path — they are read on host (pyarrow.csv / pyarrow.json) and
uploaded; indexes BUILT from them are native Parquet, so queries served
from an index still run the device pipeline.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

from .interfaces import FileBasedRelation, FileBasedSourceProvider
from .parquet_source import list_data_files
from ..exceptions import HyperspaceException
from ..log.entry import FileInfo, Relation, Schema
from ..utils.hashing import md5_hex


class TextFormatRelation(FileBasedRelation):
    """Shared implementation for csv/json/orc/avro directories."""

    def __init__(self, fmt: str, root_paths: List[str],
                 options: Optional[Dict[str, str]] = None):
        assert fmt in ("csv", "json", "orc", "avro", "text")
        self._fmt = fmt
        self._root_paths = [os.path.abspath(p) for p in root_paths]
        self._options = dict(options or {})
        self._schema: Optional[Schema] = None

    @property
    def root_paths(self):
        return self._root_paths

    @property
    def file_format(self):
        return self._fmt

    @property
    def options(self):
        return self._options

    def _suffix(self):
        return ".txt" if self._fmt == "text" else "." + self._fmt

    def all_files(self) -> List[FileInfo]:
        infos = []
        for p in list_data_files(self._root_paths, suffix=self._suffix()):
            st = os.stat(p)
            infos.append(FileInfo(p, st.st_size, int(st.st_mtime * 1000)))
        return infos

    def read_table(self, path: str):
        if self._fmt == "csv":
            from pyarrow import csv
            return csv.read_csv(path)
        if self._fmt == "orc":
            from pyarrow import orc
            return orc.read_table(path)
        if self._fmt == "avro":
            from .avro_io import read_avro
            return read_avro(path)
        if self._fmt == "text":
            # one string column "value" per line (Spark text source)
            import pyarrow as pa
            with open(path, "r", errors="replace") as f:
                lines = f.read().split("\n")
            if lines and lines[-1] == "":
                lines.pop()
            return pa.table({"value": lines})
        from pyarrow import json as pa_json
        return pa_json.read_json(path)

    @property
    def schema(self) -> Schema:
        if self._schema is None:
            files = self.all_files()
            if not files:
                raise HyperspaceException(
                    f"No {self._fmt} files under {self._root_paths}")
            self._schema = Schema.from_arrow(
                self.read_table(files[0].name).schema)
        return self._schema

    def signature(self) -> str:
        parts = [f"{f.name},{f.size},{f.modifiedTime}"
                 for f in sorted(self.all_files(), key=lambda f: f.name)]
        return md5_hex(self._fmt + "\n" + "\n".join(parts))

    def refreshed(self):
        return TextFormatRelation(self._fmt, self._root_paths,
                                  self._options)

    def describe(self) -> str:
        return f"{self._fmt}:{','.join(self._root_paths)}"

    def read_files(self, paths, columns, device):
        import pyarrow as pa
        from ..execution.columnar import ColumnBatch
        tables = [self.read_table(p) for p in paths]
        row_counts = [t.num_rows for t in tables]
        if not tables:
            return ColumnBatch({}), []
        table = pa.concat_tables(tables, promote_options="default")
        if columns:
            want = [c for c in table.column_names
                    if c.lower() in {x.lower() for x in columns}]
            table = table.select(want)
        batch = ColumnBatch.from_arrow(table)
        if getattr(device, "type", "cpu") == "cuda":
            batch = batch.to(device)
        return batch, row_counts


class TextFormatSourceProvider(FileBasedSourceProvider):
    def supports(self, relation) -> bool:
        return isinstance(relation, TextFormatRelation)

    def from_metadata(self, metadata: Relation):
        if metadata.fileFormat not in ("csv", "json", "orc",
                                       "avro", "text"):
            return None
        return TextFormatRelation(metadata.fileFormat, metadata.rootPaths,
                                  metadata.options)
