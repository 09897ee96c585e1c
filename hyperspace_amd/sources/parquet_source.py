"""Default Parquet source provider.

Reference: index/sources/default/ (DefaultFileBasedRelation: signature =
md5 chain over sorted (path,size,mtime), default/DefaultFileBasedRelation.scala:45-53;
DataPathFilter excludes files starting with '_' or '.', util/PathUtils.scala).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

from .interfaces import FileBasedRelation, FileBasedSourceProvider
from ..log.entry import FileInfo, Relation, Schema
from ..utils.hashing import md5_hex
from ..exceptions import HyperspaceException


def _is_data_file(name: str) -> bool:
    base = os.path.basename(name)
    return not (base.startswith("_") or base.startswith("."))


def list_data_files(root_paths: List[str], suffix: str = "") -> List[str]:
    out: List[str] = []
    for root in root_paths:
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
            self._schema = Schema.from_arrow(
                pq.read_schema(files[0].name))
        return self._schema

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


class ParquetSourceProvider(FileBasedSourceProvider):
    def supports(self, relation) -> bool:
        return isinstance(relation, ParquetRelation)

    def from_metadata(self, metadata: Relation
                      ) -> Optional[FileBasedRelation]:
        if metadata.fileFormat != "parquet":
            return None
        return ParquetRelation(metadata.rootPaths, metadata.options)
