"""Delta-style transactional table source.

Reference: index/sources/delta/ — DeltaLakeRelation (signature = table
version + path, delta/DeltaLakeRelation.scala:40-44; versionAsOf time
travel; files from the transaction log) and DeltaLakeRelationMetadata
(deltaVersions index-to-table version history property,
delta/DeltaLakeRelationMetadata.scala:39-59).

The table format here is a minimal native transaction log (the engine is
Spark-free, so it defines its own):

    <table>/_delta_log/<version>.json   {"version", "timestamp",
                                         "add": [{path,size,mtime}...],
                                         "remove": [path...]}
    <table>/*.parquet                   data files

Appends and deletes commit new log versions with atomic hard-link claims
(same optimistic protocol as the index op log).
"""

from __future__ import annotations

import json
import os
import tempfile
import time
from typing import Dict, List, Optional

from .interfaces import FileBasedRelation, FileBasedSourceProvider
from ..exceptions import HyperspaceException
from ..log.entry import FileInfo, Relation, Schema

LOG_DIR = "_delta_log"


class DeltaTable:
    """Minimal append/delete table with a versioned commit log."""

    def __init__(self, path: str):
        self.path = os.path.abspath(path)
        self.log_dir = os.path.join(self.path, LOG_DIR)

    # -- log --------------------------------------------------------------
    def versions(self) -> List[int]:
        if not os.path.isdir(self.log_dir):
            return []
        return sorted(int(f[:-5]) for f in os.listdir(self.log_dir)
                      if f.endswith(".json") and f[:-5].isdigit())

    @property
    def version(self) -> int:
        vs = self.versions()
        if not vs:
            raise HyperspaceException(f"Not a delta table: {self.path}")
        return vs[-1]

    def _commit(self, version: int, add: List[Dict], remove: List[str]):
        os.makedirs(self.log_dir, exist_ok=True)
        entry = {"version": version, "timestamp": int(time.time() * 1000),
                 "add": add, "remove": remove}
        target = os.path.join(self.log_dir, f"{version:010d}.json")
        fd, tmp = tempfile.mkstemp(dir=self.log_dir, prefix=".tmp_")
        try:
            with os.fdopen(fd, "w") as f:
                json.dump(entry, f)
            try:
                os.link(tmp, target)
            except FileExistsError:
                raise HyperspaceException(
                    f"Concurrent commit lost race for version {version}")
        finally:
            os.unlink(tmp)

    # -- operations -------------------------------------------------------
    @staticmethod
    def create(path: str) -> "DeltaTable":
        t = DeltaTable(path)
        os.makedirs(path, exist_ok=True)
        if not t.versions():
            t._commit(0, [], [])
        return t

    def append_files(self, paths: List[str]) -> int:
        add = []
        for p in paths:
            st = os.stat(p)
            add.append({"path": os.path.abspath(p), "size": st.st_size,
                        "mtime": int(st.st_mtime * 1000)})
        v = self.version + 1
        self._commit(v, add, [])
        return v

    def append_batch(self, batch, name_hint: str = "part") -> int:
        """Write a ColumnBatch as a new data file + commit."""
        from .parquet_io import write_batch_parquet
        import uuid
        p = os.path.join(self.path,
                         f"{name_hint}-{uuid.uuid4().hex[:12]}.parquet")
        write_batch_parquet(batch, p)
        return self.append_files([p])

    def remove_files(self, paths: List[str]) -> int:
        v = self.version + 1
        self._commit(v, [], [os.path.abspath(p) for p in paths])
        return v

    # -- snapshots ---------------------------------------------------------
    def files_at(self, version: Optional[int] = None) -> List[FileInfo]:
        target = self.version if version is None else version
        live: Dict[str, FileInfo] = {}
        for v in self.versions():
            if v > target:
                break
            with open(os.path.join(self.log_dir, f"{v:010d}.json")) as f:
                entry = json.load(f)
            for a in entry.get("add", []):
                live[a["path"]] = FileInfo(a["path"], a["size"], a["mtime"])
            for r in entry.get("remove", []):
                live.pop(r, None)
        return sorted(live.values(), key=lambda f: f.name)


class DeltaTableRelation(FileBasedRelation):
    """Relation over a DeltaTable snapshot.

    Signature = table version + path (reference
    delta/DeltaLakeRelation.scala:40-44): any commit changes the
    signature; time travel pins ``version_as_of``.
    """

    def __init__(self, path: str, version_as_of: Optional[int] = None,
                 options: Optional[Dict[str, str]] = None):
        self.table = DeltaTable(path)
        self.version_as_of = version_as_of
        self._options = dict(options or {})
        if version_as_of is not None:
            self._options["versionAsOf"] = str(version_as_of)
        self._schema: Optional[Schema] = None

    @property
    def root_paths(self):
        return [self.table.path]

    @property
    def file_format(self):
        return "delta"

    @property
    def options(self):
        return self._options

    @property
    def snapshot_version(self) -> int:
        return (self.version_as_of if self.version_as_of is not None
                else self.table.version)

    @property
    def schema(self) -> Schema:
        if self._schema is None:
            files = self.all_files()
            if not files:
                raise HyperspaceException(
                    f"Empty delta table: {self.table.path}")
            import pyarrow.parquet as pq
            self._schema = Schema.from_arrow(pq.read_schema(files[0].name))
        return self._schema

    def all_files(self) -> List[FileInfo]:
        return self.table.files_at(self.version_as_of)

    def signature(self) -> str:
        return f"{self.snapshot_version}.{self.table.path}"

    def refreshed(self) -> "DeltaTableRelation":
        # pinned snapshots stay pinned; live relations follow the log
        return DeltaTableRelation(self.table.path, self.version_as_of)

    def describe(self) -> str:
        v = f"@v{self.version_as_of}" if self.version_as_of is not None \
            else ""
        return f"delta:{self.table.path}{v}"

    def closest_index_log_entry(self, entry, log_manager):
        """Time-travel index selection (reference
        delta/DeltaLakeRelation.scala:179-251, closestIndex): for a query
        pinned at table version Q, pick the retained index log version
        whose recorded delta version is nearest to Q — preferring the
        newest version at or below Q (its data is a subset-superset
        closest in bytes) — and load that log entry."""
        if self.version_as_of is None:
            return None
        hist = (entry.properties or {}).get(
            "deltaVersions",
            entry.derivedDataset.properties.get("deltaVersions", ""))
        pairs = []
        for tok in hist.split(","):
            if ":" in tok:
                k, v = tok.split(":", 1)
                pairs.append((int(k), int(v)))
        if not pairs:
            return None
        q = self.snapshot_version
        below = [(k, v) for k, v in pairs if v <= q]
        log_id, _ = (max(below, key=lambda p: (p[1], p[0])) if below
                     else min(pairs, key=lambda p: (p[1], -p[0])))
        from ..log.constants import States
        e = log_manager.get_log(log_id)
        if e is None or e.state != States.ACTIVE:
            return None
        return e

    def enrich_index_properties(self, properties: Dict[str, str],
                                index_log_version: int) -> Dict[str, str]:
        """Maintain the index->table version history (reference
        DeltaLakeRelationMetadata: 'deltaVersions' property of
        indexVersion:tableVersion pairs)."""
        out = dict(properties)
        history = out.get("deltaVersions", "")
        pair = f"{index_log_version}:{self.snapshot_version}"
        out["deltaVersions"] = f"{history},{pair}" if history else pair
        return out


class DeltaTableSourceProvider(FileBasedSourceProvider):
    def supports(self, relation) -> bool:
        return isinstance(relation, DeltaTableRelation)

    def from_metadata(self, metadata: Relation
                      ) -> Optional[FileBasedRelation]:
        if metadata.fileFormat != "delta":
            return None
        version_as_of = metadata.options.get("versionAsOf")
        return DeltaTableRelation(
            metadata.rootPaths[0],
            int(version_as_of) if version_as_of is not None else None)
