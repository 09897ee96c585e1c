"""Iceberg-style snapshot table source.

Reference: index/sources/iceberg/ — IcebergRelation (signature =
snapshotId + table location, iceberg/IcebergRelation.scala:63-68; files
from file-scan tasks; snapshot time travel).

The table format here mirrors Iceberg's HadoopTables layout minimally:

    <table>/metadata/v<N>.metadata.json   {"snapshot-id", "timestamp-ms",
                                           "manifest": [{path,size,mtime}]}
    <table>/metadata/version-hint.text    current N
    <table>/*.parquet                     data files

Each commit writes a full manifest (Iceberg snapshots are self-contained
file listings, unlike the delta source's log of deltas).
"""

from __future__ import annotations

import json
import os
import tempfile
import time
import uuid
from typing import Dict, List, Optional

from .interfaces import FileBasedRelation, FileBasedSourceProvider
from ..exceptions import HyperspaceException
from ..log.entry import FileInfo, Relation, Schema

META_DIR = "metadata"


class IcebergTable:
    def __init__(self, path: str):
        self.path = os.path.abspath(path)
        self.meta_dir = os.path.join(self.path, META_DIR)

    @staticmethod
    def create(path: str) -> "IcebergTable":
        t = IcebergTable(path)
        os.makedirs(t.meta_dir, exist_ok=True)
        if t.current_version() is None:
            t._commit([], snapshot_id=0)
        return t

    # -- metadata ----------------------------------------------------------
    def current_version(self) -> Optional[int]:
        hint = os.path.join(self.meta_dir, "version-hint.text")
        try:
            with open(hint) as f:
                return int(f.read().strip())
        except FileNotFoundError:
            return None

    def _snapshot(self, version: int) -> Dict:
        with open(os.path.join(self.meta_dir,
                               f"v{version}.metadata.json")) as f:
            return json.load(f)

    def snapshots(self) -> List[Dict]:
        if not os.path.isdir(self.meta_dir):
            return []
        out = []
        for name in sorted(os.listdir(self.meta_dir)):
            if name.startswith("v") and name.endswith(".metadata.json"):
                with open(os.path.join(self.meta_dir, name)) as f:
                    out.append(json.load(f))
        return out

    @property
    def snapshot_id(self) -> int:
        v = self.current_version()
        if v is None:
            raise HyperspaceException(f"Not an iceberg table: {self.path}")
        return self._snapshot(v)["snapshot-id"]

    def _commit(self, manifest: List[Dict], snapshot_id: Optional[int]
                = None):
        v = (self.current_version() or 0) + 1 \
            if self.current_version() is not None else 0
        if snapshot_id is None:
            snapshot_id = uuid.uuid4().int & 0x7FFFFFFFFFFFFFFF
        entry = {"snapshot-id": snapshot_id,
                 "timestamp-ms": int(time.time() * 1000),
                 "manifest": manifest}
        target = os.path.join(self.meta_dir, f"v{v}.metadata.json")
        fd, tmp = tempfile.mkstemp(dir=self.meta_dir, prefix=".tmp_")
        try:
            with os.fdopen(fd, "w") as f:
                json.dump(entry, f)
            try:
                os.link(tmp, target)
            except FileExistsError:
                raise HyperspaceException(
                    f"Concurrent commit lost race for v{v}")
        finally:
            os.unlink(tmp)
        with open(os.path.join(self.meta_dir, "version-hint.text"),
                  "w") as f:
            f.write(str(v))

    # -- operations --------------------------------------------------------
    def _manifest_now(self) -> List[Dict]:
        v = self.current_version()
        return list(self._snapshot(v)["manifest"]) if v is not None else []

    def append_files(self, paths: List[str]):
        manifest = self._manifest_now()
        for p in paths:
            st = os.stat(p)
            manifest.append({"path": os.path.abspath(p),
                             "size": st.st_size,
                             "mtime": int(st.st_mtime * 1000)})
        self._commit(manifest)

    def append_batch(self, batch, name_hint: str = "data"):
        from .parquet_io import write_batch_parquet
        p = os.path.join(self.path,
                         f"{name_hint}-{uuid.uuid4().hex[:12]}.parquet")
        write_batch_parquet(batch, p)
        self.append_files([p])

    def remove_files(self, paths: List[str]):
        gone = {os.path.abspath(p) for p in paths}
        manifest = [m for m in self._manifest_now()
                    if m["path"] not in gone]
        self._commit(manifest)

    def files_for_snapshot(self, snapshot_id: Optional[int] = None
                           ) -> List[FileInfo]:
        if snapshot_id is None:
            v = self.current_version()
            snap = self._snapshot(v)
        else:
            snap = next((s for s in self.snapshots()
                         if s["snapshot-id"] == snapshot_id), None)
            if snap is None:
                raise HyperspaceException(
                    f"No snapshot {snapshot_id} in {self.path}")
        return sorted((FileInfo(m["path"], m["size"], m["mtime"])
                       for m in snap["manifest"]),
                      key=lambda f: f.name)


class IcebergTableRelation(FileBasedRelation):
    """Signature = snapshotId + table location (reference
    iceberg/IcebergRelation.scala:63-68)."""

    def __init__(self, path: str, snapshot_id: Optional[int] = None,
                 options: Optional[Dict[str, str]] = None):
        self.table = IcebergTable(path)
        self.snapshot_id_opt = snapshot_id
        self._options = dict(options or {})
        if snapshot_id is not None:
            self._options["snapshot-id"] = str(snapshot_id)
        self._schema: Optional[Schema] = None

    @property
    def root_paths(self):
        return [self.table.path]

    @property
    def file_format(self):
        return "iceberg"

    @property
    def options(self):
        return self._options

    @property
    def effective_snapshot_id(self) -> int:
        return (self.snapshot_id_opt if self.snapshot_id_opt is not None
                else self.table.snapshot_id)

    @property
    def schema(self) -> Schema:
        if self._schema is None:
            files = self.all_files()
            if not files:
                raise HyperspaceException(
                    f"Empty iceberg table: {self.table.path}")
            import pyarrow.parquet as pq
            self._schema = Schema.from_arrow(pq.read_schema(files[0].name))
        return self._schema

    def all_files(self) -> List[FileInfo]:
        return self.table.files_for_snapshot(self.snapshot_id_opt)

    def signature(self) -> str:
        return f"{self.effective_snapshot_id}.{self.table.path}"

    def refreshed(self) -> "IcebergTableRelation":
        return IcebergTableRelation(self.table.path, self.snapshot_id_opt)

    def describe(self) -> str:
        s = (f"@snap{self.snapshot_id_opt}"
             if self.snapshot_id_opt is not None else "")
        return f"iceberg:{self.table.path}{s}"


class IcebergTableSourceProvider(FileBasedSourceProvider):
    def supports(self, relation) -> bool:
        return isinstance(relation, IcebergTableRelation)

    def from_metadata(self, metadata: Relation
                      ) -> Optional[FileBasedRelation]:
        if metadata.fileFormat != "iceberg":
            return None
        snap = metadata.options.get("snapshot-id")
        return IcebergTableRelation(
            metadata.rootPaths[0],
            int(snap) if snap is not None else None)
