"""Pluggable source-provider interfaces.

Reference: index/sources/interfaces.scala:43-277 (FileBasedRelation /
FileBasedSourceProvider / FileBasedRelationMetadata).
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Dict, List, Optional

from ..log.entry import FileInfo, Relation, Schema


class FileBasedRelation(ABC):
    """A file-based data source (immutable file set + schema)."""

    @property
    @abstractmethod
    def root_paths(self) -> List[str]: ...

    @property
    @abstractmethod
    def schema(self) -> Schema: ...

    @property
    @abstractmethod
    def file_format(self) -> str: ...

    @property
    def options(self) -> Dict[str, str]:
        return {}

    @abstractmethod
    def all_files(self) -> List[FileInfo]:
        """Leaf data files as (abs path, size, mtime, UNKNOWN id)."""

    @abstractmethod
    def signature(self) -> str:
        """Content fingerprint of the relation."""

    def read_files(self, paths: List[str], columns, device):
        """Read data files into a (ColumnBatch on ``device``, per-file row
        counts).  Default: the native/pyarrow Parquet path with device
        decode; text formats override with their host readers."""
        from .parquet_io import read_files_batch, read_files_batch_device
        if getattr(device, "type", "cpu") == "cuda":
            return read_files_batch_device(paths, device, columns)
        return read_files_batch(paths, columns)

    def file_subset_relation(self, paths: List[str]
                             ) -> "FileBasedRelation":
        """Relation over a SUBSET of this relation's data files — the
        Hybrid Scan appended-file scan (reference reads appended files
        via the provider's internalFileFormatName; for Delta/Iceberg
        the underlying files are plain parquet).  Sources with
        partition metadata override to keep partition-column
        materialization working on the subset."""
        from .parquet_source import ParquetRelation
        return ParquetRelation(paths, dict(self.options))

    def describe(self) -> str:
        return f"{self.file_format}:{','.join(self.root_paths)}"

    def create_relation_metadata(self, file_id_tracker) -> Relation:
        """Build the log-entry Relation (with tracked file ids)."""
        from ..log.entry import Content, Hdfs
        files = []
        for f in self.all_files():
            fid = file_id_tracker.add_file(f.name, f.size, f.modifiedTime)
            files.append((f.name, f.size, f.modifiedTime, fid))
        return Relation(
            rootPaths=list(self.root_paths),
            data=Hdfs(Content.from_leaf_files(sorted(files))),
            dataSchema=self.schema,
            fileFormat=self.file_format,
            options=dict(self.options))

    def refreshed(self) -> "FileBasedRelation":
        """Re-list the current state of the source (for refresh actions)."""
        return self

    def freeze(self) -> "FileBasedRelation":
        """Pin the file listing at this instant: until :meth:`unfreeze`,
        ``all_files()`` — and everything derived from it (signature,
        relation metadata, partition schema) — returns one consistent
        snapshot.  Lifecycle actions freeze their source so a concurrent
        append between the diff and the commit cannot produce a log
        entry whose signature covers files the index does not.  (Spark
        gets this for free: the InMemoryFileIndex inside the plan is
        listed once, reference index/sources/interfaces.scala:43-120
        derives everything from that plan snapshot.)"""
        snapshot = self.all_files()
        self.all_files = (  # type: ignore[method-assign]
            lambda: snapshot)
        return self

    def unfreeze(self) -> "FileBasedRelation":
        """Restore dynamic listing (drop the :meth:`freeze` snapshot)."""
        self.__dict__.pop("all_files", None)
        return self


class FileBasedSourceProvider(ABC):
    """Claims relations it supports and reconstructs them from metadata."""

    @abstractmethod
    def supports(self, relation: FileBasedRelation) -> bool: ...

    @abstractmethod
    def from_metadata(self, metadata: Relation) -> Optional[FileBasedRelation]:
        """Rebuild a relation from a logged Relation (refresh path)."""

    def enrich_index_properties(self, properties: Dict[str, str]
                                ) -> Dict[str, str]:
        return properties
