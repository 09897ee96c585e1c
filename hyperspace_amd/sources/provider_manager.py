"""Source provider registry.

Reference: index/sources/FileBasedSourceProviderManager.scala:38-174 —
loads builder classes from conf; ensures exactly one provider claims a
relation.
"""

from __future__ import annotations

from typing import List, Optional

from .interfaces import FileBasedRelation, FileBasedSourceProvider
from .parquet_source import ParquetSourceProvider
from ..exceptions import HyperspaceException
from ..log.entry import Relation


class FileBasedSourceProviderManager:
    def __init__(self, providers: Optional[List[FileBasedSourceProvider]]
                 = None):
        self.providers: List[FileBasedSourceProvider] = providers or [
            ParquetSourceProvider()]
        # table-format providers (delta-style log, iceberg-style snapshots)
        from .delta_source import DeltaTableSourceProvider
        from .iceberg_source import IcebergTableSourceProvider
        from .text_source import TextFormatSourceProvider
        self.providers.append(DeltaTableSourceProvider())
        self.providers.append(IcebergTableSourceProvider())
        self.providers.append(TextFormatSourceProvider())

    def register(self, provider: FileBasedSourceProvider) -> None:
        self.providers.append(provider)

    def is_supported(self, relation: FileBasedRelation) -> bool:
        return sum(1 for p in self.providers if p.supports(relation)) == 1

    def from_metadata(self, metadata: Relation) -> FileBasedRelation:
        matches = [r for r in
                   (p.from_metadata(metadata) for p in self.providers)
                   if r is not None]
        if len(matches) != 1:
            raise HyperspaceException(
                f"{len(matches)} providers claim relation "
                f"{metadata.fileFormat}: expected exactly 1")
        return matches[0]
