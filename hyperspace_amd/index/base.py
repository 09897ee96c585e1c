"""Index trait + config trait + indexer context.

Reference: index/Index.scala:32-168, index/IndexConfigTrait.scala:31-59,
index/IndexerContext.scala:25-43.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Dict, List, Optional, Tuple


class IndexerContext:
    """Handed to index implementations during build/refresh/optimize.

    Holds the engine session (device, conf), the FileIdTracker for lineage,
    and the destination data path for the version being written.
    """

    def __init__(self, session, file_id_tracker, index_data_path: str):
        self.session = session
        self.file_id_tracker = file_id_tracker
        self.index_data_path = index_data_path


class Index(ABC):
    """Contract of a derived dataset (reference Index trait)."""

    @property
    @abstractmethod
    def kind(self) -> str: ...

    @property
    @abstractmethod
    def kind_abbr(self) -> str: ...

    @abstractmethod
    def indexed_columns_list(self) -> List[str]: ...

    @abstractmethod
    def referenced_columns(self) -> List[str]: ...

    @abstractmethod
    def to_json(self) -> Dict[str, Any]: ...

    @property
    def properties(self) -> Dict[str, str]:
        return getattr(self, "_properties", {})

    @abstractmethod
    def with_new_properties(self, props: Dict[str, str]) -> "Index": ...

    @abstractmethod
    def write(self, ctx: IndexerContext, index_data) -> List[str]:
        """Write index data files; returns written file paths."""

    def optimize(self, ctx: IndexerContext, files_to_optimize: List[str]
                 ) -> List[str]:
        raise NotImplementedError

    def refresh_incremental(self, ctx: IndexerContext, appended_df,
                            deleted_file_ids: List[int],
                            previous_files: List[str]):
        raise NotImplementedError

    def refresh_full(self, ctx: IndexerContext, df):
        raise NotImplementedError

    @property
    def can_handle_deleted_files(self) -> bool:
        return False

    def statistics(self) -> Dict[str, Any]:
        return {}


class IndexConfigTrait(ABC):
    """User-facing index configuration."""

    @property
    @abstractmethod
    def index_name(self) -> str: ...

    @abstractmethod
    def referenced_columns(self) -> List[str]: ...

    @abstractmethod
    def create_index(self, ctx: IndexerContext, df,
                     properties: Dict[str, str]) -> Tuple[Index, Any]:
        """Returns (Index instance, index data ColumnBatch/DataFrame)."""

    def placeholder_index(self, relation, conf) -> Index:
        """Kind-appropriate metadata-only Index for the begin() log entry
        (content is unknown until op() completes)."""
        raise NotImplementedError
