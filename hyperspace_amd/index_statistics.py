"""User-visible index summary (reference: index/IndexStatistics.scala:40-164)."""

from __future__ import annotations

from typing import Any, Dict

from .log.entry import IndexLogEntry


class IndexStatistics:
    def __init__(self, entry: IndexLogEntry, extended: bool = False):
        self.entry = entry
        self.extended = extended

    def to_dict(self) -> Dict[str, Any]:
        e = self.entry
        index_files = e.content.file_infos()
        d: Dict[str, Any] = {
            "name": e.name,
            "indexedColumns": e.derivedDataset.indexed_columns_list(),
            "includedColumns": getattr(e.derivedDataset, "included_columns",
                                       []),
            "numBuckets": getattr(e.derivedDataset, "num_buckets", None),
            "schema": e.derivedDataset.to_json().get("schema"),
            "indexLocation": _common_dir(index_files),
            "state": e.state,
        }
        if self.extended:
            d.update({
                "numIndexFiles": len(index_files),
                "sizeOfIndexInBytes": e.index_files_size(),
                "numSourceFiles": len(e.source_file_infos()),
                "sizeOfSourceInBytes": e.source_files_size(),
                "appendedFilesCount": len(e.appended_files()),
                "deletedFilesCount": len(e.deleted_files()),
            })
        return d


def _common_dir(file_infos) -> str:
    import os
    dirs = {os.path.dirname(f.name) for f in file_infos}
    if not dirs:
        return ""
    return os.path.commonpath(list(dirs)) if len(dirs) > 1 else dirs.pop()
