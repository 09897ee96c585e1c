"""Streaming source scan for index builds.

Splits the source file list into byte-bounded groups and yields decoded
device batches group by group, prefetching the next group's read +
decode on a background thread while the caller processes the current
one.  This bounds build memory to O(group) + O(output buckets) and
overlaps disk/PCIe with the sort/write pipeline — the out-of-core
streaming path for sources larger than HBM (SURVEY.md §7 hard part b).
"""

from __future__ import annotations

from concurrent.futures import ThreadPoolExecutor
from typing import Iterator, List, Optional, Tuple

import torch

from .columnar import ColumnBatch
from ..config import IndexConstants


class ScanStream:
    def __init__(self, files, columns: List[str], device,
                 lineage_tracker=None, group_bytes: int = 1 << 30,
                 reader=None):
        """``files``: FileInfo list (already sharded for this rank);
        ``reader(paths, columns, device) -> (batch, row_counts)``
        defaults to the parquet path."""
        self.files = list(files)
        self.columns = list(columns)
        self.device = device
        self.tracker = lineage_tracker
        self.group_bytes = max(1, group_bytes)
        self.reader = reader

    @property
    def total_bytes(self) -> int:
        return sum(f.size for f in self.files)

    def file_groups(self) -> List[List]:
        groups: List[List] = []
        cur: List = []
        cur_bytes = 0
        for f in self.files:
            if cur and cur_bytes + f.size > self.group_bytes:
                groups.append(cur)
                cur = []
                cur_bytes = 0
            cur.append(f)
            cur_bytes += f.size
        if cur:
            groups.append(cur)
        return groups

    def _load_group(self, group) -> ColumnBatch:
        paths = [f.name for f in group]
        if self.reader is not None:
            batch, row_counts = self.reader(paths, self.columns or None,
                                            self.device)
        else:
            from ..sources.parquet_io import (read_files_batch,
                                              read_files_batch_device)
            if self.device.type == "cuda":
                batch, row_counts = read_files_batch_device(
                    paths, self.device, columns=self.columns or None)
            else:
                batch, row_counts = read_files_batch(
                    paths, columns=self.columns or None)
        if self.tracker is not None:
            # materialize the lineage column ON DEVICE (two tiny H2D
            # uploads + repeat_interleave) — a host torch.full/cat of
            # the full column costs GBs of pageable traffic
            fids = torch.tensor(
                [self.tracker.add_file(f.name, f.size, f.modifiedTime)
                 for f in group], dtype=torch.int64, device=batch.device)
            counts = torch.tensor(row_counts, dtype=torch.int64,
                                  device=batch.device)
            batch = batch.with_column(
                IndexConstants.DATA_FILE_NAME_ID_COLUMN,
                torch.repeat_interleave(fids, counts))
        return batch

    def batches(self) -> Iterator[ColumnBatch]:
        """Yield one decoded batch per file group, prefetch depth 1."""
        groups = self.file_groups()
        if not groups:
            return
        with ThreadPoolExecutor(max_workers=1) as prefetcher:
            fut = prefetcher.submit(self._load_group, groups[0])
            for i in range(len(groups)):
                batch = fut.result()
                if i + 1 < len(groups):
                    fut = prefetcher.submit(self._load_group,
                                            groups[i + 1])
                yield batch

    def materialize(self) -> ColumnBatch:
        parts = list(self.batches())
        if not parts:
            from ..execution.executor import _empty_batch
            from ..log.entry import Schema, SchemaField
            return ColumnBatch({})
        if len(parts) == 1:
            return parts[0]
        return ColumnBatch.concat(parts)
