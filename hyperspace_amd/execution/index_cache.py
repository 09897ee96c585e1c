"""Device-resident index data cache.

MI355X has 288 GB of HBM3E per GPU — indexes should live there, not be
re-decoded from Parquet on every query.  This cache keys on the index
log id + the exact file list, so any refresh/optimize/vacuum (which
changes files or log id) naturally invalidates.  Eviction is LRU by
bytes with a configurable budget (default 64 GB per process; the
reference's analog is its 300 s metadata TTL cache —
index/CachingIndexCollectionManager.scala — but data caching is an
MI355X-first design choice).
"""

from __future__ import annotations

import collections
from typing import Dict, List, Optional, Tuple

import torch

from .columnar import ColumnBatch

DEFAULT_BUDGET_BYTES = 64 << 30


class IndexDataCache:
    def __init__(self, budget_bytes: int = DEFAULT_BUDGET_BYTES):
        self.budget = budget_bytes
        self._entries: "collections.OrderedDict[tuple, Tuple[ColumnBatch, Optional[torch.Tensor]]]" = (
            collections.OrderedDict())
        self._bytes = 0
        self.hits = 0
        self.misses = 0

    @staticmethod
    def key(entry, files: List[str], columns: List[str],
            extra: tuple = ()) -> tuple:
        return (entry.name, entry.id, tuple(sorted(files)),
                tuple(c.lower() for c in columns), extra)

    def get(self, key: tuple):
        item = self._entries.get(key)
        if item is None:
            self.misses += 1
            return None
        self._entries.move_to_end(key)
        self.hits += 1
        return item

    def put(self, key: tuple, batch: ColumnBatch,
            seg: Optional[torch.Tensor]) -> None:
        nbytes = batch.nbytes()
        if nbytes > self.budget:
            return
        while self._bytes + nbytes > self.budget and self._entries:
            _, (old, _seg) = self._entries.popitem(last=False)
            self._bytes -= old.nbytes()
        self._entries[key] = (batch, seg)
        self._bytes += nbytes

    def clear(self):
        self._entries.clear()
        self._bytes = 0
