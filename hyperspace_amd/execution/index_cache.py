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
import threading
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
        # concurrent queries share the session cache: all structural
        # mutation happens under this lock (an unlocked pop/popitem
        # interleaving corrupts the byte accounting)
        self._lock = threading.Lock()

    @staticmethod
    def key(entry, files: List[str], extra: tuple = ()) -> tuple:
        # columns are NOT part of the key: one stored batch serves every
        # query whose columns are a subset (the build write-through
        # stores the full slice; filter/join queries select from it)
        return (entry.name, entry.id, tuple(sorted(files)), extra)

    def get(self, key: tuple, columns: Optional[List[str]] = None):
        with self._lock:
            item = self._entries.get(key)
            if item is not None and columns is not None:
                have = {c.lower() for c in item[0].columns}
                if not all(c.lower() in have for c in columns):
                    item = None  # stored batch lacks a needed column
            if item is None:
                self.misses += 1
                return None
            self._entries.move_to_end(key)
            self.hits += 1
            batch, seg = item
        if columns is not None:
            batch = batch.select(columns)
        return batch, seg

    def put(self, key: tuple, batch: ColumnBatch,
            seg: Optional[torch.Tensor]) -> None:
        nbytes = batch.nbytes()
        if nbytes > self.budget:
            # oversized: keep whatever incumbent we have under this key
            return
        with self._lock:
            self._put_locked(key, batch, seg, nbytes)

    def _put_locked(self, key, batch, seg, nbytes):
        old = self._entries.get(key)
        if old is not None:
            # keep the incumbent if it already serves every column of the
            # new batch (e.g. the full-slice build write-through) — a
            # narrow-column query's put must not evict the wide batch
            have = {c.lower() for c in old[0].columns}
            if all(c.lower() in have for c in batch.columns):
                self._entries.move_to_end(key)
                return
            self._entries.pop(key)
            self._bytes -= old[0].nbytes()
        while self._bytes + nbytes > self.budget and self._entries:
            _, (evicted, _seg) = self._entries.popitem(last=False)
            self._bytes -= evicted.nbytes()
        self._entries[key] = (batch, seg)
        self._bytes += nbytes

    def clear(self):
        with self._lock:
            self._entries.clear()
            self._bytes = 0
