"""Index collection management.

Reference: index/IndexCollectionManager.scala (per-index action dispatch)
and index/CachingIndexCollectionManager.scala (TTL metadata cache,
invalidated on every mutating API).
"""

from __future__ import annotations

import os
import time
from typing import List, Optional

from .actions.actions import (CancelAction, CreateAction, DeleteAction,
                              OptimizeAction, RefreshFullAction,
                              RefreshIncrementalAction, RefreshQuickAction,
                              RestoreAction, VacuumAction,
                              VacuumOutdatedAction)
from .exceptions import HyperspaceException
from .log.constants import States
from .log.data_manager import IndexDataManager
from .log.entry import IndexLogEntry
from .log.log_manager import IndexLogManager
from .log.path_resolver import PathResolver


class IndexCollectionManager:
    def __init__(self, session, log_manager_factory=None,
                 data_manager_factory=None):
        """``*_factory``: DI seams (reference index/factories.scala) —
        action/unit tests inject mocked managers through them; the
        defaults build the real filesystem-backed managers."""
        from .log import (IndexDataManagerFactory, IndexLogManagerFactory)
        self.session = session
        self.path_resolver = PathResolver(session.conf)
        self.log_manager_factory = (log_manager_factory
                                    or IndexLogManagerFactory())
        self.data_manager_factory = (data_manager_factory
                                     or IndexDataManagerFactory())

    # -- helpers -----------------------------------------------------------
    def _managers(self, name: str):
        path = self.path_resolver.get_index_path(name)
        return (path, self.log_manager_factory.create(path),
                self.data_manager_factory.create(path))

    def log_manager(self, name: str) -> IndexLogManager:
        """Log manager for one index (closestIndex time travel reads
        retained historical entries through it)."""
        return self._managers(name)[1]

    # -- mutations ---------------------------------------------------------
    def create(self, df, config) -> None:
        path, log_mgr, data_mgr = self._managers(config.index_name)
        os.makedirs(path, exist_ok=True)
        action = CreateAction(self.session, df, config, log_mgr, data_mgr)
        action.run()
        # build write-through: the just-sorted index batch is already in
        # HBM in index-scan layout; key it on the committed entry so the
        # first query serves from residency instead of re-reading files
        built = getattr(action, "built_for_cache", None)
        cache = self.session.index_data_cache()
        if built is not None and cache is not None:
            # key on the entry THIS action committed (log id base+2), not
            # latestStable — a concurrent action landing between end() and
            # here would otherwise key the batch under a mismatched entry
            entry = (log_mgr.get_log(action.base_id + 2)
                     if action.base_id is not None else None)
            if entry is not None and entry.state == States.ACTIVE:
                batch, seg, files = built
                cache.put(cache.key(entry, files), batch, seg)

    def delete(self, name: str) -> None:
        _, log_mgr, _ = self._managers(name)
        DeleteAction(self.session, log_mgr).run()

    def restore(self, name: str) -> None:
        _, log_mgr, _ = self._managers(name)
        RestoreAction(self.session, log_mgr).run()

    def vacuum(self, name: str) -> None:
        path, log_mgr, _ = self._managers(name)
        latest = log_mgr.get_latest_stable_log()
        if latest is not None and latest.state == States.ACTIVE:
            VacuumOutdatedAction(self.session, log_mgr,
                                 self.data_manager_factory
                                 .create(path)).run()
        else:
            VacuumAction(self.session, log_mgr, path).run()

    def refresh(self, name: str, mode: str = "full") -> None:
        _, log_mgr, data_mgr = self._managers(name)
        if mode == "full":
            RefreshFullAction(self.session, log_mgr, data_mgr).run()
        elif mode == "incremental":
            RefreshIncrementalAction(self.session, log_mgr, data_mgr).run()
        elif mode == "quick":
            RefreshQuickAction(self.session, log_mgr, data_mgr).run()
        else:
            raise HyperspaceException(f"Unsupported refresh mode: {mode}")

    def optimize(self, name: str, mode: str = "quick") -> None:
        _, log_mgr, data_mgr = self._managers(name)
        OptimizeAction(self.session, log_mgr, data_mgr, mode).run()

    def cancel(self, name: str) -> None:
        _, log_mgr, _ = self._managers(name)
        CancelAction(self.session, log_mgr).run()

    # -- reads -------------------------------------------------------------
    def get_index(self, name: str) -> Optional[IndexLogEntry]:
        _, log_mgr, _ = self._managers(name)
        return log_mgr.get_latest_stable_log()

    def get_indexes(self, states: Optional[List[str]] = None
                    ) -> List[IndexLogEntry]:
        root = self.path_resolver.system_path()
        out: List[IndexLogEntry] = []
        if not os.path.isdir(root):
            return out
        for name in sorted(os.listdir(root)):
            entry = IndexLogManager(os.path.join(root, name)) \
                .get_latest_stable_log()
            if entry is None:
                continue
            if states is None or entry.state in states:
                out.append(entry)
        return out

    def index_statistics(self, name: str, extended: bool = False):
        from .index_statistics import IndexStatistics
        entry = self.get_index(name)
        if entry is None:
            raise HyperspaceException(f"Index not found: {name}")
        return IndexStatistics(entry, extended).to_dict()


class CachingIndexCollectionManager(IndexCollectionManager):
    """TTL cache over get_indexes (default 300 s), cleared on mutation.

    Reference: index/CachingIndexCollectionManager.scala:38-108.
    """

    def __init__(self, session):
        super().__init__(session)
        self._cache: Optional[List[IndexLogEntry]] = None
        self._cache_time = 0.0

    def clear_cache(self):
        self._cache = None

    def get_indexes(self, states=None):
        ttl = self.session.conf.cache_expiry_seconds
        now = time.time()
        if self._cache is None or (now - self._cache_time) > ttl:
            self._cache = super().get_indexes(None)
            self._cache_time = now
        if states is None:
            return list(self._cache)
        return [e for e in self._cache if e.state in states]

    def create(self, df, config):
        self.clear_cache()
        super().create(df, config)
        self.clear_cache()

    def delete(self, name):
        self.clear_cache()
        super().delete(name)
        self.clear_cache()

    def restore(self, name):
        self.clear_cache()
        super().restore(name)
        self.clear_cache()

    def vacuum(self, name):
        self.clear_cache()
        super().vacuum(name)
        self.clear_cache()

    def refresh(self, name, mode="full"):
        self.clear_cache()
        super().refresh(name, mode)
        self.clear_cache()

    def optimize(self, name, mode="quick"):
        self.clear_cache()
        super().optimize(name, mode)
        self.clear_cache()

    def cancel(self, name):
        self.clear_cache()
        super().cancel(name)
        self.clear_cache()
