"""Versioned operation log with optimistic concurrency.

Layout (reference: index/IndexLogManager.scala):

    <index path>/_hyperspace_log/<id>          one JSON log entry per id
    <index path>/_hyperspace_log/latestStable  copy of the latest stable entry

Optimistic concurrency: ``write_log(id)`` writes a temp file then *hard-links*
it to the target name.  The link fails if the slot is already taken — the
loser of the race observes the failure and aborts
(index/IndexLogManager.scala:178-194: temp file + atomic rename).
"""

from __future__ import annotations

import json
import os
import tempfile
from typing import Optional

from .constants import States
from .entry import IndexLogEntry
from ..config import IndexConstants


class IndexLogManager:
    """File-system backed log manager (IndexLogManagerImpl)."""

    LATEST_STABLE = "latestStable"

    def __init__(self, index_path: str):
        self.index_path = index_path
        self.log_dir = os.path.join(index_path, IndexConstants.HYPERSPACE_LOG)

    # -- reads ------------------------------------------------------------
    def _log_path(self, log_id: int) -> str:
        return os.path.join(self.log_dir, str(log_id))

    def get_log(self, log_id: int) -> Optional[IndexLogEntry]:
        path = self._log_path(log_id)
        try:
            with open(path, "r") as f:
                return IndexLogEntry.from_json_str(f.read())
        except FileNotFoundError:
            # a concurrent vacuum may trim entries between listing and
            # read — absent is absent, whoever observed it
            return None

    def get_latest_id(self) -> Optional[int]:
        if not os.path.isdir(self.log_dir):
            return None
        ids = [int(n) for n in os.listdir(self.log_dir) if n.isdigit()]
        return max(ids) if ids else None

    def get_latest_log(self) -> Optional[IndexLogEntry]:
        latest = self.get_latest_id()
        return self.get_log(latest) if latest is not None else None

    def get_latest_stable_log(self) -> Optional[IndexLogEntry]:
        """latestStable copy if valid, else scan back for a stable state
        (index/IndexLogManager.scala:102-127)."""
        stable_path = os.path.join(self.log_dir, self.LATEST_STABLE)
        try:
            with open(stable_path, "r") as f:
                entry = IndexLogEntry.from_json_str(f.read())
            if entry.state in States.STABLE_STATES:
                return entry
        except (FileNotFoundError, json.JSONDecodeError, KeyError):
            # FileNotFoundError: a concurrent writer's
            # delete_latest_stable_log between our exists-check and
            # open — fall through to the scan-back, same as a corrupt
            # copy (reference: IndexLogManager.scala:102-127 tolerates
            # an unreadable latestStable the same way)
            pass
        latest = self.get_latest_id()
        if latest is None:
            return None
        for log_id in range(latest, -1, -1):
            try:
                entry = self.get_log(log_id)
            except (json.JSONDecodeError, KeyError, ValueError):
                # a torn/corrupt entry never wins; keep scanning back —
                # earlier stable entries are immutable once written
                continue
            if entry is not None and entry.state in States.STABLE_STATES:
                return entry
        return None

    # -- writes -----------------------------------------------------------
    def write_log(self, log_id: int, entry: IndexLogEntry) -> bool:
        """Atomically claim log slot ``log_id``.  Returns False on lost race."""
        os.makedirs(self.log_dir, exist_ok=True)
        entry.id = log_id
        target = self._log_path(log_id)
        fd, tmp = tempfile.mkstemp(dir=self.log_dir, prefix=".tmp_log_")
        try:
            with os.fdopen(fd, "w") as f:
                f.write(entry.to_json_str())
            try:
                os.link(tmp, target)  # fails if target exists => lost race
            except FileExistsError:
                return False
            return True
        finally:
            os.unlink(tmp)

    def create_latest_stable_log(self, log_id: int) -> bool:
        entry = self.get_log(log_id)
        if entry is None or entry.state not in States.STABLE_STATES:
            return False
        stable_path = os.path.join(self.log_dir, self.LATEST_STABLE)
        fd, tmp = tempfile.mkstemp(dir=self.log_dir, prefix=".tmp_stable_")
        with os.fdopen(fd, "w") as f:
            f.write(entry.to_json_str())
        os.replace(tmp, stable_path)
        return True

    def delete_latest_stable_log(self) -> bool:
        stable_path = os.path.join(self.log_dir, self.LATEST_STABLE)
        try:
            os.unlink(stable_path)
        except FileNotFoundError:
            pass
        return True


class IndexLogManagerFactory:
    """DI seam used by tests (reference: index/factories.scala)."""

    def create(self, index_path: str) -> IndexLogManager:
        return IndexLogManager(index_path)
