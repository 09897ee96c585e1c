"""Versioned index data directory manager.

Index data for log version ``n`` lives under ``<index path>/v__=<n>/``
(reference: index/IndexDataManager.scala:50-108,
INDEX_VERSION_DIRECTORY_PREFIX index/IndexConstants.scala:92).
"""

from __future__ import annotations

import os
import re
import shutil
from typing import List, Optional

from ..config import IndexConstants

_VERSION_RE = re.compile(
    re.escape(IndexConstants.INDEX_VERSION_DIRECTORY_PREFIX) + r"=(\d+)$")


class IndexDataManager:
    def __init__(self, index_path: str):
        self.index_path = index_path

    def _version_of(self, name: str) -> Optional[int]:
        m = _VERSION_RE.match(name)
        return int(m.group(1)) if m else None

    def get_all_versions(self) -> List[int]:
        if not os.path.isdir(self.index_path):
            return []
        out = []
        for name in os.listdir(self.index_path):
            v = self._version_of(name)
            if v is not None:
                out.append(v)
        return sorted(out)

    def get_latest_version_id(self) -> Optional[int]:
        versions = self.get_all_versions()
        return versions[-1] if versions else None

    def get_path(self, version: int) -> str:
        return os.path.join(
            self.index_path,
            f"{IndexConstants.INDEX_VERSION_DIRECTORY_PREFIX}={version}")

    def delete_version(self, version: int) -> None:
        path = self.get_path(version)
        if os.path.isdir(path):
            shutil.rmtree(path)


class IndexDataManagerFactory:
    def create(self, index_path: str) -> IndexDataManager:
        return IndexDataManager(index_path)
