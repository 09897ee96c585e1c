"""Index path resolution.

Root = conf ``spark.hyperspace.system.path`` (reference default:
``<warehouse>/indexes`` — here: ``$HYPERSPACE_SYSTEM_PATH`` or
``~/.hyperspace/indexes``).  Index-name resolution is case-insensitive
(reference: index/PathResolver.scala:31-69).
"""

from __future__ import annotations

import os

from ..config import Conf


class PathResolver:
    def __init__(self, conf: Conf):
        self.conf = conf

    def system_path(self) -> str:
        p = self.conf.system_path
        if p:
            return p
        env = os.environ.get("HYPERSPACE_SYSTEM_PATH")
        if env:
            return env
        return os.path.join(os.path.expanduser("~"), ".hyperspace", "indexes")

    def get_index_path(self, name: str) -> str:
        """Return the path for index ``name``; reuses an existing directory
        that matches case-insensitively."""
        root = self.system_path()
        if os.path.isdir(root):
            for existing in os.listdir(root):
                if existing.lower() == name.lower():
                    return os.path.join(root, existing)
        return os.path.join(root, name)
