"""The user facade.

API mirrors the reference's Hyperspace class
(Hyperspace.scala:27-193): createIndex / deleteIndex / restoreIndex /
vacuumIndex / refreshIndex / optimizeIndex / cancel / explain / whyNot /
index / indexes.  Python naming is snake_case with camelCase aliases for
drop-in familiarity.
"""

from __future__ import annotations

from typing import List, Optional

from .dataframe import DataFrame
from .log.constants import States
from .session import HyperspaceSession, get_session


class Hyperspace:
    def __init__(self, session: Optional[HyperspaceSession] = None):
        self.session = session or get_session()

    @property
    def _manager(self):
        return self.session.index_manager()

    # -- index CRUD --------------------------------------------------------
    def create_index(self, df: DataFrame, index_config) -> None:
        self._manager.create(df, index_config)

    def delete_index(self, index_name: str) -> None:
        self._manager.delete(index_name)

    def restore_index(self, index_name: str) -> None:
        self._manager.restore(index_name)

    def vacuum_index(self, index_name: str) -> None:
        self._manager.vacuum(index_name)

    def refresh_index(self, index_name: str, mode: str = "full") -> None:
        self._manager.refresh(index_name, mode)

    def optimize_index(self, index_name: str, mode: str = "quick") -> None:
        self._manager.optimize(index_name, mode)

    def cancel(self, index_name: str) -> None:
        self._manager.cancel(index_name)

    # -- introspection -----------------------------------------------------
    def indexes(self) -> List[dict]:
        """Summary rows for all indexes (reference: hs.indexes)."""
        return [self._manager.index_statistics(e.name)
                for e in self._manager.get_indexes()]

    def index(self, index_name: str) -> dict:
        return self._manager.index_statistics(index_name, extended=True)

    def explain(self, df: DataFrame, verbose: bool = False) -> str:
        from .plananalysis.plan_analyzer import PlanAnalyzer
        return PlanAnalyzer(self.session).explain_string(df, verbose)

    def why_not(self, df: DataFrame, index_name: str = "",
                extended: bool = False) -> str:
        from .plananalysis.candidate_analyzer import CandidateIndexAnalyzer
        return CandidateIndexAnalyzer(self.session).why_not_string(
            df, index_name, extended)

    # -- camelCase aliases (reference/py4j naming) --------------------------
    createIndex = create_index
    deleteIndex = delete_index
    restoreIndex = restore_index
    vacuumIndex = vacuum_index
    refreshIndex = refresh_index
    optimizeIndex = optimize_index
    whyNot = why_not
