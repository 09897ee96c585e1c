"""Engine session: device, conf, providers, optimizer hook.

The analog of the reference's SparkSession + HyperspaceContext
(Hyperspace.scala:196-223, package.scala enableHyperspace).
"""

from __future__ import annotations

import threading
from typing import Optional

import torch

from .config import Conf, IndexConstants
from .sources.provider_manager import FileBasedSourceProviderManager


class HyperspaceSession:
    def __init__(self, conf: Optional[Conf] = None,
                 device: Optional[str] = None):
        self.conf = conf or Conf()
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.provider_manager = FileBasedSourceProviderManager()
        self._hyperspace_enabled = False
        self._event_logger = None
        # thread-local maintenance kill-switch
        # (ApplyHyperspace.withHyperspaceRuleDisabled,
        #  rules/ApplyHyperspace.scala:68-75)
        self._local = threading.local()

    # -- enable/disable (reference: package.scala:31-94) -----------------
    def enable_hyperspace(self) -> "HyperspaceSession":
        self._hyperspace_enabled = True
        return self

    def disable_hyperspace(self) -> "HyperspaceSession":
        self._hyperspace_enabled = False
        return self

    def is_hyperspace_enabled(self) -> bool:
        return (self._hyperspace_enabled
                and self.conf.apply_enabled
                and not getattr(self._local, "rule_disabled", False))

    class _RuleDisabled:
        def __init__(self, session):
            self.session = session

        def __enter__(self):
            self.session._local.rule_disabled = True

        def __exit__(self, *a):
            self.session._local.rule_disabled = False

    def with_rule_disabled(self):
        return self._RuleDisabled(self)

    # -- data access -----------------------------------------------------
    def read_parquet(self, *paths: str):
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.parquet_source import ParquetRelation
        return DataFrame(self, Scan(ParquetRelation(list(paths))))

    def read_delta(self, path: str, version_as_of=None,
                   timestamp_as_of=None):
        """Read a Delta Lake table (time travel via ``version_as_of``
        or ``timestamp_as_of`` — the newest version committed at or
        before the timestamp, Delta's timestampAsOf semantics)."""
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.delta_source import DeltaTable, DeltaTableRelation
        if timestamp_as_of is not None:
            if version_as_of is not None:
                from .exceptions import HyperspaceException
                raise HyperspaceException(
                    "Specify only one of version_as_of/timestamp_as_of")
            version_as_of = DeltaTable(path).version_at_timestamp(
                timestamp_as_of)
        return DataFrame(self, Scan(DeltaTableRelation(path,
                                                       version_as_of)))

    def read_csv(self, *paths: str):
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.text_source import TextFormatRelation
        return DataFrame(self, Scan(TextFormatRelation("csv", list(paths))))

    def read_json(self, *paths: str):
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.text_source import TextFormatRelation
        return DataFrame(self, Scan(TextFormatRelation("json",
                                                       list(paths))))

    def read_text(self, *paths: str):
        """Line-oriented text files: one string column 'value' (the
        reference default source's text format)."""
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.text_source import TextFormatRelation
        return DataFrame(self, Scan(TextFormatRelation("text",
                                                       list(paths))))

    def read_avro(self, *paths: str):
        """Avro container files (decoded by the built-in reader,
        sources/avro_io.py)."""
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.text_source import TextFormatRelation
        return DataFrame(self, Scan(TextFormatRelation("avro",
                                                       list(paths))))

    def read_orc(self, *paths: str):
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.text_source import TextFormatRelation
        return DataFrame(self, Scan(TextFormatRelation("orc", list(paths))))

    def read_iceberg(self, path: str, snapshot_id=None):
        """Read an iceberg-style snapshot table (time travel via
        ``snapshot_id``)."""
        from .dataframe import DataFrame
        from .plan.nodes import Scan
        from .sources.iceberg_source import IcebergTableRelation
        return DataFrame(self, Scan(IcebergTableRelation(path,
                                                         snapshot_id)))

    def index_manager(self):
        from .index_management import CachingIndexCollectionManager
        if not hasattr(self, "_index_manager"):
            self._index_manager = CachingIndexCollectionManager(self)
        return self._index_manager

    def index_data_cache(self):
        """Device-resident index data cache (None when disabled)."""
        from .config import IndexConstants
        if not self.conf.get(
                "spark.hyperspace.index.dataCache.enabled", True):
            return None
        if not hasattr(self, "_index_data_cache"):
            from .execution.index_cache import IndexDataCache
            budget = int(self.conf.get(
                "spark.hyperspace.index.dataCache.budgetBytes", 64 << 30))
            self._index_data_cache = IndexDataCache(budget)
        return self._index_data_cache

    @property
    def event_logger(self):
        if self._event_logger is None:
            from .telemetry import default_event_logger
            self._event_logger = default_event_logger(self.conf)
        return self._event_logger

    @event_logger.setter
    def event_logger(self, logger):
        self._event_logger = logger


_default_session: Optional[HyperspaceSession] = None


def get_session() -> HyperspaceSession:
    global _default_session
    if _default_session is None:
        _default_session = HyperspaceSession()
    return _default_session


def set_session(session: HyperspaceSession) -> None:
    global _default_session
    _default_session = session
