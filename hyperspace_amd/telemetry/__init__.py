"""Telemetry: pluggable event logger + event taxonomy.

Reference: telemetry/HyperspaceEvent.scala:33-166 (event classes),
telemetry/HyperspaceEventLogging.scala:30-68 (pluggable logger, default
no-op).
"""

from __future__ import annotations

import importlib
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class HyperspaceEvent:
    app_info: Dict[str, str] = field(default_factory=dict)
    message: str = ""
    timestamp: float = field(default_factory=time.time)

    @property
    def name(self) -> str:
        return type(self).__name__


@dataclass
class HyperspaceIndexCRUDEvent(HyperspaceEvent):
    index_name: str = ""


class CreateActionEvent(HyperspaceIndexCRUDEvent):
    pass


class DeleteActionEvent(HyperspaceIndexCRUDEvent):
    pass


class RestoreActionEvent(HyperspaceIndexCRUDEvent):
    pass


class VacuumActionEvent(HyperspaceIndexCRUDEvent):
    pass


class VacuumOutdatedActionEvent(HyperspaceIndexCRUDEvent):
    pass


class RefreshActionEvent(HyperspaceIndexCRUDEvent):
    pass


class RefreshIncrementalActionEvent(HyperspaceIndexCRUDEvent):
    pass


class RefreshQuickActionEvent(HyperspaceIndexCRUDEvent):
    pass


class OptimizeActionEvent(HyperspaceIndexCRUDEvent):
    pass


class CancelActionEvent(HyperspaceIndexCRUDEvent):
    pass


@dataclass
class HyperspaceIndexUsageEvent(HyperspaceEvent):
    """Emitted when the join rule applies indexes to a plan
    (reference: index/covering/JoinIndexRule.scala:678-684)."""
    index_names: List[str] = field(default_factory=list)


class EventLogger:
    def log_event(self, event: HyperspaceEvent) -> None:  # pragma: no cover
        pass


class NoOpEventLogger(EventLogger):
    pass


class JsonlEventLogger(EventLogger):
    """Structured event sink: one JSON object per line, appended to the
    file named by env HYPERSPACE_EVENT_LOG (default
    ~/.hyperspace/events.jsonl).  Loadable via conf
    spark.hyperspace.eventLoggerClass =
    "hyperspace_amd.telemetry.JsonlEventLogger"."""

    def __init__(self, path: str = ""):
        import os
        self.path = path or os.environ.get(
            "HYPERSPACE_EVENT_LOG",
            os.path.join(os.path.expanduser("~"), ".hyperspace",
                         "events.jsonl"))

    def log_event(self, event: HyperspaceEvent) -> None:
        import json
        import os
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        record = {"event": event.name, "timestamp": event.timestamp,
                  "message": event.message}
        for attr in ("index_name", "index_names"):
            if getattr(event, attr, None):
                record[attr] = getattr(event, attr)
        with open(self.path, "a") as f:
            f.write(json.dumps(record) + "\n")


class RecordingEventLogger(EventLogger):
    """Test sink (reference: MockEventLogger, TestUtils.scala:93-110)."""

    def __init__(self):
        self.events: List[HyperspaceEvent] = []

    def log_event(self, event: HyperspaceEvent) -> None:
        self.events.append(event)

    def reset(self):
        self.events = []


def default_event_logger(conf) -> EventLogger:
    from ..config import IndexConstants
    cls_name = conf.get(IndexConstants.EVENT_LOGGER_CLASS)
    if not cls_name:
        return NoOpEventLogger()
    module, _, cls = cls_name.rpartition(".")
    return getattr(importlib.import_module(module), cls)()
