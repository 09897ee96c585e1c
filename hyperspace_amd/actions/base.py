"""Action template: the index lifecycle transaction.

``run()`` = validate -> begin (claim log id base+1, transient state) ->
op -> end (log id base+2, final state, refresh latestStable).
Reference: actions/Action.scala:34-108.

A lost race on either log write raises HyperspaceException("Could not
acquire proper state") — the optimistic concurrency protocol
(actions/Action.scala:77-82).
"""

from __future__ import annotations

import time
from abc import ABC, abstractmethod
from typing import Optional

from ..exceptions import HyperspaceException, NoChangesException
from ..log.constants import States
from ..log.entry import IndexLogEntry
from ..log.log_manager import IndexLogManager
from ..telemetry import HyperspaceEvent


def now_ms() -> int:
    return int(time.time() * 1000)


class Action(ABC):
    def __init__(self, log_manager: IndexLogManager):
        self.log_manager = log_manager
        self.base_id: Optional[int] = None

    # data-plane ops (create/refresh/optimize builds) are collective —
    # every rank participates; metadata/file-deletion ops run on rank 0
    # only in distributed mode
    collective_op = True

    # -- abstract pieces ---------------------------------------------------
    @property
    @abstractmethod
    def transient_state(self) -> str: ...

    @property
    @abstractmethod
    def final_state(self) -> str: ...

    def validate(self) -> None:
        pass

    @abstractmethod
    def op(self) -> None: ...

    @abstractmethod
    def log_entry(self) -> IndexLogEntry:
        """Entry to persist at end()."""

    def event(self, message: str) -> Optional[HyperspaceEvent]:
        return None

    # -- template ----------------------------------------------------------
    def run(self) -> None:
        from ..parallel import dist_context as dc
        if dc.is_distributed() and dc.get_world_size() > 1:
            self._run_distributed()
            return
        self._run_local()

    def _run_distributed(self) -> None:
        """Collective run: every rank executes the data-plane op();
        rank 0 alone drives the metadata log transaction."""
        from ..parallel import dist_context as dc
        rank0 = dc.get_rank() == 0
        dc.barrier()
        if rank0:
            self.validate()
            base = self.log_manager.get_latest_id()
            self.base_id = base if base is not None else -1
            self.begin()
        dc.barrier()
        no_changes = False
        if self.collective_op or rank0:
            try:
                self.op()
            except NoChangesException:
                # all ranks see the same source state -> collective signal
                no_changes = True
        if self.collective_op:
            dc.barrier()
        else:
            # non-collective: broadcast the no-op outcome from rank 0
            # (device tensor under NCCL/RCCL — a CPU tensor crashes there)
            import torch.distributed as dist
            t = dc.collective_tensor([1 if no_changes else 0])
            dist.broadcast(t, src=0)
            no_changes = bool(t[0])
        dc.barrier()
        if rank0:
            if no_changes:
                self._write_or_fail(self.base_id + 2, self._current_stable())
                self.log_manager.create_latest_stable_log(self.base_id + 2)
            else:
                self.end()
            ev = self.event("Operation Succeeded." if not no_changes
                            else "Operation needs no update.")
            if ev is not None:
                self._log_event(ev)
        dc.barrier()

    def _run_local(self) -> None:
        self.validate()
        base = self.log_manager.get_latest_id()
        self.base_id = base if base is not None else -1
        try:
            self.begin()
            self.op()
        except NoChangesException:
            # roll the log forward to a stable no-op terminal state
            self._write_or_fail(self.base_id + 2, self._current_stable())
            self.log_manager.create_latest_stable_log(self.base_id + 2)
            ev = self.event("Operation needs no update.")
            if ev is not None:
                self._log_event(ev)
            return
        self.end()
        ev = self.event("Operation Succeeded.")
        if ev is not None:
            self._log_event(ev)

    def begin(self) -> None:
        entry = self.log_entry_for_begin()
        entry.state = self.transient_state
        entry.timestamp = now_ms()
        self._write_or_fail(self.base_id + 1, entry)

    def end(self) -> None:
        entry = self.log_entry()
        entry.state = self.final_state
        entry.timestamp = now_ms()
        self.log_manager.delete_latest_stable_log()
        self._write_or_fail(self.base_id + 2, entry)
        self.log_manager.create_latest_stable_log(self.base_id + 2)

    def log_entry_for_begin(self) -> IndexLogEntry:
        return self.log_entry()

    def _current_stable(self) -> IndexLogEntry:
        entry = self.log_manager.get_latest_stable_log()
        if entry is None:
            raise HyperspaceException("No stable state to return to")
        return entry

    def _write_or_fail(self, log_id: int, entry: IndexLogEntry) -> None:
        if not self.log_manager.write_log(log_id, entry):
            raise HyperspaceException(
                "Could not acquire proper state for log id "
                f"{log_id} (concurrent modification)")

    def _log_event(self, event: HyperspaceEvent) -> None:
        logger = getattr(self, "event_logger", None)
        if logger is not None:
            logger.log_event(event)
