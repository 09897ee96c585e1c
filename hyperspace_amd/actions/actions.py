"""Concrete lifecycle actions.

Reference: actions/ — CreateAction, RefreshAction (full),
RefreshIncrementalAction, RefreshQuickAction, OptimizeAction, DeleteAction,
RestoreAction, VacuumAction, VacuumOutdatedAction, CancelAction.
"""

from __future__ import annotations

import os
import shutil
from typing import Dict, List, Optional, Set, Tuple

from .base import Action, now_ms
from ..config import IndexConstants
from ..exceptions import HyperspaceException, NoChangesException
from ..log.constants import States
from ..log.entry import (Content, FileIdTracker, FileInfo, IndexLogEntry,
                         LogicalPlanFingerprint, Signature, Source,
                         SourcePlan)
from ..log.data_manager import IndexDataManager
from ..log.log_manager import IndexLogManager
from ..sources.parquet_io import bucket_id_of_file
from ..telemetry import (CancelActionEvent, CreateActionEvent,
                         DeleteActionEvent, OptimizeActionEvent,
                         RefreshActionEvent, RefreshIncrementalActionEvent,
                         RefreshQuickActionEvent, RestoreActionEvent,
                         VacuumActionEvent, VacuumOutdatedActionEvent)

SIGNATURE_PROVIDER = "hyperspace_amd.FileBasedSignatureProvider"


def _content_from_paths(paths: List[str], start_id: int = 0) -> Content:
    files = []
    fid = start_id
    for p in sorted(paths):
        st = os.stat(p)
        files.append((os.path.abspath(p), st.st_size,
                      int(st.st_mtime * 1000), fid))
        fid += 1
    return Content.from_leaf_files(files)


def _fingerprint(relation) -> LogicalPlanFingerprint:
    return LogicalPlanFingerprint(
        [Signature(SIGNATURE_PROVIDER, relation.signature())])


def _all_written_files(data_path: str, local_written: List[str]
                       ) -> List[str]:
    """In a distributed build every rank wrote its own bucket files;
    after a barrier, enumerate the whole version directory so the log
    entry covers all ranks' output."""
    from ..parallel import dist_context as dc
    if not (dc.is_distributed() and dc.get_world_size() > 1):
        return local_written
    dc.barrier()
    return sorted(
        os.path.join(data_path, f) for f in os.listdir(data_path)
        if f.endswith(".parquet"))


def _tracker_from_entry(entry: IndexLogEntry) -> FileIdTracker:
    tracker = FileIdTracker()
    tracker.add_file_info(entry.source_file_infos())
    return tracker


def _agree_data_version(data_manager) -> int:
    """Next index data version, agreed across ranks BEFORE any rank
    creates the directory (rank 0 lists and broadcasts — a faster rank's
    makedirs must not race a slower rank's listing)."""
    latest = data_manager.get_latest_version_id()
    version = 0 if latest is None else latest + 1
    from ..parallel import dist_context as dc
    if dc.is_distributed() and dc.get_world_size() > 1:
        import torch.distributed as dist
        t = dc.collective_tensor([version])
        dist.broadcast(t, src=0)
        version = int(t[0])
    return version


def _rebuild_config(name: str, index):
    """Reconstruct the matching config for a full rebuild of any index
    kind (refresh full must not assume a covering index)."""
    kind = index.kind
    if kind == "CoveringIndex":
        from ..index.covering.config import CoveringIndexConfig
        return CoveringIndexConfig(name, index.indexed_columns,
                                   index.included_columns)
    if kind == "ZOrderCoveringIndex":
        from ..index.zorder.config import ZOrderCoveringIndexConfig
        return ZOrderCoveringIndexConfig(name, index.indexed_columns,
                                         index.included_columns)
    if kind == "DataSkippingIndex":
        from ..index.dataskipping.config import DataSkippingIndexConfig
        return DataSkippingIndexConfig(name, *index.sketches)
    raise HyperspaceException(f"Cannot rebuild index kind {kind}")


# ---------------------------------------------------------------------------
# Create
# ---------------------------------------------------------------------------

class CreateAction(Action):
    """Build a new index (reference: actions/CreateAction.scala:29-100 +
    CreateActionBase.scala:30-103)."""

    def __init__(self, session, df, config, log_manager: IndexLogManager,
                 data_manager: IndexDataManager):
        super().__init__(log_manager)
        self.session = session
        self.df = df
        self.config = config
        self.data_manager = data_manager
        self.event_logger = session.event_logger
        self._entry: Optional[IndexLogEntry] = None

    transient_state = States.CREATING
    final_state = States.ACTIVE

    def validate(self):
        from ..plan.nodes import Scan
        leaves = self.df.plan.collect_leaves()
        if len(leaves) != 1 or not isinstance(leaves[0], Scan):
            raise HyperspaceException(
                "Only single file-source relation plans are supported")
        relation = leaves[0].relation
        if not self.session.provider_manager.is_supported(relation):
            raise HyperspaceException(
                f"Unsupported relation: {relation.describe()}")
        from ..utils.resolver import resolve_all
        resolve_all(relation.schema.field_names(),
                    self.config.referenced_columns())
        latest = self.log_manager.get_latest_log()
        if latest is not None and latest.state != States.DOESNOTEXIST:
            raise HyperspaceException(
                f"Index {self.config.index_name} already exists")

    def op(self):
        from ..index.base import IndexerContext

        version = _agree_data_version(self.data_manager)
        data_path = self.data_manager.get_path(version)
        tracker = FileIdTracker()
        ctx = IndexerContext(self.session, tracker, data_path)

        properties = {}
        if self.session.conf.lineage_enabled:
            properties[IndexConstants.LINEAGE_PROPERTY] = "true"

        # freeze the source listing for the whole build: the scanned
        # files, the relation metadata and the signature must describe
        # one consistent snapshot even if the source is appended to
        # concurrently (unfrozen after — the user's DataFrame keeps its
        # dynamic listing)
        frozen = self.df.plan.collect_leaves()[0].relation.freeze()
        try:
            with self.session.with_rule_disabled():
                index, batch = self.config.create_index(
                    ctx, self.df, properties)
                written = index.write(ctx, batch)
            written = _all_written_files(data_path, written)
            # covering builds leave the sorted bucket-major batch on ctx
            # for the post-commit HBM cache write-through (local rank's
            # files only — matching the per-rank query-time file set)
            self.built_for_cache = getattr(ctx, "built_for_cache", None)

            # provider property enrichment (delta tables record the
            # index->table version history, reference CreateActionBase ->
            # FileBasedRelationMetadata.enrichIndexProperties)
            if hasattr(frozen, "enrich_index_properties"):
                index = index.with_new_properties(
                    frozen.enrich_index_properties(
                        index.properties, (self.base_id or 0) + 2))

            rel_meta = frozen.create_relation_metadata(tracker)
            self._entry = IndexLogEntry.create(
                self.config.index_name, index, _content_from_paths(written),
                Source(SourcePlan([rel_meta], _fingerprint(frozen))),
                index.properties)
        finally:
            frozen.unfreeze()

    def log_entry(self) -> IndexLogEntry:
        assert self._entry is not None
        return self._entry

    def log_entry_for_begin(self) -> IndexLogEntry:
        # content not yet known at begin(): record config-level metadata
        from ..log.entry import Directory
        relation = self.df.plan.collect_leaves()[0].relation
        rel_meta = relation.create_relation_metadata(FileIdTracker())
        placeholder = self.config.placeholder_index(
            relation, self.session.conf)
        return IndexLogEntry.create(
            self.config.index_name, placeholder, Content(Directory("")),
            Source(SourcePlan([rel_meta], _fingerprint(relation))), {})

    def event(self, message):
        return CreateActionEvent(index_name=self.config.index_name,
                                 message=message)


# ---------------------------------------------------------------------------
# Refresh
# ---------------------------------------------------------------------------

class RefreshActionBase(Action):
    """Shared refresh plumbing (reference: actions/RefreshActionBase.scala:
    37-129): reconstruct the source relation from logged metadata,
    diff current files vs logged files."""

    def __init__(self, session, log_manager, data_manager):
        super().__init__(log_manager)
        self.session = session
        self.data_manager = data_manager
        self.event_logger = session.event_logger
        self._entry: Optional[IndexLogEntry] = None
        prev = log_manager.get_latest_stable_log()
        if prev is None:
            raise HyperspaceException("Index does not exist")
        self.previous = prev

    transient_state = States.REFRESHING
    final_state = States.ACTIVE

    def validate(self):
        if self.previous.state != States.ACTIVE:
            raise HyperspaceException(
                f"Refresh requires ACTIVE index (is {self.previous.state})")

    def source_relation(self):
        # one frozen relation per action: the diff, the built index, the
        # relation metadata and the signature must all describe the SAME
        # file set — a concurrent append between any two of those would
        # otherwise commit a signature covering files the index lacks
        if getattr(self, "_src_relation", None) is None:
            rel_meta = self.previous.relations[0]
            self._src_relation = self.session.provider_manager \
                .from_metadata(rel_meta).refreshed().freeze()
        return self._src_relation

    def compute_diff(self) -> Tuple[List[FileInfo], List[FileInfo]]:
        """(appended, deleted) via FileInfo set-diff on (path,size,mtime)."""
        logged = {(f.name, f.size, f.modifiedTime): f
                  for f in self.previous.source_file_infos()}
        current = self.source_relation().all_files()
        cur_keys = set()
        appended = []
        for f in current:
            key = (f.name, f.size, f.modifiedTime)
            cur_keys.add(key)
            if key not in logged:
                appended.append(f)
        deleted = [f for k, f in logged.items() if k not in cur_keys]
        return appended, deleted

    def log_entry_for_begin(self):
        return self.previous

    def log_entry(self):
        assert self._entry is not None
        return self._entry

    def _enrich_index(self, index, relation):
        """Provider property enrichment on refresh (reference
        RefreshActionBase via FileBasedRelationMetadata
        .enrichIndexProperties — delta appends the new
        end-entry-id:table-version pair to the 'deltaVersions' history
        consumed by closestIndex time travel)."""
        if hasattr(relation, "enrich_index_properties"):
            return index.with_new_properties(
                relation.enrich_index_properties(
                    index.properties, (self.base_id or 0) + 2))
        return index


class RefreshFullAction(RefreshActionBase):
    """Full rebuild into a new data version."""

    def __init__(self, session, log_manager, data_manager):
        super().__init__(session, log_manager, data_manager)

    def op(self):
        appended, deleted = self.compute_diff()
        if not appended and not deleted:
            raise NoChangesException("Refresh full: no source changes")
        relation = self.source_relation()
        version = _agree_data_version(self.data_manager)
        data_path = self.data_manager.get_path(version)
        tracker = FileIdTracker()
        from ..index.base import IndexerContext
        ctx = IndexerContext(self.session, tracker, data_path)
        from ..dataframe import DataFrame
        from ..plan.nodes import Scan
        df = DataFrame(self.session, Scan(relation))
        index = self.previous.derivedDataset
        config = _rebuild_config(self.previous.name, index)
        with self.session.with_rule_disabled():
            new_index, batch = config.create_index(
                ctx, df, index.properties)
            written = new_index.write(ctx, batch)
        written = _all_written_files(data_path, written)
        new_index = self._enrich_index(new_index, relation)
        rel_meta = relation.create_relation_metadata(tracker)
        self._entry = IndexLogEntry.create(
            self.previous.name, new_index, _content_from_paths(written),
            Source(SourcePlan([rel_meta], _fingerprint(relation))),
            new_index.properties)

    def event(self, message):
        return RefreshActionEvent(index_name=self.previous.name,
                                  message=message)


class RefreshIncrementalAction(RefreshActionBase):
    """Index appended files only; rewrite deleted rows via lineage
    (reference: actions/RefreshIncrementalAction.scala:45-133)."""

    def op(self):
        appended, deleted = self.compute_diff()
        if not appended and not deleted:
            raise NoChangesException("Refresh incremental: no changes")
        index = self.previous.derivedDataset
        if deleted and not index.can_handle_deleted_files:
            raise HyperspaceException(
                "Index lacks lineage; cannot handle deleted source files. "
                "Recreate with lineage enabled or refresh in full mode.")
        tracker = _tracker_from_entry(self.previous)
        deleted_ids = [f.id for f in self.previous.source_file_infos()
                       if (f.name, f.size, f.modifiedTime) in
                       {(d.name, d.size, d.modifiedTime) for d in deleted}]

        version = _agree_data_version(self.data_manager)
        data_path = self.data_manager.get_path(version)
        from ..index.base import IndexerContext
        ctx = IndexerContext(self.session, tracker, data_path)

        previous_files = [p for p in self.previous.content.os_files()
                          if p.endswith(".parquet")]

        if index.kind == "DataSkippingIndex":
            # per-file sketch rows: appended files are sketched directly,
            # deleted files drop their rows
            with self.session.with_rule_disabled():
                written, kept = index.refresh_incremental_files(
                    ctx, [f.name for f in appended], deleted_ids,
                    previous_files)
        else:
            appended_batch = None
            if appended:
                relation = self.source_relation()
                from ..plan.nodes import Scan
                from ..execution.executor import Executor
                scan = Scan(relation)
                ex = Executor(self.session)
                with self.session.with_rule_disabled():
                    batch = ex._exec_scan(
                        scan, file_subset=[f.name for f in appended],
                        lineage_tracker=tracker if index.has_lineage
                        else None)
                    cols = index.indexed_columns + index.included_columns
                    if index.has_lineage:
                        cols = cols + [
                            IndexConstants.DATA_FILE_NAME_ID_COLUMN]
                    appended_batch = batch.select(cols)

            with self.session.with_rule_disabled():
                written, kept = index.refresh_incremental(
                    ctx, appended_batch, deleted_ids, previous_files)

        relation = self.source_relation()
        index = self._enrich_index(index, relation)
        rel_meta = relation.create_relation_metadata(tracker)
        all_files = sorted(set(written) | set(kept))
        self._entry = IndexLogEntry.create(
            self.previous.name, index, _content_from_paths(all_files),
            Source(SourcePlan([rel_meta], _fingerprint(relation))),
            index.properties)

    def event(self, message):
        return RefreshIncrementalActionEvent(index_name=self.previous.name,
                                             message=message)


class RefreshQuickAction(RefreshActionBase):
    """Metadata-only refresh: record appended/deleted in source.update;
    Hybrid Scan handles the delta at query time
    (reference: actions/RefreshQuickAction.scala:32-80)."""

    def op(self):
        appended, deleted = self.compute_diff()
        if not appended and not deleted:
            raise NoChangesException("Refresh quick: no changes")
        index = self.previous.derivedDataset
        if deleted and not index.can_handle_deleted_files:
            raise HyperspaceException(
                "Index lacks lineage; cannot handle deleted source files")
        tracker = _tracker_from_entry(self.previous)
        appended_fi = [
            FileInfo(f.name, f.size, f.modifiedTime,
                     tracker.add_file(f.name, f.size, f.modifiedTime))
            for f in appended]
        logged = {(f.name, f.size, f.modifiedTime): f
                  for f in self.previous.source_file_infos()}
        deleted_fi = [logged[(f.name, f.size, f.modifiedTime)]
                      for f in deleted]
        relation = self.source_relation()
        self._entry = self.previous.copy_with_update(
            _fingerprint(relation), appended_fi, deleted_fi)
        self._entry.properties = dict(self.previous.properties)

    def event(self, message):
        return RefreshQuickActionEvent(index_name=self.previous.name,
                                       message=message)


# ---------------------------------------------------------------------------
# Optimize
# ---------------------------------------------------------------------------

class OptimizeAction(Action):
    """Compact small index files per bucket
    (reference: actions/OptimizeAction.scala:57-148)."""

    def __init__(self, session, log_manager, data_manager,
                 mode: str = "quick"):
        super().__init__(log_manager)
        if mode not in ("quick", "full"):
            raise HyperspaceException(f"Invalid optimize mode: {mode}")
        self.session = session
        self.data_manager = data_manager
        self.mode = mode
        self.event_logger = session.event_logger
        prev = log_manager.get_latest_stable_log()
        if prev is None:
            raise HyperspaceException("Index does not exist")
        self.previous = prev
        self._entry: Optional[IndexLogEntry] = None

    transient_state = States.OPTIMIZING
    final_state = States.ACTIVE

    def validate(self):
        if self.previous.state != States.ACTIVE:
            raise HyperspaceException("Optimize requires ACTIVE index")

    def _select_files(self) -> Tuple[List[FileInfo], List[FileInfo]]:
        threshold = self.session.conf.optimize_file_size_threshold
        infos = [f for f in self.previous.content.os_file_infos()
                 if f.name.endswith(".parquet")]
        if self.mode == "quick":
            candidates = [f for f in infos if f.size < threshold]
        else:
            candidates = list(infos)
        # group by bucket id parsed from filename; only multi-file buckets
        by_bucket: Dict[int, List[FileInfo]] = {}
        for f in candidates:
            b = bucket_id_of_file(f.name)
            if b is not None:
                by_bucket.setdefault(b, []).append(f)
        to_optimize = [f for fs in by_bucket.values() if len(fs) >= 2
                       for f in fs]
        chosen = {f.name for f in to_optimize}
        ignored = [f for f in infos if f.name not in chosen]
        return to_optimize, ignored

    def op(self):
        to_optimize, self._ignored = self._select_files()
        if not to_optimize:
            raise NoChangesException("Optimize: no files to compact")
        version = _agree_data_version(self.data_manager)
        data_path = self.data_manager.get_path(version)
        from ..index.base import IndexerContext
        ctx = IndexerContext(self.session, _tracker_from_entry(self.previous),
                             data_path)
        index = self.previous.derivedDataset
        with self.session.with_rule_disabled():
            written = index.optimize(ctx, [f.name for f in to_optimize])
        all_files = sorted(set(written) | {f.name for f in self._ignored})
        self._entry = IndexLogEntry.create(
            self.previous.name, index, _content_from_paths(all_files),
            self.previous.source, index.properties)

    def log_entry_for_begin(self):
        return self.previous

    def log_entry(self):
        assert self._entry is not None
        return self._entry

    def event(self, message):
        return OptimizeActionEvent(index_name=self.previous.name,
                                   message=message)


# ---------------------------------------------------------------------------
# Delete / Restore / Vacuum / Cancel (log-state actions)
# ---------------------------------------------------------------------------

class _StateFlipAction(Action):
    def __init__(self, session, log_manager):
        super().__init__(log_manager)
        self.session = session
        self.event_logger = session.event_logger
        prev = log_manager.get_latest_stable_log()
        if prev is None:
            raise HyperspaceException("Index does not exist")
        self.previous = prev

    def validate(self):
        # re-read at run time: a concurrent action may have completed
        # between construction and run (validate-then-claim; the log-slot
        # claim still arbitrates true ties)
        prev = self.log_manager.get_latest_stable_log()
        if prev is None:
            raise HyperspaceException("Index does not exist")
        self.previous = prev

    def op(self):
        pass

    def log_entry(self):
        return self.previous

    def log_entry_for_begin(self):
        return self.previous


class DeleteAction(_StateFlipAction):
    """Soft delete: ACTIVE -> DELETED (reference: actions/DeleteAction)."""
    collective_op = False
    transient_state = States.DELETING
    final_state = States.DELETED

    def validate(self):
        _StateFlipAction.validate(self)
        if self.previous.state != States.ACTIVE:
            raise HyperspaceException("Delete requires ACTIVE index")

    def event(self, message):
        return DeleteActionEvent(index_name=self.previous.name,
                                 message=message)


class RestoreAction(_StateFlipAction):
    """DELETED -> ACTIVE (reference: actions/RestoreAction)."""
    collective_op = False
    transient_state = States.RESTORING
    final_state = States.ACTIVE

    def validate(self):
        _StateFlipAction.validate(self)
        if self.previous.state != States.DELETED:
            raise HyperspaceException("Restore requires DELETED index")

    def event(self, message):
        return RestoreActionEvent(index_name=self.previous.name,
                                  message=message)


class VacuumAction(_StateFlipAction):
    """Hard delete all files of a DELETED index -> DOESNOTEXIST
    (reference: actions/VacuumAction)."""
    collective_op = False
    transient_state = States.VACUUMING
    final_state = States.DOESNOTEXIST

    def __init__(self, session, log_manager, index_path: str):
        super().__init__(session, log_manager)
        self.index_path = index_path

    def validate(self):
        _StateFlipAction.validate(self)
        if self.previous.state != States.DELETED:
            raise HyperspaceException("Vacuum requires DELETED index")

    def op(self):
        for name in os.listdir(self.index_path):
            if name == IndexConstants.HYPERSPACE_LOG:
                continue
            full = os.path.join(self.index_path, name)
            if os.path.isdir(full):
                shutil.rmtree(full)
            else:
                os.unlink(full)

    def event(self, message):
        return VacuumActionEvent(index_name=self.previous.name,
                                 message=message)


class VacuumOutdatedAction(_StateFlipAction):
    """GC data versions not referenced by the latest content; keep index
    ACTIVE (reference: actions/VacuumOutdatedAction.scala:34-144)."""
    collective_op = False
    transient_state = States.VACUUMINGOUTDATED
    final_state = States.ACTIVE

    def __init__(self, session, log_manager, data_manager):
        super().__init__(session, log_manager)
        self.data_manager = data_manager

    def validate(self):
        _StateFlipAction.validate(self)
        if self.previous.state != States.ACTIVE:
            raise HyperspaceException(
                "VacuumOutdated requires ACTIVE index")

    def op(self):
        referenced = set(self.previous.content.os_files())
        ref_dirs = {os.path.dirname(p) for p in referenced}
        for version in self.data_manager.get_all_versions():
            vpath = os.path.abspath(self.data_manager.get_path(version))
            if vpath not in ref_dirs:
                self.data_manager.delete_version(version)
            else:
                for name in os.listdir(vpath):
                    full = os.path.join(vpath, name)
                    if os.path.isfile(full) and full not in referenced:
                        os.unlink(full)

    def log_entry(self):
        # outdated data versions are gone, so the delta time-travel
        # history must forget them: keep only the newest pair
        # (reference VacuumOutdatedAction.scala:56-67 via
        # DeltaLakeRelationMetadata resetting 'deltaVersions')
        entry = self.previous
        hist = entry.properties.get("deltaVersions")
        if hist and "," in hist:
            props = dict(entry.properties)
            props["deltaVersions"] = hist.split(",")[-1]
            import copy
            entry = copy.copy(entry)
            entry.properties = props
            entry.derivedDataset = entry.derivedDataset \
                .with_new_properties(props)
        return entry

    def event(self, message):
        return VacuumOutdatedActionEvent(index_name=self.previous.name,
                                         message=message)


class CancelAction(Action):
    """Escape hatch from a transient state back to the last stable log
    (reference: actions/CancelAction.scala)."""

    transient_state = States.CANCELLING
    final_state = States.ACTIVE  # replaced by stable entry's state

    def __init__(self, session, log_manager):
        super().__init__(log_manager)
        self.session = session
        self.event_logger = session.event_logger
        latest = log_manager.get_latest_log()
        if latest is None:
            raise HyperspaceException("Index does not exist")
        self.latest = latest

    def validate(self):
        if self.latest.state in States.STABLE_STATES:
            raise HyperspaceException(
                f"Cancel requires a transient state (is {self.latest.state})")

    def op(self):
        pass

    def log_entry_for_begin(self):
        return self.latest

    def log_entry(self):
        stable = self.log_manager.get_latest_stable_log()
        if stable is None:
            e = self.latest
            e.state = States.DOESNOTEXIST
            return e
        return stable

    def run(self):
        # custom: final state comes from the stable entry
        self.validate()
        base = self.log_manager.get_latest_id()
        self.base_id = base if base is not None else -1
        entry = self.log_entry_for_begin()
        entry.state = self.transient_state
        entry.timestamp = now_ms()
        self._write_or_fail(self.base_id + 1, entry)
        final = self.log_entry()
        final.timestamp = now_ms()
        self.log_manager.delete_latest_stable_log()
        self._write_or_fail(self.base_id + 2, final)
        self.log_manager.create_latest_stable_log(self.base_id + 2)
        self._log_event(CancelActionEvent(index_name=self.latest.name,
                                          message="Operation Succeeded."))
