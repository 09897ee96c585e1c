"""Candidate index collection per source relation.

Reference: index/rules/CandidateIndexCollector.scala:28-60 —
ColumnSchemaFilter (index columns ⊆ relation columns,
index/rules/ColumnSchemaFilter.scala) then FileSignatureFilter
(signature match, or Hybrid Scan file-overlap thresholds,
index/rules/FileSignatureFilter.scala:33-192).
"""

from __future__ import annotations

from typing import Dict, List, Tuple

from .filter_reason import FilterReason, FilterReasons, ReasonCollector
from ..log.constants import States
from ..log.entry import IndexLogEntry
from ..plan.nodes import Scan


# tags attached to candidate entries during collection
TAG_HYBRIDSCAN_REQUIRED = "HYBRIDSCAN_REQUIRED"
TAG_COMMON_SOURCE_SIZE = "COMMON_SOURCE_SIZE_IN_BYTES"
TAG_APPENDED_FILES = "HYBRIDSCAN_APPENDED_FILES"
TAG_DELETED_FILE_IDS = "HYBRIDSCAN_DELETED_FILE_IDS"


class Candidate:
    """An index entry admitted for a specific scan node, with hybrid-scan
    tags resolved at collection time."""

    def __init__(self, entry: IndexLogEntry):
        self.entry = entry
        self.tags: Dict[str, object] = {}

    @property
    def name(self):
        return self.entry.name

    @property
    def index(self):
        return self.entry.derivedDataset

    @property
    def hybrid_required(self) -> bool:
        return bool(self.tags.get(TAG_HYBRIDSCAN_REQUIRED))

    @property
    def common_bytes(self) -> int:
        return int(self.tags.get(TAG_COMMON_SOURCE_SIZE,
                                 self.entry.source_files_size()))


class CandidateIndexCollector:
    def __init__(self, session, reasons: ReasonCollector):
        self.session = session
        self.reasons = reasons

    def collect(self, plan, entries: List[IndexLogEntry]
                ) -> Dict[int, List[Candidate]]:
        """Map id(scan-node) -> admissible candidates."""
        out: Dict[int, List[Candidate]] = {}
        for leaf in plan.collect_leaves():
            if not isinstance(leaf, Scan):
                continue
            cands = []
            for entry in entries:
                if entry.state != States.ACTIVE:
                    continue
                c = self._admit(leaf, entry)
                if c is not None:
                    cands.append(c)
            if cands:
                out[id(leaf)] = cands
        return out

    # -- filters -----------------------------------------------------------
    def _admit(self, scan: Scan, entry: IndexLogEntry):
        if not self._column_schema_filter(scan, entry):
            return None
        return self._file_signature_filter(scan, entry)

    def _column_schema_filter(self, scan: Scan, entry: IndexLogEntry) -> bool:
        relation_cols = {c.lower() for c in scan.relation.schema
                         .field_names()}
        index_cols = {c.lower() for c in
                      entry.derivedDataset.referenced_columns()}
        if not index_cols <= relation_cols:
            self.reasons.add(entry.name, scan, FilterReason(
                FilterReasons.COL_SCHEMA_MISMATCH,
                {"indexCols": str(sorted(index_cols)),
                 "relationCols": str(sorted(relation_cols))}))
            return False
        return True

    def _file_signature_filter(self, scan: Scan, entry: IndexLogEntry):
        current_sig = scan.relation.signature()
        logged_sig = entry.signature
        cand = Candidate(entry)

        # fold in quick-refresh recorded deltas: the logged "current" state
        # is source files ∪ appended − deleted
        logged_files = {(f.name, f.size, f.modifiedTime): f
                        for f in entry.source_file_infos()}
        for f in entry.appended_files():
            logged_files[(f.name, f.size, f.modifiedTime)] = f
        for f in entry.deleted_files():
            logged_files.pop((f.name, f.size, f.modifiedTime), None)

        if logged_sig is not None and logged_sig.value == current_sig and \
                not entry.has_source_update():
            cand.tags[TAG_COMMON_SOURCE_SIZE] = entry.source_files_size()
            return cand

        # Delta time travel: swap in the closest retained index version
        # (reference FileSignatureFilter delegates closestIndex to the
        # relation, delta/DeltaLakeRelation.scala:179-251)
        closest_fn = getattr(scan.relation, "closest_index_log_entry",
                             None)
        if closest_fn is not None:
            import os as _os
            lm = self.session.index_manager().log_manager(entry.name)
            closest = closest_fn(entry, lm)
            if closest is not None and closest.signature is not None and \
                    closest.signature.value == current_sig and \
                    all(_os.path.exists(p)
                        for p in closest.content.os_files()[:1]):
                c2 = Candidate(closest)
                c2.tags[TAG_COMMON_SOURCE_SIZE] = \
                    closest.source_files_size()
                return c2

        # signature mismatch -> hybrid scan file-level overlap
        if not self.session.conf.hybrid_scan_enabled and \
                not entry.has_source_update():
            self.reasons.add(entry.name, scan, FilterReason(
                FilterReasons.SOURCE_DATA_CHANGED, {}))
            return None

        current = {(f.name, f.size, f.modifiedTime): f
                   for f in scan.relation.all_files()}
        common_keys = logged_files.keys() & current.keys()
        if not common_keys:
            self.reasons.add(entry.name, scan, FilterReason(
                FilterReasons.NO_COMMON_FILES, {}))
            return None

        common_bytes = sum(logged_files[k].size for k in common_keys)
        appended = [current[k] for k in current.keys() - logged_files.keys()]
        deleted_keys = logged_files.keys() - current.keys()
        appended_bytes = sum(f.size for f in appended)
        deleted_bytes = sum(logged_files[k].size for k in deleted_keys)

        appended_ratio = appended_bytes / max(
            1, appended_bytes + common_bytes)
        deleted_ratio = deleted_bytes / max(1, deleted_bytes + common_bytes)
        conf = self.session.conf
        if appended_ratio > conf.hybrid_scan_appended_ratio_threshold:
            self.reasons.add(entry.name, scan, FilterReason(
                FilterReasons.TOO_MUCH_APPENDED,
                {"ratio": f"{appended_ratio:.3f}"}))
            return None
        if deleted_keys:
            if not entry.derivedDataset.can_handle_deleted_files:
                self.reasons.add(entry.name, scan, FilterReason(
                    FilterReasons.NO_DELETE_SUPPORT, {}))
                return None
            if deleted_ratio > conf.hybrid_scan_deleted_ratio_threshold:
                self.reasons.add(entry.name, scan, FilterReason(
                    FilterReasons.TOO_MUCH_DELETED,
                    {"ratio": f"{deleted_ratio:.3f}"}))
                return None

        deleted_ids = [logged_files[k].id for k in deleted_keys
                       if logged_files[k].id >= 0]
        cand.tags[TAG_HYBRIDSCAN_REQUIRED] = bool(appended or deleted_keys)
        cand.tags[TAG_COMMON_SOURCE_SIZE] = common_bytes
        cand.tags[TAG_APPENDED_FILES] = [f.name for f in appended]
        cand.tags[TAG_DELETED_FILE_IDS] = deleted_ids
        return cand
