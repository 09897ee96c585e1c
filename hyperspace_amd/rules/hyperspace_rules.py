"""The rewrite rules: FilterIndexRule, JoinIndexRule, NoOpRule.

Reference:
  - HyperspaceRule template  index/rules/HyperspaceRule.scala:28-91
  - FilterIndexRule          index/covering/FilterIndexRule.scala:33-174
    (requires a filter on the FIRST indexed column; index covers
    filter ∪ project columns; score = 50 × coverage)
  - JoinIndexRule            index/covering/JoinIndexRule.scala:47-720
    (equi-join CNF, linear children, compatible index pair with
    indexedColumns == join keys; score = 70 × coverage per side)
  - JoinIndexRanker          index/covering/JoinIndexRanker.scala:52-90
  - FilterIndexRanker        index/covering/FilterIndexRanker.scala:43-64
  - plan surgery             index/covering/CoveringIndexRuleUtils.scala
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from .candidate_collector import (Candidate, TAG_APPENDED_FILES,
                                  TAG_DELETED_FILE_IDS)
from .filter_reason import FilterReason, FilterReasons, ReasonCollector
from ..plan.expr import extract_equi_join_keys
from ..plan.nodes import (BucketUnionNode, Filter, IndexScan, Join,
                          LogicalPlan, Project, Scan, UnionNode)


class HyperspaceRule:
    name = "HyperspaceRule"

    def __init__(self, session, reasons: ReasonCollector):
        self.session = session
        self.reasons = reasons

    def apply(self, plan: LogicalPlan,
              candidates: Dict[int, List[Candidate]]
              ) -> Tuple[LogicalPlan, float]:
        raise NotImplementedError


class NoOpRule(HyperspaceRule):
    """Identity participant so child-only rewrites are explored
    (reference: index/rules/NoOpRule.scala)."""
    name = "NoOpRule"

    def apply(self, plan, candidates):
        return plan, 0.0


def _decompose_linear(plan: LogicalPlan):
    """Match Project?-Filter?-Scan; returns (project, filter, scan) with
    Nones, or None if the shape doesn't match."""
    project = filt = None
    node = plan
    if isinstance(node, Project):
        project = node
        node = node.child
    if isinstance(node, Filter):
        filt = node
        node = node.child
    if isinstance(node, Scan):
        return project, filt, node
    return None


def _needed_columns(project: Optional[Project], filt: Optional[Filter],
                    scan: Scan) -> List[str]:
    if project is not None:
        cols = list(project.columns)
        if filt is not None:
            for r in sorted(filt.condition.references()):
                if r.lower() not in {c.lower() for c in cols}:
                    cols.append(r)
        return cols
    return scan.relation.schema.field_names()


def _coverage(cand: Candidate) -> float:
    total = max(1, cand.entry.source_files_size())
    return min(1.0, cand.common_bytes / total)


def _index_scan_for(cand: Candidate, columns: List[str],
                    use_bucket_spec: bool) -> LogicalPlan:
    return IndexScan(cand.entry, columns, use_bucket_spec,
                     excluded_source_file_ids=list(
                         cand.tags.get(TAG_DELETED_FILE_IDS, [])))


def _hybrid_appended_scan(cand: Candidate, scan: Scan,
                          columns: List[str]) -> Optional[LogicalPlan]:
    appended = list(cand.tags.get(TAG_APPENDED_FILES, []))
    if not appended:
        return None
    # scan the SAME relation restricted to the appended files: partition
    # materialization and table-format read paths stay intact for
    # hive/delta/iceberg sources alike
    return Project(columns, Scan(scan.relation, scan.options,
                                 file_subset=appended))


class FilterIndexRule(HyperspaceRule):
    name = "FilterIndexRule"
    SCORE = 50.0

    def apply(self, plan, candidates):
        shape = _decompose_linear(plan)
        if shape is None:
            return plan, 0.0
        project, filt, scan = shape
        if filt is None:
            return plan, 0.0
        cands = candidates.get(id(scan), [])
        if not cands:
            return plan, 0.0

        needed = _needed_columns(project, filt, scan)
        filter_refs = {r.lower() for r in filt.condition.references()}
        eligible: List[Candidate] = []
        for cand in cands:
            index = cand.index
            if index.kind != "CoveringIndex":
                continue
            first_indexed = index.indexed_columns[0].lower()
            if first_indexed not in filter_refs:
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.NO_FIRST_INDEXED_COL_COND,
                    {"firstIndexedCol": index.indexed_columns[0]}))
                continue
            covered = {c.lower() for c in index.referenced_columns()}
            if index.has_lineage:
                from ..config import IndexConstants
                covered.add(IndexConstants.DATA_FILE_NAME_ID_COLUMN.lower())
            if not {c.lower() for c in needed} <= covered:
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.MISSING_REQUIRED_COL,
                    {"needed": str(needed)}))
                continue
            eligible.append(cand)
        if not eligible:
            return plan, 0.0

        best = self._rank(eligible)
        new_scan = self._rewrite_scan(best, scan, needed)
        new_plan: LogicalPlan = Filter(filt.condition, new_scan)
        if project is not None:
            new_plan = Project(project.columns, new_plan)
        score = self.SCORE * _coverage(best)
        self.reasons.applied.setdefault(best.name, []).append(self.name)
        return new_plan, score

    def _rank(self, cands: List[Candidate]) -> Candidate:
        """Hybrid-scan era: max common bytes; else min index size
        (reference: FilterIndexRanker.scala:43-64)."""
        if any(c.hybrid_required for c in cands):
            return max(cands, key=lambda c: c.common_bytes)
        return min(cands, key=lambda c: c.entry.index_files_size())

    def _rewrite_scan(self, cand: Candidate, scan: Scan,
                      needed: List[str]) -> LogicalPlan:
        use_bucket_spec = self.session.conf.filter_rule_use_bucket_spec
        index_scan = _index_scan_for(cand, needed, use_bucket_spec)
        appended = _hybrid_appended_scan(cand, scan, needed)
        if appended is None:
            return index_scan
        return UnionNode([index_scan, appended])


class JoinIndexRule(HyperspaceRule):
    name = "JoinIndexRule"
    SCORE = 70.0

    def apply(self, plan, candidates):
        if not isinstance(plan, Join):
            return plan, 0.0
        pairs = extract_equi_join_keys(plan.condition)
        if not pairs:
            self.reasons.add("", plan, FilterReason(
                FilterReasons.NOT_ELIGIBLE_JOIN, {"reason": "non equi-join"}))
            return plan, 0.0
        lshape = _decompose_linear(plan.left)
        rshape = _decompose_linear(plan.right)
        if lshape is None or rshape is None:
            self.reasons.add("", plan, FilterReason(
                FilterReasons.NOT_ELIGIBLE_JOIN,
                {"reason": "non-linear children"}))
            return plan, 0.0
        lproj, lfilt, lscan = lshape
        rproj, rfilt, rscan = rshape

        # resolve which side each key belongs to
        lcols = {c.lower() for c in lscan.relation.schema.field_names()}
        rcols = {c.lower() for c in rscan.relation.schema.field_names()}
        lkeys, rkeys = [], []
        for a, b in pairs:
            if a.lower() in lcols and b.lower() in rcols:
                lkeys.append(a)
                rkeys.append(b)
            elif b.lower() in lcols and a.lower() in rcols:
                lkeys.append(b)
                rkeys.append(a)
            else:
                self.reasons.add("", plan, FilterReason(
                    FilterReasons.NOT_ELIGIBLE_JOIN,
                    {"reason": f"cannot resolve join keys {a},{b}"}))
                return plan, 0.0

        lneeded = _needed_columns(lproj, lfilt, lscan)
        rneeded = _needed_columns(rproj, rfilt, rscan)
        lcands = self._eligible(candidates.get(id(lscan), []), lkeys,
                                lneeded, plan)
        rcands = self._eligible(candidates.get(id(rscan), []), rkeys,
                                rneeded, plan)
        if not lcands or not rcands:
            self.reasons.add("", plan, FilterReason(
                FilterReasons.NO_AVAIL_JOIN_INDEX_PAIR, {}))
            return plan, 0.0

        lbest, rbest = self._rank_pairs(lcands, rcands, lkeys, rkeys)
        if lbest is None:
            return plan, 0.0

        lnew = self._rewrite_side(lbest, lscan, lproj, lfilt, lneeded)
        rnew = self._rewrite_side(rbest, rscan, rproj, rfilt, rneeded)
        score = (self.SCORE * _coverage(lbest)
                 + self.SCORE * _coverage(rbest))
        self.reasons.applied.setdefault(lbest.name, []).append(self.name)
        self.reasons.applied.setdefault(rbest.name, []).append(self.name)
        from ..telemetry import HyperspaceIndexUsageEvent
        self.session.event_logger.log_event(HyperspaceIndexUsageEvent(
            index_names=[lbest.name, rbest.name],
            message="JoinIndexRule applied"))
        return Join(lnew, rnew, plan.condition, plan.join_type), score

    def _eligible(self, cands: List[Candidate], keys: List[str],
                  needed: List[str], plan) -> List[Candidate]:
        out = []
        for cand in cands:
            index = cand.index
            if index.kind != "CoveringIndex":
                continue
            # indexed columns must be exactly the join keys (set equality;
            # order compatibility across the pair is checked by the ranker)
            if set(c.lower() for c in index.indexed_columns) != \
                    set(k.lower() for k in keys):
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.NOT_ALL_JOIN_COL_INDEXED,
                    {"keys": str(keys)}))
                continue
            covered = {c.lower() for c in index.referenced_columns()}
            if index.has_lineage:
                from ..config import IndexConstants
                covered.add(IndexConstants.DATA_FILE_NAME_ID_COLUMN.lower())
            if not {c.lower() for c in needed} <= covered:
                self.reasons.add(cand.name, plan, FilterReason(
                    FilterReasons.MISSING_REQUIRED_COL,
                    {"needed": str(needed)}))
                continue
            out.append(cand)
        return out

    def _rank_pairs(self, lcands, rcands, lkeys, rkeys):
        """Compatible pairs ranked: equal bucket counts first, then more
        buckets, then common bytes (reference: JoinIndexRanker.scala:52-90)."""
        best = None
        best_rank = None
        for lc in lcands:
            for rc in rcands:
                if not self._compatible(lc, rc, lkeys, rkeys):
                    continue
                ln, rn = lc.index.num_buckets, rc.index.num_buckets
                rank = (1 if ln == rn else 0, min(ln, rn),
                        lc.common_bytes + rc.common_bytes)
                if best_rank is None or rank > best_rank:
                    best_rank = rank
                    best = (lc, rc)
        return best if best else (None, None)

    def _compatible(self, lc: Candidate, rc: Candidate,
                    lkeys: List[str], rkeys: List[str]) -> bool:
        """Indexed column order must correspond through the key pairing so
        both sides bucket/sort identically."""
        lmap = {k.lower(): v.lower() for k, v in zip(lkeys, rkeys)}
        lidx = [c.lower() for c in lc.index.indexed_columns]
        ridx = [c.lower() for c in rc.index.indexed_columns]
        if len(lidx) != len(ridx):
            return False
        try:
            return [lmap[c] for c in lidx] == ridx
        except KeyError:
            return False

    def _rewrite_side(self, cand: Candidate, scan: Scan,
                      proj: Optional[Project], filt: Optional[Filter],
                      needed: List[str]) -> LogicalPlan:
        index_scan = _index_scan_for(cand, needed, use_bucket_spec=True)
        appended = _hybrid_appended_scan(cand, scan, needed)
        if appended is not None:
            node: LogicalPlan = BucketUnionNode(
                [index_scan, appended], cand.index.num_buckets,
                cand.index.indexed_columns)
        else:
            node = index_scan
        if filt is not None:
            node = Filter(filt.condition, node)
        if proj is not None:
            node = Project(proj.columns, node)
        return node
