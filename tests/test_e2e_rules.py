"""E2E query-acceleration tests (the reference's E2EHyperspaceRulesTest).

Covers: filter rewrite to index-only scan, bucket pruning, join rewrite to
the zero-shuffle co-bucketed merge join, result equivalence with the
unindexed run, explain/whyNot output.
"""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.plan.nodes import IndexScan, Join, Scan

N = 20000


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(3)
    left_dir = tmp_path / "left"
    right_dir = tmp_path / "right"
    left_dir.mkdir()
    right_dir.mkdir()
    for i in range(2):
        t = pa.table({
            "orderkey": rng.integers(0, 5000, N),
            "qty": rng.integers(1, 50, N),
            "price": rng.random(N) * 100,
        })
        pq.write_table(t, str(left_dir / f"part-{i}.parquet"))
    t = pa.table({
        "orderkey": np.arange(5000, dtype=np.int64),
        "status": rng.integers(0, 3, 5000),
    })
    pq.write_table(t, str(right_dir / "part-0.parquet"))

    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 16)
    h = hs.Hyperspace(session)
    left = session.read_parquet(str(left_dir))
    right = session.read_parquet(str(right_dir))
    return session, h, left, right


def _sorted_rows(batch, cols):
    arrs = batch.to_numpy()
    rows = sorted(zip(*[arrs[c].tolist() for c in cols]))
    return rows


def test_filter_rewrite_and_equivalence(env):
    session, h, left, _ = env
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    q = left.filter("qty = 7").select("qty", "price")

    baseline = q.collect()  # rules disabled by default
    session.enable_hyperspace()
    plan = q.optimized_plan()
    leaves = plan.collect_leaves()
    assert any(isinstance(l, IndexScan) for l in leaves), plan.pretty()
    accel = q.collect()
    assert _sorted_rows(accel, ["qty", "price"]) == \
        _sorted_rows(baseline, ["qty", "price"])


def test_filter_not_applied_without_first_indexed_col(env):
    session, h, left, _ = env
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    session.enable_hyperspace()
    # filter on price (an included, not indexed, column) -> no rewrite
    plan = left.filter("price >= 50.0").select("qty", "price") \
        .optimized_plan()
    assert not any(isinstance(l, IndexScan)
                   for l in plan.collect_leaves())


def test_filter_bucket_pruning(env):
    session, h, left, _ = env
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     True)
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    session.enable_hyperspace()
    q = left.filter("qty = 7").select("qty", "price")
    ex = Executor(session)
    out = ex.execute(q.optimized_plan())
    assert ex.stats.bucket_pruned_files > 0
    # correctness vs unindexed
    session.disable_hyperspace()
    base = q.collect()
    assert _sorted_rows(out, ["qty", "price"]) == \
        _sorted_rows(base, ["qty", "price"])


def test_join_rewrite_zero_shuffle(env):
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig(
        "lidx", ["orderkey"], ["qty"]))
    h.create_index(right, hs.CoveringIndexConfig(
        "ridx", ["orderkey"], ["status"]))
    q = left.select("orderkey", "qty").join(
        right.select("orderkey", "status"), on="orderkey")

    baseline = q.collect()
    session.enable_hyperspace()
    plan = q.optimized_plan()
    # both leaves are bucketed index scans
    leaves = plan.collect_leaves()
    index_leaves = [l for l in leaves if isinstance(l, IndexScan)]
    assert len(index_leaves) == 2, plan.pretty()
    assert all(l.use_bucket_spec for l in index_leaves)

    ex = Executor(session)
    accel = ex.execute(plan)
    assert ex.stats.shuffles == 0            # the whole point
    assert ex.stats.merge_joins == 1
    assert accel.num_rows == baseline.num_rows
    cols = ["orderkey", "qty", "status"]
    assert _sorted_rows(accel, cols) == _sorted_rows(baseline, cols)


def test_join_no_rewrite_when_keys_not_indexed(env):
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("lidx", ["qty"], ["price"]))
    session.enable_hyperspace()
    plan = left.select("orderkey", "qty").join(
        right.select("orderkey", "status"), on="orderkey").optimized_plan()
    assert not any(isinstance(l, IndexScan)
                   for l in plan.collect_leaves())


def test_source_change_blocks_index(env, tmp_path):
    session, h, left, _ = env
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    # append a file -> signature mismatch, hybrid scan off -> no rewrite
    rng = np.random.default_rng(9)
    t = pa.table({"orderkey": rng.integers(0, 5000, 100),
                  "qty": rng.integers(1, 50, 100),
                  "price": rng.random(100)})
    pq.write_table(t, str(tmp_path / "left" / "part-9.parquet"))
    session.enable_hyperspace()
    plan = left.filter("qty = 7").select("qty", "price").optimized_plan()
    assert not any(isinstance(l, IndexScan)
                   for l in plan.collect_leaves())


def test_explain_output(env):
    session, h, left, _ = env
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    q = left.filter("qty = 7").select("qty", "price")
    out = h.explain(q, verbose=True)
    assert "Plan with indexes" in out
    assert "fidx" in out
    assert "IndexScan" in out


def test_why_not_output(env):
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    # query that cannot use the index
    q = left.filter("price >= 50.0").select("price")
    out = h.why_not(q, extended=True)
    assert "fidx" in out
    assert "NO_FIRST_INDEXED_COL_COND" in out
    # and one that can
    q2 = left.filter("qty = 7").select("qty", "price")
    out2 = h.why_not(q2)
    assert "APPLIED" in out2


def test_score_prefers_join_over_filter(env):
    # when both a filter index and join indexes could apply to a join
    # query's sides, the join pair (70x2) beats per-side filter (50)
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig(
        "lidx", ["orderkey"], ["qty"]))
    h.create_index(right, hs.CoveringIndexConfig(
        "ridx", ["orderkey"], ["status"]))
    session.enable_hyperspace()
    q = left.select("orderkey", "qty").join(
        right.select("orderkey", "status"), on="orderkey")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves() if isinstance(l, IndexScan)]
    assert {s.entry.name for s in scans} == {"lidx", "ridx"}
    assert all(s.use_bucket_spec for s in scans)


def test_explain_display_modes(env):
    session, h, left, _ = env
    h.create_index(left, hs.CoveringIndexConfig("fidx", ["qty"], ["price"]))
    q = left.filter("qty = 7").select("qty", "price")
    session.conf.set(hs.IndexConstants.DISPLAY_MODE, "html")
    out = h.explain(q)
    assert "<b>" in out and "<br>" in out
    session.conf.set(hs.IndexConstants.DISPLAY_MODE, "console")
    out = h.explain(q)
    assert "\x1b[92m" in out
    session.conf.set(hs.IndexConstants.DISPLAY_MODE, "plaintext")
    out = h.explain(q)
    assert "<----" in out and "IndexScan" in out


def test_string_indexed_column(env, tmp_path):
    """Covering index on a string column: bucket hashing must depend on
    the VALUE (per-dictionary murmur3 LUT), not per-batch codes."""
    session, h, _, _ = env
    rng = np.random.default_rng(17)
    sdir = tmp_path / "sdata"
    sdir.mkdir()
    cities = np.array(["oslo", "lima", "pune", "kiel", "bern",
                       "reno", "kobe", "baku"])
    for i in range(2):
        # different value distributions per file -> different dicts
        vals = cities[rng.integers(i * 2, 4 + i * 4, 5000)]
        t = pa.table({"city": vals, "pop": rng.integers(0, 10**6, 5000)})
        pq.write_table(t, str(sdir / f"part-{i}.parquet"))
    df = session.read_parquet(str(sdir))
    h.create_index(df, hs.CoveringIndexConfig("six", ["city"], ["pop"]))
    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     True)
    import pyarrow.compute as pc
    full = pq.read_table(str(sdir))
    for city in cities:
        expected = int(pc.sum(
            pc.equal(full.column("city"), str(city))).as_py() or 0)
        q = df.filter(f"city = '{city}'").select("city", "pop")
        plan = q.optimized_plan()
        assert any(isinstance(l, IndexScan)
                   for l in plan.collect_leaves()), city
        ex = Executor(session)
        out = ex.execute(plan)
        assert out.num_rows == expected, city


def test_string_join_keys_not_rewritten(env, tmp_path):
    session, h, _, _ = env
    rng = np.random.default_rng(18)
    adir = tmp_path / "sa"
    bdir = tmp_path / "sb"
    adir.mkdir()
    bdir.mkdir()
    names = np.array(["a", "b", "c", "d"])
    pq.write_table(pa.table({"name": names[rng.integers(0, 4, 1000)],
                             "v": rng.random(1000)}),
                   str(adir / "part-0.parquet"))
    pq.write_table(pa.table({"name": names, "s": np.arange(4)}),
                   str(bdir / "part-0.parquet"))
    left = session.read_parquet(str(adir))
    right = session.read_parquet(str(bdir))
    h.create_index(left, hs.CoveringIndexConfig("sjl", ["name"], ["v"]))
    h.create_index(right, hs.CoveringIndexConfig("sjr", ["name"], ["s"]))
    session.enable_hyperspace()
    q = left.select("name", "v").join(right.select("name", "s"),
                                      on="name")
    plan = q.optimized_plan()
    # string join keys rewrite via the merged-dictionary remap
    assert all(isinstance(l, IndexScan) for l in plan.collect_leaves())
    ex = Executor(session)
    out = ex.execute(plan)
    assert ex.stats.merge_joins == 1 and ex.stats.shuffles == 0
    assert out.num_rows == 1000


def test_join_correct_after_incremental_refresh(env, tmp_path):
    """Regression: incremental refresh leaves multiple sorted files per
    bucket; the bucket must be re-sorted at scan time or the merge join
    returns wrong results."""
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig(
        "lidx", ["orderkey"], ["qty"]))
    h.create_index(right, hs.CoveringIndexConfig(
        "ridx", ["orderkey"], ["status"]))
    # append to the RIGHT source (the searched side of the merge join —
    # the side whose sortedness the join depends on) and refresh
    # incrementally -> second file per bucket in ridx
    rng = np.random.default_rng(23)
    t = pa.table({"orderkey": rng.integers(0, 5000, 3000),
                  "status": rng.integers(0, 3, 3000)})
    pq.write_table(t, str(tmp_path / "right" / "part-extra.parquet"))
    h.refresh_index("ridx", "incremental")
    entry = session.index_manager().get_index("ridx")
    from collections import Counter
    from hyperspace_amd.sources.parquet_io import bucket_id_of_file
    per_bucket = Counter(bucket_id_of_file(f)
                         for f in entry.content.os_files())
    assert max(per_bucket.values()) > 1  # multi-file buckets exist

    q = left.select("orderkey", "qty").join(
        right.select("orderkey", "status"), on="orderkey")
    baseline = q.collect()
    session.enable_hyperspace()
    from hyperspace_amd.execution.executor import Executor
    ex = Executor(session)
    accel = ex.execute(q.optimized_plan())
    assert ex.stats.merge_joins == 1
    assert accel.num_rows == baseline.num_rows
    cols = ["orderkey", "qty", "status"]
    assert _sorted_rows(accel, cols) == _sorted_rows(baseline, cols)


def test_covering_index_serves_arithmetic_filter(env):
    """An arithmetic predicate over the first indexed column still
    rewrites to the index-only scan; the executor evaluates the
    expression over index data."""
    from hyperspace_amd.plan.expr import col as _col
    session, h, left, _ = env
    h.create_index(left, hs.CoveringIndexConfig("arx", ["qty"],
                                                ["price"]))
    session.enable_hyperspace()
    q = left.filter((_col("qty") % 7) == 3).select("qty", "price")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = q.collect()
    session.disable_hyperspace()
    base = q.collect()
    assert out.num_rows == base.num_rows
    assert bool((out.tensor("qty") % 7 == 3).all())


def test_why_not_per_subplan_matrix(env):
    """whyNot renders the per-(index, subplan) reason matrix (reference
    CandidateIndexAnalyzer renders reasons against each sub-plan)."""
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("mfix", ["qty"],
                                                ["price"]))
    q = left.filter("price >= 50.0").select("price")
    out = h.why_not(q, extended=True)
    assert "Reasons per sub-plan:" in out
    assert "SubPlan #1:" in out
    assert "mfix" in out
    # the matrix names the sub-plan shape
    assert "Filter" in out or "Scan" in out


def test_minmax_html_report(env, tmp_path):
    session, h, left, right = env
    from hyperspace_amd.utils.minmax_analysis import analyze_html
    html = analyze_html(left, ["qty"])
    assert html.startswith("<!DOCTYPE html>")
    assert "<svg" in html and "rect" in html
    assert "qty" in html


def test_deleted_index_not_used_restored_is(env):
    """Soft-deleted indexes are invisible to the optimizer; restore
    brings them back (reference E2EHyperspaceRulesTest behavior)."""
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("del1", ["qty"],
                                                ["price"]))
    session.enable_hyperspace()
    q = left.filter("qty = 7").select("qty", "price")
    assert any(isinstance(l, IndexScan)
               for l in q.optimized_plan().collect_leaves())
    h.delete_index("del1")
    assert not any(isinstance(l, IndexScan)
                   for l in q.optimized_plan().collect_leaves())
    h.restore_index("del1")
    assert any(isinstance(l, IndexScan)
               for l in q.optimized_plan().collect_leaves())


def test_disable_enable_toggle(env):
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("tog", ["qty"],
                                                ["price"]))
    q = left.filter("qty = 7").select("qty", "price")
    session.disable_hyperspace()
    assert not any(isinstance(l, IndexScan)
                   for l in q.optimized_plan().collect_leaves())
    session.enable_hyperspace()
    assert any(isinstance(l, IndexScan)
               for l in q.optimized_plan().collect_leaves())


def test_uncovered_projection_blocks_rewrite(env):
    """A filter query projecting a column the index does not include
    cannot rewrite (MISSING_REQUIRED_COL)."""
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("cov1", ["qty"], []))
    session.enable_hyperspace()
    q = left.filter("qty = 7").select("qty", "price")
    assert not any(isinstance(l, IndexScan)
                   for l in q.optimized_plan().collect_leaves())
    out = h.why_not(q, index_name="cov1")
    assert "MISSING_REQUIRED_COL" in out


def test_projection_only_no_rewrite(env):
    """Project-without-filter plans are not FilterIndexRule targets
    (reference requires the Filter node)."""
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("pj1", ["qty"],
                                                ["price"]))
    session.enable_hyperspace()
    q = left.select("qty", "price")
    assert not any(isinstance(l, IndexScan)
                   for l in q.optimized_plan().collect_leaves())


def test_self_join_uses_same_index_both_sides(env):
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("sj", ["orderkey"],
                                                ["qty"]))
    session.enable_hyperspace()
    from hyperspace_amd.plan.expr import col
    q = left.select("orderkey", "qty").join(
        left.select("orderkey", "qty"), on="orderkey")
    plan = q.optimized_plan()
    scans = [l for l in plan.collect_leaves()
             if isinstance(l, IndexScan)]
    assert len(scans) == 2
    assert all(s.entry.name == "sj" for s in scans)
    from hyperspace_amd.execution.executor import Executor
    ex = Executor(session)
    out = ex.execute(plan)
    assert ex.stats.shuffles == 0
    session.disable_hyperspace()
    assert out.num_rows == q.collect().num_rows


def test_invalid_refresh_mode(env):
    session, h, left, right = env
    h.create_index(left, hs.CoveringIndexConfig("rm1", ["qty"], []))
    with pytest.raises(Exception, match="Unsupported refresh mode"):
        h.refresh_index("rm1", "bogus")
