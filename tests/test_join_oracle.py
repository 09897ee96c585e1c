"""Join oracle: indexed join results validated against a ground truth
computed independently with numpy/collections (not against the engine's
own unindexed run), across randomized table shapes — int keys, string
keys, nullable keys, skew, and empty intersections.

Deterministic seeds (no flaky CI); each case builds covering indexes on
both sides so the plan rewrites to the zero-shuffle co-bucketed merge
join (reference: index/covering/JoinIndexRule.scala:47-720 semantics,
validated here at the result level).
"""

import collections

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs


def _expected_pairs(lkeys, rkeys):
    """Inner-join row count; None entries never match (SQL nulls)."""
    rc = collections.Counter(k for k in rkeys if k is not None)
    return sum(rc[k] for k in lkeys if k is not None)


def _write(dir_, name, tables):
    d = dir_ / name
    d.mkdir()
    for i, t in enumerate(tables):
        pq.write_table(t, str(d / f"part-{i}.parquet"))
    return str(d)


def _run_case(tmp_path, monkeypatch, case_id, left_tables, right_tables,
              lkeys, rkeys, buckets=8):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH",
                       str(tmp_path / f"idx{case_id}"))
    lpath = _write(tmp_path, f"l{case_id}", left_tables)
    rpath = _write(tmp_path, f"r{case_id}", right_tables)
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, buckets)
    h = hs.Hyperspace(session)
    left = session.read_parquet(lpath)
    right = session.read_parquet(rpath)
    h.create_index(left, hs.CoveringIndexConfig(
        f"jl{case_id}", ["k"], ["lv"]))
    h.create_index(right, hs.CoveringIndexConfig(
        f"jr{case_id}", ["k"], ["rv"]))
    session.enable_hyperspace()

    from hyperspace_amd.execution.executor import Executor
    from hyperspace_amd.plan.nodes import IndexScan
    q = left.select("k", "lv").join(right.select("k", "rv"), on="k")
    plan = q.optimized_plan()
    n_idx = sum(isinstance(l, IndexScan) for l in plan.collect_leaves())
    ex = Executor(session)
    out = ex.execute(plan)
    want = _expected_pairs(lkeys, rkeys)
    assert out.num_rows == want, (case_id, out.num_rows, want)
    assert n_idx == 2, f"{case_id}: join did not rewrite to both indexes"
    assert ex.stats.shuffles == 0, case_id
    return out


def test_join_oracle_int_keys_random_shapes(tmp_path, monkeypatch):
    rng = np.random.default_rng(2024)
    for case in range(6):
        n_l = int(rng.integers(1000, 20_000))
        n_r = int(rng.integers(100, 5000))
        dom = int(rng.integers(10, 3000))
        n_files_l = int(rng.integers(1, 4))
        lkeys = rng.integers(0, dom, n_l)
        rkeys = rng.integers(0, dom, n_r)
        bounds = np.linspace(0, n_l, n_files_l + 1).astype(int)
        left_tables = [pa.table({
            "k": lkeys[a:b], "lv": rng.random(b - a)})
            for a, b in zip(bounds[:-1], bounds[1:])]
        right_tables = [pa.table({"k": rkeys, "rv": rng.random(n_r)})]
        _run_case(tmp_path, monkeypatch, f"i{case}", left_tables,
                  right_tables, lkeys.tolist(), rkeys.tolist())


def test_join_oracle_skewed_and_empty(tmp_path, monkeypatch):
    rng = np.random.default_rng(7)
    # heavy skew: 80% of left rows share one key present 50x on right
    lkeys = np.where(rng.random(8000) < 0.8, 42,
                     rng.integers(1000, 2000, 8000))
    rkeys = np.concatenate([np.full(50, 42),
                            rng.integers(1000, 2000, 500)])
    _run_case(tmp_path, monkeypatch, "skew",
              [pa.table({"k": lkeys, "lv": rng.random(8000)})],
              [pa.table({"k": rkeys, "rv": rng.random(550)})],
              lkeys.tolist(), rkeys.tolist())
    # disjoint domains: zero matches
    lkeys = rng.integers(0, 100, 3000)
    rkeys = rng.integers(1000, 1100, 300)
    _run_case(tmp_path, monkeypatch, "disj",
              [pa.table({"k": lkeys, "lv": rng.random(3000)})],
              [pa.table({"k": rkeys, "rv": rng.random(300)})],
              lkeys.tolist(), rkeys.tolist())


def test_join_oracle_nullable_keys(tmp_path, monkeypatch):
    rng = np.random.default_rng(13)
    for case in range(3):
        n_l, n_r = 6000, 800
        lkeys = rng.integers(0, 500, n_l)
        rkeys = rng.integers(0, 500, n_r)
        lmask = rng.random(n_l) < 0.15   # True = null
        rmask = rng.random(n_r) < 0.15
        lt = pa.table({"k": pa.array(lkeys, mask=lmask),
                       "lv": rng.random(n_l)})
        rt = pa.table({"k": pa.array(rkeys, mask=rmask),
                       "rv": rng.random(n_r)})
        llist = [None if m else int(k) for k, m in zip(lkeys, lmask)]
        rlist = [None if m else int(k) for k, m in zip(rkeys, rmask)]
        _run_case(tmp_path, monkeypatch, f"n{case}", [lt], [rt],
                  llist, rlist)


def test_join_oracle_string_keys(tmp_path, monkeypatch):
    rng = np.random.default_rng(29)
    vocab_l = [f"sku-{i:04d}" for i in range(400)]
    # right vocab overlaps half of left's
    vocab_r = [f"sku-{i:04d}" for i in range(200, 600)]
    lkeys = [vocab_l[i] for i in rng.integers(0, 400, 7000)]
    rkeys = [vocab_r[i] for i in rng.integers(0, 400, 900)]
    _run_case(tmp_path, monkeypatch, "s",
              [pa.table({"k": lkeys, "lv": rng.random(7000)})],
              [pa.table({"k": rkeys, "rv": rng.random(900)})],
              lkeys, rkeys)


def test_join_oracle_value_integrity(tmp_path, monkeypatch):
    """Beyond counts: every output (k, lv, rv) triple must be a real
    pairing — checked via per-key value-sum aggregation."""
    rng = np.random.default_rng(31)
    n_l, n_r = 5000, 600
    lkeys = rng.integers(0, 300, n_l)
    rkeys = rng.integers(0, 300, n_r)
    lv = rng.integers(1, 1000, n_l).astype(np.float64)
    rv = rng.integers(1, 1000, n_r).astype(np.float64)
    out = _run_case(tmp_path, monkeypatch, "v",
                    [pa.table({"k": lkeys, "lv": lv})],
                    [pa.table({"k": rkeys, "rv": rv})],
                    lkeys.tolist(), rkeys.tolist())
    # oracle: per key, sum over the cross product = sum_l * count_r
    # (for lv) — so total lv-sum = sum_k (sum_lv[k] * n_r[k])
    sum_lv = collections.defaultdict(float)
    cnt_l = collections.Counter()
    for k, v in zip(lkeys, lv):
        sum_lv[int(k)] += v
        cnt_l[int(k)] += 1
    sum_rv = collections.defaultdict(float)
    cnt_r = collections.Counter()
    for k, v in zip(rkeys, rv):
        sum_rv[int(k)] += v
        cnt_r[int(k)] += 1
    want_lv = sum(sum_lv[k] * cnt_r[k] for k in sum_lv)
    want_rv = sum(sum_rv[k] * cnt_l[k] for k in sum_rv)
    got_lv = float(out.tensor("lv").sum())
    got_rv = float(out.tensor("rv").sum())
    assert got_lv == pytest.approx(want_lv, rel=1e-9)
    assert got_rv == pytest.approx(want_rv, rel=1e-9)
