"""Property-based fuzzing: random schemas, data, predicates and
maintenance sequences — indexed results must equal an independent numpy
oracle.  (hypothesis drives the cases; the oracle never goes through the
engine's own evaluator.)"""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
from hypothesis import HealthCheck, given, settings, strategies as st

import hyperspace_amd as hs

OPS = ["=", "!=", "<", "<=", ">", ">="]
_NP_OP = {
    "=": lambda a, b: a == b, "!=": lambda a, b: a != b,
    "<": lambda a, b: a < b, "<=": lambda a, b: a <= b,
    ">": lambda a, b: a > b, ">=": lambda a, b: a >= b,
}


@st.composite
def scenario(draw):
    n_files = draw(st.integers(1, 4))
    rows_per_file = draw(st.integers(1, 400))
    key_lo = draw(st.integers(-1000, 0))
    key_hi = draw(st.integers(1, 1000))
    num_buckets = draw(st.sampled_from([1, 2, 7, 16]))
    op = draw(st.sampled_from(OPS))
    value = draw(st.integers(key_lo - 10, key_hi + 10))
    seed = draw(st.integers(0, 2**31 - 1))
    use_bucket_spec = draw(st.booleans())
    maintenance = draw(st.sampled_from(
        ["none", "append_refresh_incremental", "append_refresh_full",
         "optimize"]))
    return (n_files, rows_per_file, key_lo, key_hi, num_buckets, op,
            value, seed, use_bucket_spec, maintenance)


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(scenario())
def test_indexed_filter_matches_numpy_oracle(tmp_path_factory, sc):
    (n_files, rows_per_file, key_lo, key_hi, num_buckets, op, value,
     seed, use_bucket_spec, maintenance) = sc
    tmp = tmp_path_factory.mktemp("fuzz")
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp / "indexes")
    data = tmp / "data"
    data.mkdir()
    rng = np.random.default_rng(seed)
    all_keys = []
    all_vals = []
    for i in range(n_files):
        k = rng.integers(key_lo, key_hi + 1, rows_per_file)
        v = rng.random(rows_per_file)
        all_keys.append(k)
        all_vals.append(v)
        pq.write_table(pa.table({"key": k, "val": v}),
                       str(data / f"part-{i}.parquet"))

    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, num_buckets)
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     use_bucket_spec)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    h.create_index(df, hs.CoveringIndexConfig("fz", ["key"], ["val"]))

    if maintenance.startswith("append"):
        k = rng.integers(key_lo, key_hi + 1, 50)
        v = rng.random(50)
        all_keys.append(k)
        all_vals.append(v)
        pq.write_table(pa.table({"key": k, "val": v}),
                       str(data / "part-extra.parquet"))
        h.refresh_index("fz", maintenance.split("_")[-1]
                        if maintenance.endswith("full") else "incremental")
    elif maintenance == "optimize":
        h.optimize_index("fz", "full")

    keys = np.concatenate(all_keys)
    vals = np.concatenate(all_vals)
    mask = _NP_OP[op](keys, value)
    expected = sorted(zip(keys[mask].tolist(), vals[mask].tolist()))

    session.enable_hyperspace()
    q = df.filter(f"key {op} {value}").select("key", "val")
    from hyperspace_amd.plan.nodes import IndexScan
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan)
               for l in plan.collect_leaves()), plan.pretty()
    out = q.collect().to_numpy()
    got = sorted(zip(out["key"].tolist(), out["val"].tolist()))
    assert got == expected


@settings(max_examples=10, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(st.integers(0, 2**31 - 1), st.sampled_from([1, 3, 8]),
       st.integers(1, 50), st.integers(2, 500))
def test_indexed_join_matches_numpy_oracle(tmp_path_factory, seed,
                                           num_buckets, dup, key_space):
    tmp = tmp_path_factory.mktemp("fuzzj")
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp / "indexes")
    ldir, rdir = tmp / "l", tmp / "r"
    ldir.mkdir()
    rdir.mkdir()
    rng = np.random.default_rng(seed)
    lk = rng.integers(0, key_space, 300 * dup)
    lv = rng.random(300 * dup)
    rk = rng.integers(0, key_space, 200)
    rv = rng.integers(0, 9, 200)
    pq.write_table(pa.table({"key": lk, "val": lv}),
                   str(ldir / "p.parquet"))
    pq.write_table(pa.table({"key": rk, "status": rv}),
                   str(rdir / "p.parquet"))

    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, num_buckets)
    h = hs.Hyperspace(session)
    left = session.read_parquet(str(ldir))
    right = session.read_parquet(str(rdir))
    h.create_index(left, hs.CoveringIndexConfig("fl", ["key"], ["val"]))
    h.create_index(right, hs.CoveringIndexConfig("fr", ["key"], ["status"]))
    session.enable_hyperspace()

    q = left.select("key", "val").join(right.select("key", "status"),
                                       on="key")
    from hyperspace_amd.plan.nodes import IndexScan
    plan = q.optimized_plan()
    assert sum(isinstance(l, IndexScan)
               for l in plan.collect_leaves()) == 2
    out = q.collect().to_numpy()
    got = sorted(zip(out["key"].tolist(), out["val"].tolist(),
                     out["status"].tolist()))

    # numpy oracle join
    expected = []
    import collections
    rmap = collections.defaultdict(list)
    for k, s in zip(rk.tolist(), rv.tolist()):
        rmap[k].append(s)
    for k, v in zip(lk.tolist(), lv.tolist()):
        for s in rmap.get(k, []):
            expected.append((k, v, s))
    assert got == sorted(expected)
