"""String join keys: merge join across independently built indexes via
merged-dictionary code remapping (the reference joins strings through
Spark's UTF8String ordering; here each side's sorted dictionary remaps
monotonically onto the union dictionary, preserving bucket sortedness).
"""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor

N = 30000


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(11)
    # fact: string codes drawn from a large vocab; dim: the full vocab
    vocab = np.array([f"sku-{i:05d}" for i in range(2000)], dtype=object)
    fact_dir = tmp_path / "fact"
    dim_dir = tmp_path / "dim"
    fact_dir.mkdir()
    dim_dir.mkdir()
    fkeys = vocab[rng.integers(0, 2000, N)]
    fval = rng.random(N)
    for i in range(2):
        sl = slice(i * N // 2, (i + 1) * N // 2)
        pq.write_table(pa.table({"sku": fkeys[sl], "v": fval[sl]}),
                       str(fact_dir / f"part-{i}.parquet"))
    # dim covers only half the vocab so the join filters rows
    dkeys = vocab[:1000]
    pq.write_table(pa.table({"sku": dkeys,
                             "w": np.arange(1000, dtype=np.int64)}),
                   str(dim_dir / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    fact = session.read_parquet(str(fact_dir))
    dim = session.read_parquet(str(dim_dir))
    return session, h, fact, dim, fkeys, fval, dkeys


def _expected_rows(fkeys, fval, dkeys):
    dset = {k: i for i, k in enumerate(dkeys)}
    return sorted((k, round(v, 9), dset[k])
                  for k, v in zip(fkeys, fval) if k in dset)


def _got_rows(out):
    arrs = out.to_numpy()
    return sorted((k, round(v, 9), int(w))
                  for k, v, w in zip(arrs["sku"], arrs["v"], arrs["w"]))


def test_string_join_unindexed(env):
    session, h, fact, dim, fkeys, fval, dkeys = env
    out = fact.join(dim, on="sku").collect()
    assert _got_rows(out) == _expected_rows(fkeys, fval, dkeys)


def test_string_join_indexed_cobucketed(env):
    session, h, fact, dim, fkeys, fval, dkeys = env
    h.create_index(fact, hs.CoveringIndexConfig("sf", ["sku"], ["v"]))
    h.create_index(dim, hs.CoveringIndexConfig("sd", ["sku"], ["w"]))
    session.enable_hyperspace()
    q = fact.join(dim, on="sku")
    plan = q.optimized_plan()
    ex = Executor(session)
    out = ex.execute(plan)
    assert ex.stats.merge_joins == 1 and ex.stats.shuffles == 0, \
        plan.pretty()
    assert _got_rows(out) == _expected_rows(fkeys, fval, dkeys)


def test_string_join_with_nulls(env, tmp_path):
    session, h, fact, dim, fkeys, fval, dkeys = env
    rng = np.random.default_rng(12)
    nd = tmp_path / "nfact"
    nd.mkdir()
    mask = rng.random(1000) > 0.3
    keys = np.array([f"sku-{i:05d}" for i in
                     rng.integers(0, 1000, 1000)], dtype=object)
    pq.write_table(
        pa.table({"sku": pa.array(keys.tolist(), mask=~mask),
                  "v": rng.random(1000)}),
        str(nd / "part-0.parquet"))
    nfact = session.read_parquet(str(nd))
    out = nfact.join(dim, on="sku").collect()
    assert out.num_rows == int(mask.sum())  # null keys dropped
