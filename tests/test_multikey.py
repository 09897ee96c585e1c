"""Composite-key covering indexes: multi-column bucket hashing, LSD
multi-column sort, multi-key equi-join rewrites with secondary-key
verification (reference JoinIndexRule CNF extraction + JoinIndexRanker
same-order compatibility)."""

import numpy as np
import pandas as pd
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.plan.expr import col
from hyperspace_amd.plan.nodes import IndexScan

N = 30000


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(66)
    fd, dd = tmp_path / "f", tmp_path / "d"
    fd.mkdir()
    dd.mkdir()
    k1 = rng.integers(0, 50, N)
    k2 = rng.integers(0, 40, N)
    v = rng.random(N)
    pq.write_table(pa.table({"k1": k1, "k2": k2, "v": v}),
                   str(fd / "part-0.parquet"))
    dk1, dk2 = np.meshgrid(np.arange(50), np.arange(20))
    dk1, dk2 = dk1.ravel(), dk2.ravel()  # dim covers k2 < 20 only
    w = rng.random(dk1.size)
    pq.write_table(pa.table({"k1": dk1, "k2": dk2, "w": w}),
                   str(dd / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    return (session, hs.Hyperspace(session),
            session.read_parquet(str(fd)), session.read_parquet(str(dd)),
            pd.DataFrame({"k1": k1, "k2": k2, "v": v}),
            pd.DataFrame({"k1": dk1, "k2": dk2, "w": w}))


def test_multikey_join_cobucketed(env):
    session, h, fact, dim, fpd, dpd = env
    h.create_index(fact, hs.CoveringIndexConfig("mf", ["k1", "k2"], ["v"]))
    h.create_index(dim, hs.CoveringIndexConfig("md", ["k1", "k2"], ["w"]))
    session.enable_hyperspace()
    q = fact.join(dim, on=["k1", "k2"])
    plan = q.optimized_plan()
    assert sum(isinstance(l, IndexScan)
               for l in plan.collect_leaves()) == 2, plan.pretty()
    ex = Executor(session)
    out = ex.execute(plan)
    assert ex.stats.merge_joins == 1 and ex.stats.shuffles == 0
    expected = fpd.merge(dpd, on=["k1", "k2"])
    assert out.num_rows == len(expected)
    got = sorted(zip(out.tensor("k1").tolist(), out.tensor("k2").tolist(),
                     np.round(out.tensor("v").numpy(), 9)))
    want = sorted(zip(expected.k1, expected.k2,
                      np.round(expected.v.to_numpy(), 9)))
    assert got == want


def test_multikey_filter_first_col(env):
    session, h, fact, dim, fpd, dpd = env
    h.create_index(fact, hs.CoveringIndexConfig("mfx", ["k1", "k2"],
                                                ["v"]))
    session.enable_hyperspace()
    q = fact.filter((col("k1") == 7) & (col("k2") >= 20)) \
        .select("k1", "k2", "v")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = q.collect()
    expected = fpd[(fpd.k1 == 7) & (fpd.k2 >= 20)]
    assert out.num_rows == len(expected)


def test_multikey_sorted_layout(env):
    """Index files are sorted by (k1, k2) within each bucket."""
    session, h, fact, dim, fpd, dpd = env
    h.create_index(fact, hs.CoveringIndexConfig("ml", ["k1", "k2"],
                                                ["v"]))
    entry = session.index_manager().get_index("ml")
    from hyperspace_amd.sources.parquet_io import read_files_batch
    for f in entry.content.os_files():
        b, _ = read_files_batch([f])
        k1 = b.tensor("k1").numpy()
        k2 = b.tensor("k2").numpy()
        comp = k1.astype(np.int64) * (1 << 32) + k2
        assert (np.diff(comp) >= 0).all(), f
