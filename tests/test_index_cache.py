"""Device/host-resident index data cache tests (reference analog:
IndexCacheTest for the metadata cache; the data cache is MI355X-first)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.execution.executor import Executor


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(41)
    data = tmp_path / "data"
    data.mkdir()
    t = pa.table({"key": rng.integers(0, 100, 5000),
                  "val": rng.random(5000)})
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    h.create_index(df, hs.CoveringIndexConfig("cix", ["key"], ["val"]))
    session.enable_hyperspace()
    return session, h, df, data, rng


def test_cache_hit_on_second_query(env):
    session, h, df, _, _ = env
    q = df.filter("key = 7").select("key", "val")
    q.collect()
    cache = session.index_data_cache()
    misses0 = cache.misses
    hits0 = cache.hits
    q.collect()
    assert cache.hits == hits0 + 1
    assert cache.misses == misses0
    # zero files scanned on the cached query
    ex = Executor(session)
    ex.execute(q.optimized_plan())
    assert ex.stats.scanned_files == 0


def test_cache_invalidated_by_refresh(env):
    session, h, df, data, rng = env
    q = df.filter("key = 7").select("key", "val")
    before = q.collect().num_rows
    # append data and refresh -> new log id -> new cache key
    t = pa.table({"key": np.full(100, 7, dtype=np.int64),
                  "val": rng.random(100)})
    pq.write_table(t, str(data / "part-1.parquet"))
    h.refresh_index("cix", "full")
    after = q.collect().num_rows
    assert after == before + 100


def test_cache_disabled_by_conf(env):
    session, h, df, _, _ = env
    session.conf.set("spark.hyperspace.index.dataCache.enabled", False)
    assert session.index_data_cache() is None
    q = df.filter("key = 7").select("key", "val")
    ex = Executor(session)
    ex.execute(q.optimized_plan())
    assert ex.stats.scanned_files > 0


def test_cache_eviction_budget():
    from hyperspace_amd.execution.index_cache import IndexDataCache
    from hyperspace_amd.execution.columnar import ColumnBatch
    import torch
    cache = IndexDataCache(budget_bytes=3000)

    class E:
        name = "x"
        id = 1
    b1 = ColumnBatch({"a": torch.zeros(128, dtype=torch.int64)})  # 1 KiB
    b2 = ColumnBatch({"a": torch.zeros(128, dtype=torch.int64)})
    b3 = ColumnBatch({"a": torch.zeros(128, dtype=torch.int64)})
    cache.put(("k1",), b1, None)
    cache.put(("k2",), b2, None)
    cache.put(("k3",), b3, None)  # evicts k1 (3 KiB > 3000 B budget)
    assert cache.get(("k1",)) is None
    assert cache.get(("k2",)) is not None
    assert cache.get(("k3",)) is not None


def test_build_write_through(env):
    """The first query after create must hit the cache (the build's
    sorted bucket-major batch is already HBM/host-resident in index-scan
    layout) and scan zero files."""
    session, h, df, _, _ = env
    from hyperspace_amd.execution.executor import Executor
    cache = session.index_data_cache()
    h.create_index(df, hs.CoveringIndexConfig("wt", ["key"], ["val"]))
    session.enable_hyperspace()
    hits0, misses0 = cache.hits, cache.misses
    ex = Executor(session)
    out = ex.execute(df.filter("key >= 0").select("key", "val")
                     .optimized_plan())
    assert out.num_rows == df.collect().num_rows
    assert cache.hits == hits0 + 1, (cache.hits, cache.misses)
    assert ex.stats.scanned_files == 0
