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


def test_put_keeps_superset_incumbent():
    """A narrow-column put must not evict a wide incumbent under the same
    key (the build write-through stores the full slice; a later query's
    narrow put would otherwise thrash it)."""
    from hyperspace_amd.execution.index_cache import IndexDataCache
    from hyperspace_amd.execution.columnar import ColumnBatch
    import torch
    cache = IndexDataCache(budget_bytes=1 << 20)
    wide = ColumnBatch({"key": torch.arange(10), "val": torch.arange(10)})
    narrow = ColumnBatch({"key": torch.arange(10)})
    cache.put(("k",), wide, None)
    cache.put(("k",), narrow, None)
    got, _ = cache.get(("k",), columns=["key", "val"])
    assert set(got.names) == {"key", "val"}
    # a genuinely different-column put does replace
    other = ColumnBatch({"other": torch.arange(10)})
    cache.put(("k",), other, None)
    assert cache.get(("k",), columns=["key", "val"]) is None


def test_oversized_put_keeps_incumbent():
    """An oversized replacement must not drop a valid cached entry."""
    from hyperspace_amd.execution.index_cache import IndexDataCache
    from hyperspace_amd.execution.columnar import ColumnBatch
    import torch
    cache = IndexDataCache(budget_bytes=2000)
    small = ColumnBatch({"a": torch.zeros(100, dtype=torch.int64)})
    big = ColumnBatch({"b": torch.zeros(10000, dtype=torch.int64)})
    cache.put(("k",), small, None)
    cache.put(("k",), big, None)  # over budget: refused
    assert cache.get(("k",), columns=["a"]) is not None


def test_nbytes_counts_masks_and_strings():
    from hyperspace_amd.execution.columnar import ColumnBatch, StringColumn
    import torch
    b = ColumnBatch(
        {"s": StringColumn(torch.zeros(4, dtype=torch.int32), ["aa", "bb"]),
         "x": torch.zeros(4, dtype=torch.int64)},
        masks={"x": torch.ones(4, dtype=torch.bool)})
    # codes 16 + dict 4 + x 32 + mask 4
    assert b.nbytes() == 16 + 4 + 32 + 4
