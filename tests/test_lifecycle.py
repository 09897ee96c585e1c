"""Index lifecycle (actions state machine) integration tests on CPU.

Mirrors the reference's IndexManagerTest / RefreshIndexTest behaviors.
"""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.exceptions import HyperspaceException
from hyperspace_amd.log import IndexLogManager, States
from hyperspace_amd.sources.parquet_io import bucket_id_of_file
from hyperspace_amd.telemetry import (CreateActionEvent, DeleteActionEvent,
                                      RecordingEventLogger)

N_ROWS = 5000


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    data_dir = tmp_path / "data"
    data_dir.mkdir()
    rng = np.random.default_rng(7)
    for i in range(3):
        t = pa.table({
            "key": rng.integers(0, 500, N_ROWS),
            "val": rng.random(N_ROWS),
        })
        pq.write_table(t, str(data_dir / f"part-{i}.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 10)
    session.event_logger = RecordingEventLogger()
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data_dir))
    return session, h, df, data_dir, rng


def _log_states(tmp_path_root, name):
    mgr = IndexLogManager(os.path.join(tmp_path_root, name))
    latest = mgr.get_latest_id()
    return [mgr.get_log(i).state for i in range(latest + 1)]


def test_create_lifecycle(env, tmp_path):
    session, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    states = _log_states(str(tmp_path / "indexes"), "ix")
    assert states == [States.CREATING, States.ACTIVE]
    # bucket filename contract
    entry = session.index_manager().get_index("ix")
    for f in entry.content.os_files():
        assert bucket_id_of_file(f) is not None
    # events
    evs = [e for e in session.event_logger.events
           if isinstance(e, CreateActionEvent)]
    assert len(evs) == 1


def test_create_duplicate_fails(env):
    _, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    with pytest.raises(HyperspaceException, match="already exists"):
        h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))


def test_delete_restore_vacuum(env, tmp_path):
    session, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    h.delete_index("ix")
    assert session.index_manager().get_index("ix").state == States.DELETED
    h.restore_index("ix")
    assert session.index_manager().get_index("ix").state == States.ACTIVE
    h.delete_index("ix")
    h.vacuum_index("ix")
    entry = session.index_manager().get_index("ix")
    assert entry.state == States.DOESNOTEXIST
    # data files are gone
    idx_dir = tmp_path / "indexes" / "ix"
    remaining = [p for p in os.listdir(idx_dir) if p != "_hyperspace_log"]
    assert remaining == []
    assert any(isinstance(e, DeleteActionEvent)
               for e in session.event_logger.events)


def test_delete_requires_active(env):
    _, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    h.delete_index("ix")
    with pytest.raises(HyperspaceException):
        h.delete_index("ix")


def test_refresh_full_new_version(env, tmp_path):
    session, h, df, data_dir, rng = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    # append a source file
    t = pa.table({"key": rng.integers(0, 500, 1000),
                  "val": rng.random(1000)})
    pq.write_table(t, str(data_dir / "part-new.parquet"))
    h.refresh_index("ix", "full")
    entry = session.index_manager().get_index("ix")
    assert entry.state == States.ACTIVE
    assert all("v__=1" in f for f in entry.content.os_files())
    assert len(entry.source_file_infos()) == 4


def test_refresh_no_changes_is_noop(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    h.refresh_index("ix", "full")  # no source change
    entry = session.index_manager().get_index("ix")
    assert entry.state == States.ACTIVE
    assert all("v__=0" in f for f in entry.content.os_files())


def test_refresh_incremental_appends(env, tmp_path):
    session, h, df, data_dir, rng = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    old_entry = session.index_manager().get_index("ix")
    old_files = set(old_entry.content.os_files())
    t = pa.table({"key": rng.integers(0, 500, 1000),
                  "val": rng.random(1000)})
    pq.write_table(t, str(data_dir / "part-new.parquet"))
    h.refresh_index("ix", "incremental")
    entry = session.index_manager().get_index("ix")
    new_files = set(entry.content.os_files())
    assert old_files < new_files  # old files kept, new files added
    assert any("v__=1" in f for f in new_files - old_files)
    assert len(entry.source_file_infos()) == 4


def test_refresh_quick_records_update(env, data_dir=None):
    session, h, df, data_dir, rng = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    t = pa.table({"key": rng.integers(0, 500, 1000),
                  "val": rng.random(1000)})
    pq.write_table(t, str(data_dir / "part-new.parquet"))
    h.refresh_index("ix", "quick")
    entry = session.index_manager().get_index("ix")
    assert len(entry.appended_files()) == 1
    # index data untouched
    assert all("v__=0" in f for f in entry.content.os_files())


def test_optimize_compacts_buckets(env):
    session, h, df, data_dir, rng = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    # create second batch of files per bucket via incremental refresh
    t = pa.table({"key": rng.integers(0, 500, 4000),
                  "val": rng.random(4000)})
    pq.write_table(t, str(data_dir / "part-new.parquet"))
    h.refresh_index("ix", "incremental")
    before = session.index_manager().get_index("ix")
    n_before = len(before.content.os_files())
    h.optimize_index("ix", "quick")
    after = session.index_manager().get_index("ix")
    n_after = len(after.content.os_files())
    assert n_after < n_before
    # every bucket now has exactly one file
    buckets = [bucket_id_of_file(f) for f in after.content.os_files()]
    assert len(buckets) == len(set(buckets))


def test_optimize_noop_when_single_file_buckets(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    h.optimize_index("ix", "quick")  # single file per bucket -> no-op
    entry = session.index_manager().get_index("ix")
    assert entry.state == States.ACTIVE


def test_cancel_from_transient(env, tmp_path):
    session, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    # manufacture a stuck transient state
    mgr = IndexLogManager(str(tmp_path / "indexes" / "ix"))
    entry = mgr.get_latest_log()
    entry.state = States.REFRESHING
    assert mgr.write_log(mgr.get_latest_id() + 1, entry)
    h.cancel("ix")
    latest = mgr.get_latest_log()
    assert latest.state == States.ACTIVE


def test_index_stats(env):
    session, h, df, _, _ = env
    h.create_index(df, hs.CoveringIndexConfig("ix", ["key"], ["val"]))
    stats = h.index("ix")
    assert stats["numBuckets"] == 10
    assert stats["numSourceFiles"] == 3
    assert stats["sizeOfIndexInBytes"] > 0
    assert [d["name"] for d in h.indexes()] == ["ix"]


def test_jsonl_event_logger(env, tmp_path, monkeypatch):
    import json
    session, h, df, _, _ = env
    log_path = str(tmp_path / "events.jsonl")
    monkeypatch.setenv("HYPERSPACE_EVENT_LOG", log_path)
    session.conf.set(hs.IndexConstants.EVENT_LOGGER_CLASS,
                     "hyperspace_amd.telemetry.JsonlEventLogger")
    session.event_logger = None  # force re-resolution from conf
    h.create_index(df, hs.CoveringIndexConfig("jx", ["key"], ["val"]))
    h.delete_index("jx")
    with open(log_path) as f:
        events = [json.loads(l) for l in f]
    assert [e["event"] for e in events] == ["CreateActionEvent",
                                           "DeleteActionEvent"]
    assert events[0]["index_name"] == "jx"


def test_streaming_build_multi_group(env, tmp_path):
    """Small group size forces a multi-group streaming build: several
    sorted files per bucket, queries still exact."""
    session, h, df, data_dir, rng = env
    # each source file ~78KB; 100KB groups -> 3 groups
    session.conf.set("spark.hyperspace.index.build.groupBytes", 100_000)
    h.create_index(df, hs.CoveringIndexConfig("sg", ["key"], ["val"]))
    entry = session.index_manager().get_index("sg")
    from collections import Counter
    per_bucket = Counter(bucket_id_of_file(f)
                         for f in entry.content.os_files())
    assert max(per_bucket.values()) > 1  # one file per (group, bucket)
    session.enable_hyperspace()
    session.conf.set(hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC,
                     True)
    q = df.filter("key = 77").select("key", "val")
    accel = q.collect()
    session.disable_hyperspace()
    base = q.collect()
    a = sorted(zip(accel.to_numpy()["key"].tolist(),
                   accel.to_numpy()["val"].tolist()))
    b = sorted(zip(base.to_numpy()["key"].tolist(),
                   base.to_numpy()["val"].tolist()))
    assert a == b and len(a) > 0


def test_streaming_vs_materialized_same_rows(env, tmp_path):
    session, h, df, _, _ = env
    session.conf.set("spark.hyperspace.index.build.groupBytes", 100_000)
    h.create_index(df, hs.CoveringIndexConfig("s1", ["key"], ["val"]))
    session.conf.set("spark.hyperspace.index.build.groupBytes", 0)
    h.create_index(df, hs.CoveringIndexConfig("s2", ["key"], ["val"]))
    import pyarrow.parquet as pq2
    e1 = session.index_manager().get_index("s1")
    e2 = session.index_manager().get_index("s2")
    t1 = pq2.read_table(e1.content.os_files()).sort_by(
        [("key", "ascending"), ("val", "ascending")])
    t2 = pq2.read_table(e2.content.os_files()).sort_by(
        [("key", "ascending"), ("val", "ascending")])
    assert t1.equals(t2)


def test_optimize_after_full_bucket_deletion(tmp_path, monkeypatch):
    """Regression (found by the lifecycle fuzzer): deleting every source
    file that fed a bucket file used to leave a 0-row index file whose
    empty native read broke optimize's compaction."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    import pyarrow as pa2
    import pyarrow.parquet as pq2
    rng = np.random.default_rng(0)
    d = tmp_path / "data"
    d.mkdir()

    def wf(i):
        pq2.write_table(
            pa2.table({"key": rng.integers(0, 200, 1500),
                       "val": rng.random(1500)}),
            str(d / f"part-{i:04d}.parquet"))

    wf(0)
    wf(1)
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("opt0", ["key"], ["val"]))
    wf(2)
    os.unlink(str(d / "part-0000.parquet"))
    os.unlink(str(d / "part-0001.parquet"))
    h.refresh_index("opt0", mode="incremental")
    h.optimize_index("opt0")  # used to raise KeyError('key')
    session.enable_hyperspace()
    t2 = pq2.read_table(str(d / "part-0002.parquet"))
    k2 = t2.column("key").to_numpy()
    out = df.filter("key = 5").select("key", "val").collect()
    assert out.num_rows == int((k2 == 5).sum())


def test_optimize_full_vs_quick_thresholds(env, tmp_path):
    """quick mode only compacts buckets whose files sit under the
    fileSizeThreshold; full mode compacts every multi-file bucket
    (reference OptimizeAction.scala:57-148)."""
    session, h, df, data, rng = env
    import pyarrow as pa
    import pyarrow.parquet as pq
    h.create_index(df, hs.CoveringIndexConfig("opt2", ["key"], ["val"]))
    # two refreshes -> up to 3 files per bucket
    for i in range(2):
        t = pa.table({"key": rng.integers(0, 100, 2000),
                      "val": rng.random(2000)})
        pq.write_table(t, str(data / f"part-opt{i}.parquet"))
        h.refresh_index("opt2", "incremental")
    entry = session.index_manager().get_index("opt2")
    from collections import Counter
    from hyperspace_amd.sources.parquet_io import bucket_id_of_file
    per_bucket = Counter(bucket_id_of_file(p)
                         for p in entry.content.os_files())
    assert max(per_bucket.values()) >= 2
    # quick with a tiny threshold: nothing qualifies -> NoChanges no-op
    session.conf.set(hs.IndexConstants.OPTIMIZE_FILE_SIZE_THRESHOLD, 1)
    h.optimize_index("opt2", "quick")
    entry_q = session.index_manager().get_index("opt2")
    per_bucket_q = Counter(bucket_id_of_file(p)
                           for p in entry_q.content.os_files())
    assert per_bucket_q == per_bucket  # unchanged
    # full ignores the threshold: every multi-file bucket compacts to 1
    h.optimize_index("opt2", "full")
    entry_f = session.index_manager().get_index("opt2")
    per_bucket_f = Counter(bucket_id_of_file(p)
                           for p in entry_f.content.os_files())
    assert max(per_bucket_f.values()) == 1
    # results unchanged
    session.enable_hyperspace()
    n = df.filter("key = 7").collect().num_rows
    session.disable_hyperspace()
    assert n == df.filter("key = 7").collect().num_rows
