"""Crash recovery / fault injection (the state machine is
the recovery story — a crash mid-action leaves a transient state;
readers trust only stable logs; cancel() restores; VacuumOutdated
collects stray files)."""

import os
import shutil

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.exceptions import HyperspaceException
from hyperspace_amd.log import IndexLogManager, States
from hyperspace_amd.plan.nodes import IndexScan


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    data = tmp_path / "data"
    data.mkdir()
    rng = np.random.default_rng(141)
    t = pa.table({"key": rng.integers(0, 100, 5000),
                  "val": rng.random(5000)})
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    return session, hs.Hyperspace(session), \
        session.read_parquet(str(data)), tmp_path


class _crash_during:
    """Context manager: the given covering-index method raises
    (simulated crash) inside the block."""

    def __init__(self, phase):
        from hyperspace_amd.index.covering.index import CoveringIndex
        self.cls = CoveringIndex
        self.phase = phase

    def __enter__(self):
        self.orig = getattr(self.cls, self.phase)

        def boom(self_, *a, **k):
            raise RuntimeError("injected crash")

        setattr(self.cls, self.phase, boom)

    def __exit__(self, *a):
        setattr(self.cls, self.phase, self.orig)


def test_crash_during_create_leaves_transient_then_cancel(env):
    session, h, df, tmp = env
    with _crash_during("write"):
        with pytest.raises(RuntimeError, match="injected"):
            h.create_index(df,
                           hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    mgr = IndexLogManager(str(tmp / "indexes" / "cx"))
    assert mgr.get_latest_log().state == States.CREATING  # dangling
    assert mgr.get_latest_stable_log() is None
    # readers see nothing usable
    assert session.index_manager().get_indexes([States.ACTIVE]) == []
    # cancel -> DOESNOTEXIST (no prior stable state)
    h.cancel("cx")
    assert mgr.get_latest_stable_log().state == States.DOESNOTEXIST
    # and the index can be created again afterwards
    h.create_index(df, hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    assert session.index_manager().get_index("cx").state == States.ACTIVE


def test_crash_during_refresh_keeps_old_index_usable(env):
    session, h, df, tmp = env
    h.create_index(df, hs.CoveringIndexConfig("rx", ["key"], ["val"]))
    old = session.index_manager().get_index("rx")
    # append + crash mid-refresh
    rng = np.random.default_rng(1)
    t = pa.table({"key": rng.integers(0, 100, 500),
                  "val": rng.random(500)})
    pq.write_table(t, str(tmp / "data" / "part-1.parquet"))
    with _crash_during("refresh_incremental"):
        with pytest.raises(RuntimeError, match="injected"):
            h.refresh_index("rx", "incremental")
    mgr = IndexLogManager(str(tmp / "indexes" / "rx"))
    assert mgr.get_latest_log().state == States.REFRESHING
    # stable readers still serve the OLD version
    entry = session.index_manager().get_index("rx")
    assert entry.state == States.ACTIVE
    assert entry.id == old.id
    # cancel back to ACTIVE, then the refresh can rerun cleanly
    h.cancel("rx")
    h.refresh_index("rx", "incremental")
    assert len(session.index_manager().get_index("rx")
               .source_file_infos()) == 2


def test_vacuum_outdated_collects_orphans(env):
    session, h, df, tmp = env
    h.create_index(df, hs.CoveringIndexConfig("vx", ["key"], ["val"]))
    # leave an orphaned version dir + a stray file in the live dir
    idx_dir = tmp / "indexes" / "vx"
    orphan = idx_dir / "v__=7"
    orphan.mkdir()
    (orphan / "junk.parquet").write_bytes(b"junk")
    live = idx_dir / "v__=0"
    stray = live / "stray.tmp"
    stray.write_bytes(b"stray")
    h.vacuum_index("vx")  # ACTIVE -> VacuumOutdated
    assert not orphan.exists()
    assert not stray.exists()
    entry = session.index_manager().get_index("vx")
    assert entry.state == States.ACTIVE
    # index still serves queries
    session.enable_hyperspace()
    plan = df.filter("key = 7").select("key", "val").optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())


def test_corrupt_latest_stable_falls_back_to_scan(env, tmp_path):
    session, h, df, tmp = env
    h.create_index(df, hs.CoveringIndexConfig("kx", ["key"], ["val"]))
    stable = tmp / "indexes" / "kx" / "_hyperspace_log" / "latestStable"
    stable.write_text("{ corrupted json !!")
    mgr = IndexLogManager(str(tmp / "indexes" / "kx"))
    entry = mgr.get_latest_stable_log()  # scan-back path
    assert entry is not None and entry.state == States.ACTIVE


def test_missing_index_files_fail_loudly(env):
    session, h, df, tmp = env
    h.create_index(df, hs.CoveringIndexConfig("mx", ["key"], ["val"]))
    entry = session.index_manager().get_index("mx")
    os.unlink(entry.content.os_files()[0])
    session.enable_hyperspace()
    # drop the build write-through residency: this test targets the
    # file READ path (a resident cache may legitimately keep serving
    # after out-of-band deletion; lifecycle ops invalidate via log id)
    cache = session.index_data_cache()
    if cache is not None:
        cache.clear()
    q = df.filter("key = 7").select("key", "val")
    with pytest.raises(Exception):
        q.collect()
