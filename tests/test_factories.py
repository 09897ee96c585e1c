"""DI factory-seam tests: actions driven through mocked log/data
managers injected via the IndexCollectionManager factories (reference:
actions/CreateActionTest.scala and index/factories.scala — the action
tests mock managers through exactly these seams)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.exceptions import HyperspaceException
from hyperspace_amd.index_management import IndexCollectionManager
from hyperspace_amd.log import (IndexDataManager, IndexDataManagerFactory,
                                IndexLogManager, IndexLogManagerFactory)
from hyperspace_amd.log.constants import States


class RecordingLogManager(IndexLogManager):
    """Real behavior + a call log the tests assert on."""

    def __init__(self, path, calls):
        super().__init__(path)
        self.calls = calls

    def write_log(self, log_id, entry):
        self.calls.append(("write_log", log_id, entry.state))
        return super().write_log(log_id, entry)

    def create_latest_stable_log(self, log_id):
        self.calls.append(("stable", log_id))
        return super().create_latest_stable_log(log_id)


class FailingWriteLogManager(RecordingLogManager):
    """Refuses the begin write -> the optimistic-concurrency loss."""

    def write_log(self, log_id, entry):
        self.calls.append(("write_log", log_id, entry.state))
        return False


class RecordingFactory(IndexLogManagerFactory):
    def __init__(self, cls):
        self.cls = cls
        self.calls = []
        self.created = []

    def create(self, path):
        m = self.cls(path, self.calls)
        self.created.append(m)
        return m


class RecordingDataFactory(IndexDataManagerFactory):
    def __init__(self):
        self.paths = []

    def create(self, path):
        self.paths.append(path)
        return IndexDataManager(path)


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(5)
    d = tmp_path / "src"
    d.mkdir()
    pq.write_table(pa.table({"k": rng.integers(0, 50, 2000),
                             "v": rng.random(2000)}),
                   str(d / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    return session, session.read_parquet(str(d))


def test_create_action_through_factory_seam(env):
    session, df = env
    log_f = RecordingFactory(RecordingLogManager)
    data_f = RecordingDataFactory()
    mgr = IndexCollectionManager(session, log_manager_factory=log_f,
                                 data_manager_factory=data_f)
    mgr.create(df, hs.CoveringIndexConfig("fix", ["k"], ["v"]))
    # begin (id 0, CREATING) then end (id 1, ACTIVE) + latestStable
    writes = [c for c in log_f.calls if c[0] == "write_log"]
    assert writes == [("write_log", 0, States.CREATING),
                      ("write_log", 1, States.ACTIVE)]
    assert ("stable", 1) in log_f.calls
    assert len(data_f.paths) == 1  # data manager resolved via factory
    entry = mgr.get_index("fix")
    assert entry is not None and entry.state == States.ACTIVE


def test_lost_race_surfaces_through_factory_seam(env):
    session, df = env
    log_f = RecordingFactory(FailingWriteLogManager)
    mgr = IndexCollectionManager(session, log_manager_factory=log_f,
                                 data_manager_factory=RecordingDataFactory())
    with pytest.raises(HyperspaceException, match="Could not acquire"):
        mgr.create(df, hs.CoveringIndexConfig("fx2", ["k"], ["v"]))
    # exactly the begin write was attempted, nothing was committed
    assert log_f.calls == [("write_log", 0, States.CREATING)]
    assert log_f.created[-1].get_latest_stable_log() is None


def test_delete_restore_vacuum_through_seams(env):
    session, df = env
    log_f = RecordingFactory(RecordingLogManager)
    mgr = IndexCollectionManager(session, log_manager_factory=log_f,
                                 data_manager_factory=RecordingDataFactory())
    mgr.create(df, hs.CoveringIndexConfig("fx3", ["k"], ["v"]))
    log_f.calls.clear()
    mgr.delete("fx3")
    assert [c[2] for c in log_f.calls if c[0] == "write_log"] == \
        [States.DELETING, States.DELETED]
    mgr.restore("fx3")
    mgr.delete("fx3")
    log_f.calls.clear()
    mgr.vacuum("fx3")
    states = [c[2] for c in log_f.calls if c[0] == "write_log"]
    assert states == [States.VACUUMING, States.DOESNOTEXIST]
