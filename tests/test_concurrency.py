"""Optimistic concurrency on the metadata log (reference §5.2:
atomic-rename claims a log id; the loser of the race aborts)."""

import os
import threading

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.exceptions import HyperspaceException
from hyperspace_amd.log import IndexLogManager, States
from hyperspace_amd.log.entry import IndexLogEntry


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    data = tmp_path / "data"
    data.mkdir()
    rng = np.random.default_rng(71)
    t = pa.table({"key": rng.integers(0, 100, 5000),
                  "val": rng.random(5000)})
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    return session, hs.Hyperspace(session), session.read_parquet(str(data))


def test_write_log_claims_slot_once(env, tmp_path):
    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    mgr = IndexLogManager(str(tmp_path / "indexes" / "cx"))
    entry = mgr.get_latest_log()
    next_id = mgr.get_latest_id() + 1
    assert mgr.write_log(next_id, entry) is True
    # second claim of the same slot loses the race
    assert mgr.write_log(next_id, entry) is False


def test_concurrent_log_writers_one_wins(env, tmp_path):
    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    mgr = IndexLogManager(str(tmp_path / "indexes" / "cx"))
    entry = mgr.get_latest_log()
    next_id = mgr.get_latest_id() + 1
    results = []
    barrier = threading.Barrier(8)

    def writer(i):
        m = IndexLogManager(str(tmp_path / "indexes" / "cx"))
        barrier.wait()
        results.append(m.write_log(next_id, entry))

    threads = [threading.Thread(target=writer, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert sum(results) == 1  # exactly one winner


def test_concurrent_maintenance_raises_cleanly(env, tmp_path):
    """Two maintenance ops racing on the same index: one succeeds, the
    other fails with 'Could not acquire proper state'."""
    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    outcomes = []
    barrier = threading.Barrier(2)

    def do_delete():
        s2 = hs.HyperspaceSession(device="cpu")
        h2 = hs.Hyperspace(s2)
        barrier.wait()
        try:
            h2.delete_index("cx")
            outcomes.append("ok")
        except HyperspaceException:
            outcomes.append("lost")

    threads = [threading.Thread(target=do_delete) for _ in range(2)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert sorted(outcomes) == ["lost", "ok"]
    # log ends in a stable DELETED state either way
    mgr = IndexLogManager(str(tmp_path / "indexes" / "cx"))
    assert mgr.get_latest_stable_log().state == States.DELETED


def test_reader_ignores_inflight_transient(env, tmp_path):
    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    mgr = IndexLogManager(str(tmp_path / "indexes" / "cx"))
    # simulate a crashed action: dangling transient state
    entry = mgr.get_latest_log()
    entry.state = States.REFRESHING
    mgr.write_log(mgr.get_latest_id() + 1, entry)
    stable = mgr.get_latest_stable_log()
    assert stable.state == States.ACTIVE  # readers only trust stable


def test_minmax_analysis_output(env, tmp_path):
    session, h, df = env
    from hyperspace_amd.utils.minmax_analysis import analyze
    out = analyze(df, ["key"])
    assert "column: key" in out
    assert "avg files per point lookup" in out


def test_corrupt_latest_stable_scan_back(env, tmp_path):
    """A torn latestStable copy falls back to the scan-back path
    (reference getLatestStableLog semantics)."""
    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("cx", ["key"], ["val"]))
    log_dir = os.path.join(str(tmp_path / "indexes"), "cx",
                           "_hyperspace_log")
    with open(os.path.join(log_dir, "latestStable"), "w") as f:
        f.write("{ torn json")
    entry = session.index_manager().get_index("cx")
    assert entry is not None and entry.state == States.ACTIVE
    session.enable_hyperspace()
    assert df.filter("key = 5").collect().num_rows >= 0


def test_corrupt_tip_entry_scan_back(env, tmp_path):
    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("cy", ["key"], ["val"]))
    log_dir = os.path.join(str(tmp_path / "indexes"), "cy",
                           "_hyperspace_log")
    # fake a torn in-flight entry above the stable tip + torn stable copy
    with open(os.path.join(log_dir, "7"), "w") as f:
        f.write("not json at all")
    os.unlink(os.path.join(log_dir, "latestStable"))
    entry = session.index_manager().get_index("cy")
    assert entry is not None and entry.state == States.ACTIVE


def _race_create(rank, tmpdir, results):
    # spawned fresh process (torch mp.spawn): real cross-process race
    os.environ["HYPERSPACE_SYSTEM_PATH"] = os.path.join(tmpdir, "indexes")
    import hyperspace_amd as hs2
    session = hs2.HyperspaceSession(device="cpu")
    session.conf.set(hs2.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h2 = hs2.Hyperspace(session)
    df2 = session.read_parquet(os.path.join(tmpdir, "data"))
    try:
        h2.create_index(df2, hs2.CoveringIndexConfig(
            "race", ["key"], ["val"]))
        results[rank] = "ok"
    except Exception as e:  # noqa: BLE001
        results[rank] = f"lost: {type(e).__name__}"


def test_cross_process_create_race(env, tmp_path):
    """Two PROCESSES race to create the same index: the hard-link log
    claim arbitrates — at least one wins, the log ends consistent, and
    the surviving index serves queries."""
    import torch.multiprocessing as mp
    session, h, df = env
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_race_create, args=(str(tmp_path), results), nprocs=2,
                 join=True)
        res = dict(results)
    assert sum(1 for v in res.values() if v == "ok") >= 1, res
    entry = session.index_manager().get_index("race")
    assert entry is not None and entry.state == States.ACTIVE
    session.enable_hyperspace()
    out = df.filter("key = 5").select("key", "val").collect()
    assert out.num_rows >= 0


def test_query_during_refresh_sees_consistent_snapshot(env, tmp_path):
    """Readers resolve indexes through latestStable only: while a full
    refresh rewrites the index, concurrent queries must return either
    the old or the new answer — never a torn state."""
    import threading

    session, h, df = env
    h.create_index(df, hs.CoveringIndexConfig("qr", ["key"], ["val"]))
    session.enable_hyperspace()
    baseline = df.filter("key = 5").select("key", "val").collect().num_rows

    data_dir = df.plan.collect_leaves()[0].relation.root_paths[0]
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    rng = np.random.default_rng(123)
    extra_key = rng.integers(0, 100, 2000)
    pq.write_table(pa.table({"key": extra_key,
                             "val": rng.random(2000)}),
                   data_dir + "/part-9.parquet")
    new_expected = baseline + int((extra_key == 5).sum())

    results = []
    errors = []
    stop = threading.Event()

    def reader():
        while not stop.is_set():
            try:
                n = df.filter("key = 5").select("key", "val") \
                    .collect().num_rows
                results.append(n)
            except Exception as e:  # noqa: BLE001
                errors.append(repr(e))

    t = threading.Thread(target=reader)
    t.start()
    try:
        h.refresh_index("qr", mode="full")
    finally:
        stop.set()
        t.join(30)
    assert not errors, errors[:3]
    # every observation is one of the two legal snapshots
    assert set(results) <= {baseline, new_expected}, set(results)
    # and the post-refresh answer is the new one
    final = df.filter("key = 5").select("key", "val").collect().num_rows
    assert final == new_expected


def test_concurrent_maintenance_fuzz(env, tmp_path):
    """Three threads fire random maintenance ops at one index while a
    reader queries continuously: losers of log races surface as
    HyperspaceException (or no-op), state converges to a stable one,
    and the surviving index answers correctly."""
    import random
    import threading
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq

    session, h, df = env
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h.create_index(df, hs.CoveringIndexConfig("cfz", ["key"], ["val"]))
    session.enable_hyperspace()
    data_dir = df.plan.collect_leaves()[0].relation.root_paths[0]
    rng = np.random.default_rng(5)
    stop = threading.Event()
    hard_errors = []

    def writer(seed):
        r = random.Random(seed)
        i = 100 + seed
        while not stop.is_set():
            # vacuumOutdated is excluded: deleting outdated versions
            # while queries are in flight is documented-unsafe in the
            # reference too (retention is the operator's contract)
            op = r.choice(["append_refresh", "optimize", "del_restore"])
            try:
                if op == "append_refresh":
                    # atomic publish (write temp, rename): readers must
                    # never see half-written source files — the same
                    # contract Spark ingestion relies on
                    tmp_p = data_dir + f"/.tmp-{seed}-{i}"
                    pq.write_table(
                        pa.table({"key": rng.integers(0, 100, 500),
                                  "val": rng.random(500)}), tmp_p)
                    os.rename(tmp_p,
                              data_dir + f"/part-{seed}-{i}.parquet")
                    i += 1
                    h.refresh_index("cfz", mode="incremental")
                elif op == "optimize":
                    h.optimize_index("cfz")
                else:
                    h.delete_index("cfz")
                    h.restore_index("cfz")
            except HyperspaceException:
                pass  # lost race / wrong-state: the designed outcome
            except Exception:  # noqa: BLE001
                import traceback as _tb
                hard_errors.append(_tb.format_exc())

    def reader():
        while not stop.is_set():
            try:
                df.filter("key = 5").select("key", "val").collect()
            except HyperspaceException:
                pass
            except Exception as e:  # noqa: BLE001
                hard_errors.append("reader: " + repr(e))

    threads = [threading.Thread(target=writer, args=(s,))
               for s in range(3)] + [threading.Thread(target=reader)]
    for t in threads:
        t.start()
    import time
    time.sleep(6)
    stop.set()
    for t in threads:
        t.join(60)
    assert not hard_errors, hard_errors[:5]
    # converged: a stable log entry exists; restore if soft-deleted
    from hyperspace_amd.log.constants import States as St
    entry = session.index_manager().get_index("cfz")
    assert entry is not None and entry.state in (St.ACTIVE, St.DELETED)
    if entry.state == St.DELETED:
        h.restore_index("cfz")
    # final correctness vs the surviving source files
    got = df.filter("key = 5").select("key", "val").collect()
    import pyarrow.parquet as pq2
    t_all = pq2.read_table(data_dir)
    want = int((t_all.column("key").to_numpy() == 5).sum())
    assert got.num_rows == want


def test_relation_freeze_pins_listing(env, tmp_path):
    """freeze() pins all_files/signature; unfreeze restores dynamic
    listing (the invariant behind the action-layer snapshot fix)."""
    session, h, df = env
    rel = df.plan.collect_leaves()[0].relation
    frozen = rel.freeze()
    sig0 = frozen.signature()
    n0 = len(frozen.all_files())
    pq.write_table(pa.table({"key": np.arange(10),
                             "val": np.arange(10.0)}),
                   str(tmp_path / "data" / "part-new.parquet"))
    # frozen view: the concurrent append is invisible
    assert len(frozen.all_files()) == n0
    assert frozen.signature() == sig0
    frozen.unfreeze()
    assert len(rel.all_files()) == n0 + 1
    assert rel.signature() != sig0


def test_refresh_commits_signature_of_built_snapshot(env, tmp_path):
    """A file appended AFTER a refresh computed its diff must not be
    covered by the committed signature: the post-refresh query must see
    a signature mismatch and fall back to a full scan (deterministic
    replay of the fuzz-found stale-result bug)."""
    from hyperspace_amd.actions.actions import RefreshIncrementalAction

    session, h, df = env
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h.create_index(df, hs.CoveringIndexConfig("frz", ["key"], ["val"]))
    data = tmp_path / "data"
    pq.write_table(pa.table({"key": np.full(100, 5),
                             "val": np.zeros(100)}),
                   str(data / "part-1.parquet"))

    mgr = session.index_manager()
    _, log_mgr, data_mgr = mgr._managers("frz")
    action = RefreshIncrementalAction(session, log_mgr, data_mgr)
    # diff computed (freezes the listing) ...
    appended, deleted = action.compute_diff()
    assert [os.path.basename(f.name) for f in appended] == \
        ["part-1.parquet"]
    # ... then a racing writer appends another file mid-action
    pq.write_table(pa.table({"key": np.full(50, 5),
                             "val": np.ones(50)}),
                   str(data / "part-2.parquet"))
    action.run()

    # committed signature describes the frozen snapshot, NOT the current
    # source -> the index is stale and must NOT serve this query
    entry = mgr.get_index("frz")
    live_sig = df.plan.collect_leaves()[0].relation.signature()
    assert entry.signature.value != live_sig

    session.enable_hyperspace()
    got = df.filter("key = 5").select("key", "val").collect()
    t_all = pq.read_table(str(data))
    want = int((t_all.column("key").to_numpy() == 5).sum())
    assert got.num_rows == want


def test_concurrent_queries_during_refresh(tmp_path, monkeypatch):
    """N query threads hammer an indexed filter while a refresh runs:
    every result must be correct (pre- or post-refresh view), the
    shared HBM/CPU index-data cache must stay consistent (locked LRU),
    and the maintenance kill-switch must keep the refresh's own reads
    index-free."""
    import threading
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "ix"))
    rng = np.random.default_rng(17)
    d = tmp_path / "src"
    d.mkdir()
    key = rng.integers(0, 500, 20_000)
    expected1 = int((key == 77).sum())
    pq.write_table(pa.table({"key": key, "val": rng.random(20_000)}),
                   str(d / "p0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("cqx", ["key"], ["val"]))
    session.enable_hyperspace()
    # append then refresh concurrently with queries
    key2 = rng.integers(0, 500, 5_000)
    expected2 = expected1 + int((key2 == 77).sum())
    pq.write_table(pa.table({"key": key2, "val": rng.random(5_000)}),
                   str(d / "p1.parquet"))
    errors = []
    counts = set()

    def worker():
        try:
            for _ in range(8):
                q = session.read_parquet(str(d)).filter("key = 77") \
                    .select("key", "val")
                counts.add(q.collect().num_rows)
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=worker) for _ in range(6)]
    for t in threads:
        t.start()
    h.refresh_index("cqx", "incremental")
    for t in threads:
        t.join()
    assert not errors, errors
    # every observed count is one of the two consistent views
    assert counts <= {expected1, expected2}, (counts, expected1,
                                              expected2)
    # post-refresh queries see the merged view
    q = session.read_parquet(str(d)).filter("key = 77").select("key")
    assert q.collect().num_rows == expected2
