"""CSV / JSON-lines source tests (reference: default source supports
csv/json/etc — util/HyperspaceConf.scala:110-115)."""

import json

import numpy as np
import pytest

import hyperspace_amd as hs
from hyperspace_amd.plan.nodes import IndexScan


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    rng = np.random.default_rng(131)
    csv_dir = tmp_path / "csvdata"
    json_dir = tmp_path / "jsondata"
    csv_dir.mkdir()
    json_dir.mkdir()
    for i in range(2):
        keys = rng.integers(0, 200, 2000)
        vals = rng.random(2000)
        with open(csv_dir / f"part-{i}.csv", "w") as f:
            f.write("key,val\n")
            for k, v in zip(keys, vals):
                f.write(f"{k},{v}\n")
        with open(json_dir / f"part-{i}.json", "w") as f:
            for k, v in zip(keys, vals):
                f.write(json.dumps({"key": int(k), "val": float(v)}) + "\n")
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    return session, hs.Hyperspace(session), csv_dir, json_dir


def test_csv_scan_and_filter(env):
    session, h, csv_dir, _ = env
    df = session.read_csv(str(csv_dir))
    assert df.count() == 4000
    n = df.filter("key = 77").count()
    assert n > 0


def test_csv_index_build_and_query(env):
    session, h, csv_dir, _ = env
    df = session.read_csv(str(csv_dir))
    h.create_index(df, hs.CoveringIndexConfig("cix", ["key"], ["val"]))
    entry = session.index_manager().get_index("cix")
    assert entry.relations[0].fileFormat == "csv"
    # index data itself is parquet regardless of source format
    assert all(f.endswith(".parquet") for f in entry.content.os_files())
    session.enable_hyperspace()
    q = df.filter("key = 77").select("key", "val")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    accel = q.collect()
    session.disable_hyperspace()
    assert accel.num_rows == q.collect().num_rows


def test_json_index_build_and_query(env):
    session, h, _, json_dir = env
    df = session.read_json(str(json_dir))
    assert df.count() == 4000
    h.create_index(df, hs.CoveringIndexConfig("jix", ["key"], ["val"]))
    session.enable_hyperspace()
    q = df.filter("key = 77").select("key", "val")
    accel = q.collect()
    session.disable_hyperspace()
    assert accel.num_rows == q.collect().num_rows


def test_csv_refresh_incremental(env):
    session, h, csv_dir, _ = env
    df = session.read_csv(str(csv_dir))
    h.create_index(df, hs.CoveringIndexConfig("cix", ["key"], ["val"]))
    with open(csv_dir / "part-new.csv", "w") as f:
        f.write("key,val\n77,0.5\n77,0.6\n")
    h.refresh_index("cix", "incremental")
    session.enable_hyperspace()
    q = df.filter("key = 77").select("key", "val")
    accel = q.collect()
    session.disable_hyperspace()
    assert accel.num_rows == q.collect().num_rows


def test_orc_index_build_and_query(tmp_path, monkeypatch):
    import pyarrow as pa
    from pyarrow import orc
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    d = tmp_path / "orcdata"
    d.mkdir()
    rng = np.random.default_rng(7)
    t = pa.table({"key": rng.integers(0, 100, 3000),
                  "val": rng.random(3000)})
    orc.write_table(t, str(d / "part-0.orc"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_orc(str(d))
    assert df.count() == 3000
    h.create_index(df, hs.CoveringIndexConfig("oix", ["key"], ["val"]))
    session.enable_hyperspace()
    q = df.filter("key = 42").select("key", "val")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    accel = q.collect()
    session.disable_hyperspace()
    assert accel.num_rows == q.collect().num_rows


def test_camelcase_api_aliases(tmp_path, monkeypatch):
    """Reference python binding naming (py4j mirror) works verbatim."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    d = tmp_path / "d"
    d.mkdir()
    rng = np.random.default_rng(5)
    import pyarrow.parquet as pq2
    import pyarrow as pa
    pq2.write_table(pa.table({"key": rng.integers(0, 50, 1000),
                              "val": rng.random(1000)}),
                    str(d / "p.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 2)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.createIndex(df, hs.CoveringIndexConfig("ax", ["key"], ["val"]))
    h.refreshIndex("ax", "full")
    assert h.index("ax")["name"] == "ax"
    assert h.whyNot(df.filter("key = 1")) is not None
    h.deleteIndex("ax")
    h.restoreIndex("ax")
    h.deleteIndex("ax")
    h.vacuumIndex("ax")
