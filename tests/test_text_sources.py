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


# ---------------------------------------------------------------------------
# Avro (built-in container-format reader, sources/avro_io.py)
# ---------------------------------------------------------------------------

def test_avro_roundtrip(tmp_path):
    import numpy as np
    from hyperspace_amd.sources.avro_io import read_avro, write_avro
    rng = np.random.default_rng(3)
    vals = rng.integers(-(10**9), 10**9, 2000)
    d = rng.random(2000)
    names = np.array(["x", "yy", "zzz"])[rng.integers(0, 3, 2000)]
    mask = rng.random(2000) > 0.2
    p = str(tmp_path / "t.avro")
    write_avro({"k": vals, "v": d, "name": names.tolist()}, p,
               masks={"v": mask})
    t = read_avro(p)
    assert t.num_rows == 2000
    assert t.column("k").to_pylist() == vals.tolist()
    assert t.column("name").to_pylist() == names.tolist()
    got_v = t.column("v")
    assert got_v.null_count == int((~mask).sum())
    import pyarrow.compute as pc
    assert np.allclose(np.asarray(got_v.drop_null()), d[mask])


def test_avro_deflate_codec(tmp_path):
    # hand-build a deflate-codec file to cover the codec branch
    import json as _json
    import os
    import struct
    import zlib
    from hyperspace_amd.sources.avro_io import (MAGIC, _write_long,
                                                read_avro)
    schema = {"type": "record", "name": "r",
              "fields": [{"name": "a", "type": "long"}]}
    rows = [7, -3, 1 << 40]
    body = b"".join(_write_long(v) for v in rows)
    comp = zlib.compress(body)[2:-4]  # raw deflate
    sync = os.urandom(16)
    p = str(tmp_path / "d.avro")
    with open(p, "wb") as f:
        f.write(MAGIC)
        f.write(_write_long(2))
        for k, v in (("avro.schema", _json.dumps(schema).encode()),
                     ("avro.codec", b"deflate")):
            kb = k.encode()
            f.write(_write_long(len(kb)) + kb)
            f.write(_write_long(len(v)) + v)
        f.write(_write_long(0))
        f.write(sync)
        f.write(_write_long(len(rows)))
        f.write(_write_long(len(comp)))
        f.write(comp)
        f.write(sync)
    t = read_avro(p)
    assert t.column("a").to_pylist() == rows


def test_index_on_avro_source(tmp_path, monkeypatch):
    import numpy as np
    import hyperspace_amd as hs
    from hyperspace_amd.plan.nodes import IndexScan
    from hyperspace_amd.sources.avro_io import write_avro
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    d = tmp_path / "av"
    d.mkdir()
    rng = np.random.default_rng(4)
    key = rng.integers(0, 100, 5000)
    val = rng.random(5000)
    write_avro({"key": key, "val": val}, str(d / "part-0.avro"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_avro(str(d))
    h.create_index(df, hs.CoveringIndexConfig("avx", ["key"], ["val"]))
    session.enable_hyperspace()
    q = df.filter("key = 42").select("key", "val")
    assert any(isinstance(l, IndexScan)
               for l in q.optimized_plan().collect_leaves())
    out = q.collect()
    assert out.num_rows == int((key == 42).sum())


def test_text_format_lines(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    import hyperspace_amd as hs
    d = tmp_path / "logs"
    d.mkdir()
    (d / "a.txt").write_text("alpha\nbeta\ngamma\n")
    (d / "b.txt").write_text("beta\ndelta\n")
    session = hs.HyperspaceSession(device="cpu")
    df = session.read_text(str(d))
    out = df.collect()
    assert out.num_rows == 5
    vals = sorted(out.column("value").to_numpy().tolist())
    assert vals == ["alpha", "beta", "beta", "delta", "gamma"]
    got = df.filter("value = 'beta'").collect()
    assert got.num_rows == 2


def test_glob_paths(tmp_path):
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    import hyperspace_amd as hs
    from hyperspace_amd.exceptions import HyperspaceException
    rng = np.random.default_rng(5)
    for sub in ("d1", "d2", "other"):
        d = tmp_path / sub
        d.mkdir()
        pq.write_table(pa.table({"key": rng.integers(0, 10, 100)}),
                       str(d / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    df = session.read_parquet(str(tmp_path / "d*"))
    assert df.collect().num_rows == 200  # d1 + d2, not "other"
    with pytest.raises(HyperspaceException, match="matched nothing"):
        session.read_parquet(str(tmp_path / "zz*")).collect()
