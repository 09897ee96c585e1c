"""Wide-table coverage: a 10-column source (ints of all widths, floats,
strings, nullable columns) through covering-index build, projection
coverage checks, refresh and optimize — catches layout/dtype
assumptions that 2-column tests miss."""

import numpy as np
import pandas as pd
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.plan.nodes import IndexScan

N = 20000


@pytest.fixture
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(77)
    d = tmp_path / "wide"
    d.mkdir()
    nmask = rng.random(N) > 0.15
    cols = {
        "key": rng.integers(0, 500, N),
        "i32": rng.integers(-(2**31), 2**31 - 1, N).astype(np.int32),
        "i16": rng.integers(-(2**15), 2**15 - 1, N).astype(np.int16),
        "i8": rng.integers(-128, 127, N).astype(np.int8),
        "f64": rng.random(N),
        "f32": rng.random(N).astype(np.float32),
        "flag": rng.integers(0, 2, N).astype(bool),
        "name": np.array(["x", "yy", "zzz", "wwww"])[
            rng.integers(0, 4, N)],
        "nullable_v": rng.random(N),
    }
    t = pa.table({**{k: pa.array(v) for k, v in cols.items()
                     if k != "nullable_v"},
                  "nullable_v": pa.array(cols["nullable_v"],
                                         mask=~nmask)})
    pq.write_table(t.slice(0, N // 2), str(d / "part-0.parquet"))
    pq.write_table(t.slice(N // 2), str(d / "part-1.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    pdf = pd.DataFrame({**{k: v for k, v in cols.items()},
                        "valid": nmask})
    return session, hs.Hyperspace(session), \
        session.read_parquet(str(d)), pdf


def test_wide_covering_index(env):
    session, h, df, pdf = env
    h.create_index(df, hs.CoveringIndexConfig(
        "wix", ["key"],
        ["i32", "i16", "i8", "f64", "f32", "flag", "name",
         "nullable_v"]))
    session.enable_hyperspace()
    q = df.filter("key = 42").select(
        "key", "i32", "i16", "i8", "f64", "f32", "flag", "name",
        "nullable_v")
    plan = q.optimized_plan()
    assert any(isinstance(l, IndexScan) for l in plan.collect_leaves())
    out = q.collect()
    exp = pdf[pdf.key == 42]
    assert out.num_rows == len(exp)
    # every dtype round-trips
    for c, rounding in [("i32", None), ("i16", None), ("i8", None),
                        ("f64", 9), ("f32", 4)]:
        got = out.tensor(c).numpy()
        want = exp[c].to_numpy()
        if rounding:
            got, want = np.round(got, rounding), np.round(want, rounding)
        assert sorted(got.tolist()) == sorted(want.tolist()), c
    assert sorted(out.column("name").to_numpy().tolist()) == \
        sorted(exp.name.tolist())
    # nullable column kept its mask through the index
    m = out.mask("nullable_v")
    n_null_exp = int((~exp.valid).sum())
    if n_null_exp:
        assert m is not None and int((~m).sum()) == n_null_exp


def test_wide_refresh_and_optimize(env, tmp_path):
    session, h, df, pdf = env
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h.create_index(df, hs.CoveringIndexConfig(
        "wr", ["key"], ["f64", "name", "nullable_v"]))
    # append a third file, refresh incrementally, compact
    rng = np.random.default_rng(78)
    src = df.plan.collect_leaves()[0].relation.root_paths[0]
    pq.write_table(
        pa.table({
            "key": rng.integers(0, 500, 1000),
            "i32": rng.integers(0, 10, 1000).astype(np.int32),
            "i16": rng.integers(0, 10, 1000).astype(np.int16),
            "i8": rng.integers(0, 10, 1000).astype(np.int8),
            "f64": rng.random(1000),
            "f32": rng.random(1000).astype(np.float32),
            "flag": rng.integers(0, 2, 1000).astype(bool),
            "name": ["x"] * 1000,
            "nullable_v": rng.random(1000),
        }), src + "/part-9.parquet")
    h.refresh_index("wr", mode="incremental")
    h.optimize_index("wr", mode="full")
    session.enable_hyperspace()
    out = df.filter("key = 7").select("key", "f64", "name").collect()
    want = int((pdf.key == 7).sum())
    t9 = pq.read_table(src + "/part-9.parquet")
    want += sum(1 for k in t9.column("key").to_pylist() if k == 7)
    assert out.num_rows == want


def test_timestamp_column_index(tmp_path, monkeypatch):
    """Timestamp columns map to int64 epochs in the columnar model and
    index/sort/filter like any numeric column (reference indexes Spark
    TimestampType natively)."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(81)
    d = tmp_path / "ts"
    d.mkdir()
    base = np.datetime64("2025-01-01T00:00:00", "us")
    offs = rng.integers(0, 365 * 24 * 3600, 10_000).astype("timedelta64[s]")
    ts = base + offs
    v = rng.random(10_000)
    pq.write_table(pa.table({"ts": ts, "v": v}),
                   str(d / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("tsx", ["ts"], ["v"]))
    session.enable_hyperspace()
    target = int(ts[0].astype("datetime64[us]").astype(np.int64))
    out = df.filter(f"ts = {target}").select("ts", "v").collect()
    want = int((ts.astype("datetime64[us]").astype(np.int64)
                == target).sum())
    assert out.num_rows == want and want >= 1
