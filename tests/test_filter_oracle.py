"""Query-level filter oracle: expected row counts computed independently
with numpy (NOT by comparing to the engine's own unindexed run — the
range-sentinel bug showed equivalence tests are blind to shared-path
evaluator errors)."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs

N = 30_000


@pytest.fixture(params=["unindexed", "indexed", "indexed_bucketspec"])
def env(tmp_path, monkeypatch, request):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "indexes"))
    data = tmp_path / "data"
    data.mkdir()
    rng = np.random.default_rng(111)
    key = rng.integers(-5000, 5000, N)
    qty = rng.integers(0, 100, N).astype(np.int32)
    price = rng.random(N) * 200 - 100
    t = pa.table({"key": key, "qty": qty, "price": price})
    pq.write_table(t, str(data / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(data))
    if request.param != "unindexed":
        h.create_index(df, hs.CoveringIndexConfig(
            "oix", ["key"], ["qty", "price"]))
        session.enable_hyperspace()
        if request.param == "indexed_bucketspec":
            session.conf.set(
                hs.IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC, True)
    return session, df, {"key": key, "qty": qty, "price": price}


CASES = [
    ("key = 777", lambda c: c["key"] == 777),
    ("key != 777", lambda c: c["key"] != 777),
    ("key < -4000", lambda c: c["key"] < -4000),
    ("key <= 0", lambda c: c["key"] <= 0),
    ("key > 4000", lambda c: c["key"] > 4000),
    ("key >= 4999", lambda c: c["key"] >= 4999),
    ("key >= -5000", lambda c: c["key"] >= -5000),   # full range
    ("key < -5000", lambda c: c["key"] < -5000),     # empty
    ("qty >= 90", lambda c: c["qty"] >= 90),
    ("qty < 10", lambda c: c["qty"] < 10),
    ("price >= 0.0", lambda c: c["price"] >= 0.0),
    ("price < -99.0", lambda c: c["price"] < -99.0),
    ("key >= 1000 AND key < 2000",
     lambda c: (c["key"] >= 1000) & (c["key"] < 2000)),
    ("key >= 1000 AND qty < 50",
     lambda c: (c["key"] >= 1000) & (c["qty"] < 50)),
]


@pytest.mark.parametrize("pred,oracle", CASES,
                         ids=[c[0] for c in CASES])
def test_filter_counts_match_numpy(env, pred, oracle):
    session, df, cols = env
    expected = int(oracle(cols).sum())
    got = df.filter(pred).select("key", "qty", "price").count()
    assert got == expected, pred


def test_filter_values_match_numpy(env):
    session, df, cols = env
    mask = (cols["key"] >= 1000) & (cols["key"] < 1100)
    expected = sorted(zip(cols["key"][mask].tolist(),
                          cols["qty"][mask].tolist()))
    out = df.filter("key >= 1000 AND key < 1100").select("key", "qty") \
        .collect().to_numpy()
    got = sorted(zip(out["key"].tolist(), out["qty"].tolist()))
    assert got == expected


def test_arith_filter_oracle_with_nulls(tmp_path, monkeypatch):
    """Arithmetic predicates vs a pandas oracle over nullable data:
    null inputs never match, Java-style remainder on negatives."""
    import pandas as pd
    from hyperspace_amd.plan.expr import col
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(37)
    d = tmp_path / "ar"
    d.mkdir()
    k = rng.integers(-1000, 1000, 20000)
    mask = rng.random(20000) > 0.1
    pq.write_table(pa.table({"k": pa.array(k, mask=~mask),
                             "v": rng.random(20000)}),
                   str(d / "p.parquet"), compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
    session = hs.HyperspaceSession(device="cpu")
    df = session.read_parquet(str(d))
    kv = pd.Series(k).where(mask)  # NaN at nulls

    def java_mod(s, m):
        # sign of the dividend (np.fmod), unlike python %
        return np.fmod(s, m)

    cases = [
        ((col("k") % 7) == 3, java_mod(kv, 7) == 3),
        ((col("k") * 2) >= 500, kv * 2 >= 500),
        ((col("k") + 10) < -200, kv + 10 < -200),
        ((col("k") - 1) != 0, (kv - 1).notna() & ((kv - 1) != 0)),
        ((col("k") / 4) <= -100.0, kv / 4 <= -100.0),
        ((col("k") % 9).isin([0, 4]), java_mod(kv, 9).isin([0, 4])),
    ]
    for pred, oracle in cases:
        got = df.filter(pred).collect().num_rows
        want = int(oracle.fillna(False).sum())
        assert got == want, repr(pred)
