"""Golden-file plan stability (reference: goldstandard/PlanStabilitySuite
ported from Spark — normalized plan strings compared against checked-in
goldens; regenerate with HYPERSPACE_GENERATE_GOLDEN_FILES=1)."""

import os
import re

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "goldens")
GENERATE = os.environ.get("HYPERSPACE_GENERATE_GOLDEN_FILES") == "1"


def normalize(plan_str: str) -> str:
    """Strip run-dependent detail: absolute paths, uuids, file counts."""
    s = re.sub(r"(parquet|delta|iceberg):[^,)\s@]*", r"\1:<PATH>", plan_str)
    s = re.sub(r"@v\d+|@snap\d+", "@<VER>", s)
    s = re.sub(r"-\d+files", "-<N>files", s)
    return s


@pytest.fixture(scope="module")
def env(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("stab")
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp / "indexes")
    rng = np.random.default_rng(101)
    for name, cols in [("fact", {"key": rng.integers(0, 1000, 8000),
                                 "qty": rng.integers(0, 50, 8000),
                                 "val": rng.random(8000)}),
                       ("dim", {"key": np.arange(1000, dtype=np.int64),
                                "status": rng.integers(0, 5, 1000)})]:
        d = tmp / name
        d.mkdir()
        pq.write_table(pa.table(cols), str(d / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    fact = session.read_parquet(str(tmp / "fact"))
    dim = session.read_parquet(str(tmp / "dim"))
    h.create_index(fact, hs.CoveringIndexConfig("s_fidx", ["qty"], ["val"]))
    h.create_index(fact, hs.CoveringIndexConfig("s_jl", ["key"],
                                                ["qty", "val"]))
    h.create_index(dim, hs.CoveringIndexConfig("s_jr", ["key"], ["status"]))
    h.create_index(fact, hs.DataSkippingIndexConfig(
        "s_ds", hs.MinMaxSketch("val")))
    session.enable_hyperspace()
    return session, fact, dim


QUERIES = {
    "q1_filter_eq": lambda f, d: f.filter("qty = 7").select("qty", "val"),
    "q2_filter_range": lambda f, d: f.filter("qty >= 40")
        .select("qty", "val"),
    "q3_join": lambda f, d: f.select("key", "val")
        .join(d.select("key", "status"), on="key"),
    "q4_filter_unindexed": lambda f, d: f.filter("val <= 0.001")
        .select("val"),
    "q5_join_then_nothing_matches": lambda f, d: f.select("key", "qty")
        .join(d.select("key", "status"), on="key"),
}


@pytest.mark.parametrize("name", sorted(QUERIES))
def test_plan_stability(env, name):
    session, fact, dim = env
    q = QUERIES[name](fact, dim)
    plan = normalize(q.optimized_plan().pretty())
    golden_path = os.path.join(GOLDEN_DIR, f"{name}.txt")
    if GENERATE:
        os.makedirs(GOLDEN_DIR, exist_ok=True)
        with open(golden_path, "w") as f:
            f.write(plan + "\n")
        pytest.skip("golden regenerated")
    assert os.path.exists(golden_path), (
        f"missing golden {golden_path}; regenerate with "
        "HYPERSPACE_GENERATE_GOLDEN_FILES=1")
    with open(golden_path) as f:
        expected = f.read().rstrip("\n")
    assert plan == expected, (
        f"plan drifted for {name}:\n--- got ---\n{plan}\n"
        f"--- golden ---\n{expected}")


def test_explain_golden(env):
    """Explain output golden (reference plananalysis/ExplainTest against
    resources/expected/) — paths and version dirs normalized."""
    session, fact, dim = env
    import hyperspace_amd as hs
    h = hs.Hyperspace(session)
    q = fact.filter("qty = 7").select("qty", "val")
    out = h.explain(q, verbose=True)
    out = re.sub(r"(parquet|delta|iceberg):[^,)\s@]*", r"\1:<PATH>", out)
    out = re.sub(r":/[^\s]*/indexes/", ":<SYS>/", out)
    golden_path = os.path.join(GOLDEN_DIR, "explain_filter.txt")
    if GENERATE:
        with open(golden_path, "w") as f:
            f.write(out + "\n")
        pytest.skip("golden regenerated")
    with open(golden_path) as f:
        expected = f.read().rstrip("\n")
    assert out == expected


def test_hybrid_delete_plan_stability(tmp_path):
    """Hybrid Scan over DELETED source files: the plan excludes the
    deleted file's rows via the lineage filter, shown as the index
    scan's -Nfiles annotation (reference:
    CoveringIndexRuleUtils.scala:247-252 lineage NOT-IN filter)."""
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp_path / "ix")
    rng = np.random.default_rng(5)
    d = tmp_path / "src"
    d.mkdir()
    for i in range(8):
        pq.write_table(pa.table({"key": rng.integers(0, 500, 4000),
                                 "val": rng.random(4000)}),
                       str(d / f"p{i}.parquet"))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    session.conf.set(hs.IndexConstants.INDEX_LINEAGE_ENABLED, True)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("hdel", ["key"], ["val"]))
    (d / "p7.parquet").unlink()
    session.conf.set(hs.IndexConstants.INDEX_HYBRID_SCAN_ENABLED, True)
    session.enable_hyperspace()
    q = session.read_parquet(str(d)).filter("key = 77") \
        .select("key", "val")
    plan = normalize(q.optimized_plan().pretty())
    assert plan == ("Project(['key', 'val'])\n"
                    "  Filter((key = 77))\n"
                    "    IndexScan(hdel, -<N>files)"), plan
