"""Driver-contract guard: `python bench.py` must print one JSON line
with the agreed fields (the round driver parses this on real MI355X
boxes; a schema regression would invalidate the round's BENCH/SCALE
artifacts)."""

import json
import os
import subprocess
import sys


def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"),
         "--device", "cpu", "--gb-per-gpu", "0.02",
         "--steps", "1", "--warmup", "0",
         "--workdir", str(tmp_path / "w")],
        capture_output=True, text=True, timeout=600, env=env, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"].startswith("index-build GB/s")
    for key in ("value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    for key in ("model", "parallelism", "filter_query_s",
                "join_query_s", "build_s"):
        assert key in cfg, key


def test_bench_forces_inline_cleanup_on_long_runs(tmp_path):
    """Long benchmark runs must vacuum each step's index inline —
    deferring tens of per-step versions to the end exhausts tmpfs/RAM
    (observed killing 40-step GPU runs)."""
    import subprocess
    import sys
    env = dict(os.environ, BENCH_DEBUG="1",
               BENCH_WORKDIR=str(tmp_path / "w"))
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "26",
         "--warmup", "0", "--gb-per-gpu", "0.005",
         "--num-buckets", "4"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        env=env, capture_output=True, text=True, timeout=600)
    assert "inline_cleanup=True" in out.stderr, out.stderr[:2000]
    # and the workdir holds at most a couple of index versions at exit
    import glob
    vdirs = glob.glob(str(tmp_path / "w" / "indexes" / "bench_ix_*"))
    leftover = [d for d in vdirs if os.path.isdir(d)
                and glob.glob(os.path.join(d, "v__=*", "*.parquet"))]
    assert len(leftover) <= 2, leftover
