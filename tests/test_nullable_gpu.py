"""GPU nullable-column path: device parquet decode of OPTIONAL pages,
null-aware bucketing/sort on device kernels, SQL null semantics for
indexed filters and the co-bucketed join, CPU-engine equivalence."""

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.ops import native
from hyperspace_amd.plan.expr import col
from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                               read_files_batch_device)

pytestmark = pytest.mark.gpu

N = 400_000


@pytest.fixture(scope="module", autouse=True)
def _require():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert native.available()


@pytest.fixture
def data(tmp_path):
    rng = np.random.default_rng(33)
    d = tmp_path / "data"
    d.mkdir()
    key = rng.integers(0, 5000, N)
    kmask = rng.random(N) > 0.1
    key[::101] = 0
    val = rng.random(N)
    for i in range(2):
        sl = slice(i * N // 2, (i + 1) * N // 2)
        pq.write_table(
            pa.table({"k": pa.array(key[sl], mask=~kmask[sl]),
                      "v": pa.array(val[sl])}),
            str(d / f"part-{i}.parquet"), compression="NONE",
            use_dictionary=False, data_page_version="1.0")
    return d, key, kmask, val


def test_device_decode_masks(data):
    d, key, kmask, val = data
    paths = sorted(str(p) for p in d.glob("*.parquet"))
    dev_batch, counts = read_files_batch_device(
        paths, torch.device("cuda:0"))
    host_batch, hcounts = read_files_batch(paths)
    assert counts == hcounts
    m = dev_batch.mask("k")
    assert m is not None and m.is_cuda
    assert np.array_equal(m.cpu().numpy(), kmask)
    got = dev_batch.tensor("k").cpu().numpy()
    assert np.array_equal(got[kmask], key[kmask])
    assert dev_batch.mask("v") is None


def test_gpu_null_semantics_match_cpu(data, tmp_path, monkeypatch):
    d, key, kmask, val = data
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    session = hs.HyperspaceSession(device="cuda:0")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 16)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("gn", ["k"], ["v"]))
    session.enable_hyperspace()

    # literal-0 filter must exclude the null rows (stored as 0)
    out = df.filter("k = 0").select("k", "v").collect()
    expected = int((kmask & (key == 0)).sum())
    assert out.num_rows == expected

    out = df.filter(col("k").is_null()).collect()
    assert out.num_rows == int((~kmask).sum())
    out = df.filter(col("k").is_not_null()).collect()
    assert out.num_rows == int(kmask.sum())

    rng_pred = kmask & (key >= 4900)
    out = df.filter("k >= 4900").select("k", "v").collect()
    assert out.num_rows == int(rng_pred.sum())


def test_gpu_join_drops_null_keys(data, tmp_path, monkeypatch):
    d, key, kmask, val = data
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rdir = tmp_path / "right"
    rdir.mkdir()
    pq.write_table(
        pa.table({"k": np.arange(5000, dtype=np.int64),
                  "s": np.arange(5000, dtype=np.int64) % 7}),
        str(rdir / "part-0.parquet"))
    session = hs.HyperspaceSession(device="cuda:0")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 16)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    dim = session.read_parquet(str(rdir))
    h.create_index(df, hs.CoveringIndexConfig("gl", ["k"], ["v"]))
    h.create_index(dim, hs.CoveringIndexConfig("gr", ["k"], ["s"]))
    session.enable_hyperspace()
    out = df.join(dim, on="k").collect()
    assert out.num_rows == int(kmask.sum())
