"""HIP kernel numerics tests vs the CPU torch reference (oracle).

Run on a GPU box: python -m pytest tests/test_ops_gpu.py -m gpu -x -q
"""

import numpy as np
import pytest
import torch

import hyperspace_amd.ops as ops
from hyperspace_amd.ops import cpu_ref, native

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs GPU")


@pytest.fixture(scope="module", autouse=True)
def _require_native():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert native.available(), "HIP extension must load on a GPU box"


def _rand_i64(n, lo=-2**62, hi=2**62, seed=0):
    rng = np.random.default_rng(seed)
    return torch.from_numpy(rng.integers(lo, hi, n, dtype=np.int64))


def test_murmur3_bucket_matches_cpu():
    for n in (1, 63, 64, 1000, 1_000_000):
        keys = _rand_i64(n, seed=n)
        cpu = cpu_ref.murmur3_bucket([keys], 200)
        gpu = ops.murmur3_bucket([keys.cuda()], 200).cpu()
        assert torch.equal(cpu, gpu), n


def test_murmur3_bucket_multi_col():
    a = _rand_i64(10_000, seed=1)
    b = torch.from_numpy(
        np.random.default_rng(2).integers(0, 100, 10_000,
                                          dtype=np.int32).astype(np.int32))
    cpu = cpu_ref.murmur3_bucket([a, b], 128)
    gpu = ops.murmur3_bucket([a.cuda(), b.cuda()], 128).cpu()
    assert torch.equal(cpu, gpu)


def test_normalize_key_matches_cpu():
    for dtype in (torch.int64, torch.int32, torch.float64, torch.float32):
        if dtype.is_floating_point:
            vals = torch.randn(100_000, dtype=torch.float64).to(dtype)
        else:
            vals = _rand_i64(100_000).to(dtype)
        cpu = cpu_ref.normalize_key(vals)
        gpu = ops.normalize_key(vals.cuda()).cpu()
        assert torch.equal(cpu, gpu), dtype


@pytest.mark.parametrize("n", [1, 2, 255, 256, 257, 100_000, 3_000_000])
def test_radix_sort_matches_cpu(n):
    keys = cpu_ref.normalize_key(_rand_i64(n, seed=n % 97))
    payload = torch.arange(n, dtype=torch.int64)
    ck, cp = cpu_ref.stable_sort_u64(keys, payload)
    gk, gp = ops.sort_pairs(keys.cuda(), payload.cuda())
    assert torch.equal(ck, gk.cpu())
    assert torch.equal(cp, gp.cpu())


def test_radix_sort_stability():
    # many duplicate keys: payload order must be preserved per key
    rng = np.random.default_rng(5)
    keys = cpu_ref.normalize_key(
        torch.from_numpy(rng.integers(0, 50, 500_000, dtype=np.int64)))
    payload = torch.arange(500_000, dtype=torch.int64)
    ck, cp = cpu_ref.stable_sort_u64(keys, payload)
    gk, gp = ops.sort_pairs(keys.cuda(), payload.cuda())
    assert torch.equal(cp, gp.cpu())


def test_radix_sort_small_range_pass_skipping():
    # keys in [0, 255]: only one nibble varies after normalization offset —
    # exercises the constant-nibble skip path
    keys = cpu_ref.normalize_key(
        torch.from_numpy(np.random.default_rng(6).integers(
            0, 256, 100_000, dtype=np.int64)))
    payload = torch.arange(100_000, dtype=torch.int64)
    ck, cp = cpu_ref.stable_sort_u64(keys, payload)
    gk, gp = ops.sort_pairs(keys.cuda(), payload.cuda())
    assert torch.equal(ck, gk.cpu())
    assert torch.equal(cp, gp.cpu())


@pytest.mark.parametrize("lo,hi", [
    (0, 1 << 26),            # bench shape: 26 bits -> three 9-bit passes
    (1 << 8, 1 << 17),       # bits 8..16 -> one 9-bit pass
    (0, 1 << 18),            # 18 bits -> two 9-bit passes
])
def test_radix_sort_nine_bit_planner(lo, hi):
    """Key shapes where the 9-bit pass planner beats byte passes must
    still sort stably and bit-identically to the CPU oracle."""
    rng = np.random.default_rng(lo % 13 + hi % 7)
    keys = cpu_ref.normalize_key(torch.from_numpy(
        rng.integers(lo, hi, 2_000_000, dtype=np.int64)))
    payload = torch.arange(2_000_000, dtype=torch.int64)
    ck, cp = cpu_ref.stable_sort_u64(keys, payload)
    gk, gp = ops.sort_pairs(keys.cuda(), payload.cuda())
    assert torch.equal(ck, gk.cpu())
    assert torch.equal(cp, gp.cpu())


def test_merge_join_matches_cpu():
    rng = np.random.default_rng(7)
    nseg = 16
    lk, rk, lseg, rseg = [], [], [0], [0]
    for s in range(nseg):
        ln = int(rng.integers(0, 2000))
        rn = int(rng.integers(0, 1500))
        lk.append(np.sort(rng.integers(0, 500, ln)))
        rk.append(np.sort(rng.integers(0, 500, rn)))
        lseg.append(lseg[-1] + ln)
        rseg.append(rseg[-1] + rn)
    lkeys = cpu_ref.normalize_key(
        torch.from_numpy(np.concatenate(lk).astype(np.int64)))
    rkeys = cpu_ref.normalize_key(
        torch.from_numpy(np.concatenate(rk).astype(np.int64)))
    lseg_t = torch.tensor(lseg, dtype=torch.int64)
    rseg_t = torch.tensor(rseg, dtype=torch.int64)
    cl, cr = cpu_ref.merge_join(lkeys, rkeys, lseg_t, rseg_t)
    gl, gr = ops.merge_join(lkeys.cuda(), rkeys.cuda(), lseg_t, rseg_t)
    # same pair sets (order may differ within equal-key runs; ours matches
    # exactly since both iterate left rows in order)
    assert torch.equal(cl, gl.cpu())
    assert torch.equal(cr, gr.cpu())


def test_select_range_matches_cpu():
    vals = _rand_i64(2_000_000, lo=-1000, hi=1000, seed=8)
    keys = cpu_ref.normalize_key(vals)
    lo = int(cpu_ref.normalize_key(torch.tensor([-500]))[0])
    hi = int(cpu_ref.normalize_key(torch.tensor([500]))[0])
    for li, hi_incl in [(True, True), (False, False), (True, False)]:
        cpu = cpu_ref.select_range_u64(keys, lo, hi, li, hi_incl)
        gpu = ops.select_range_u64(keys.cuda(), lo, hi, li, hi_incl).cpu()
        assert torch.equal(cpu, gpu)


def test_isin_sorted_matches_cpu():
    vals = _rand_i64(1_000_000, lo=0, hi=10_000, seed=9)
    s = torch.unique(_rand_i64(100, lo=0, hi=10_000, seed=10))
    cpu = cpu_ref.isin_sorted(vals, s)
    gpu = ops.isin_sorted(vals.cuda(), s.cuda()).cpu()
    assert torch.equal(cpu, gpu)


def test_segmented_minmax_matches_cpu():
    vals = _rand_i64(500_000, seed=11)
    seg = torch.tensor([0, 100, 100, 250_000, 500_000], dtype=torch.int64)
    cmin, cmax = cpu_ref.segmented_minmax(vals, seg)
    gmin, gmax = ops.segmented_minmax(vals.cuda(), seg)
    assert torch.equal(cmin, gmin.cpu())
    assert torch.equal(cmax, gmax.cpu())


def test_bloom_matches_cpu():
    vals = _rand_i64(100_000, lo=0, hi=10**9, seed=12)
    m_bits, k = 1 << 20, 5
    cpu_words = cpu_ref.bloom_build(vals, m_bits, k)
    gpu_words = ops.bloom_build(vals.cuda(), m_bits, k).cpu()
    assert torch.equal(cpu_words, gpu_words)
    probe = _rand_i64(100_000, lo=0, hi=10**9, seed=13)
    cpu_hit = cpu_ref.bloom_probe(probe, cpu_words, m_bits, k)
    gpu_hit = ops.bloom_probe(probe.cuda(), gpu_words.cuda(), m_bits,
                              k).cpu()
    assert torch.equal(cpu_hit, gpu_hit)


def test_zorder_matches_cpu():
    a = cpu_ref.normalize_key(_rand_i64(200_000, seed=14))
    b = cpu_ref.normalize_key(_rand_i64(200_000, seed=15))
    cpu = cpu_ref.zorder_key([a, b], 16)
    gpu = ops.zorder_key([a.cuda(), b.cuda()], 16).cpu()
    assert torch.equal(cpu, gpu)


def test_gather_matches_cpu():
    vals = _rand_i64(1_000_000, seed=16)
    idx = torch.from_numpy(np.random.default_rng(17).integers(
        0, 1_000_000, 300_000, dtype=np.int64))
    cpu = vals[idx]
    gpu = ops.gather_rows(vals.cuda(), idx.cuda()).cpu()
    assert torch.equal(cpu, gpu)
    f32 = torch.randn(1_000_000)
    assert torch.equal(f32[idx], ops.gather_rows(f32.cuda(),
                                                 idx.cuda()).cpu())


def test_device_parquet_decode(tmp_path):
    """K1 device decode: file bytes -> HBM -> unaligned-copy kernel must
    equal the host decode."""
    import numpy as np
    from hyperspace_amd.sources.native_parquet import write_parquet_native
    from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                                   read_files_batch_device)
    rng = np.random.default_rng(23)
    cols = {"key": rng.integers(0, 10**9, 300_000),
            "val": rng.random(300_000),
            "i": rng.integers(0, 100, 300_000).astype(np.int32)}
    p = str(tmp_path / "n.parquet")
    write_parquet_native(cols, p)
    host, hc = read_files_batch([p])
    dev, dc = read_files_batch_device([p], "cuda")
    assert hc == dc
    for name in cols:
        assert torch.equal(host.tensor(name), dev.tensor(name).cpu()), name


def test_device_parquet_decode_pyarrow_file(tmp_path):
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                                   read_files_batch_device)
    rng = np.random.default_rng(24)
    cols = {"key": rng.integers(0, 10**9, 100_000),
            "val": rng.random(100_000)}
    p = str(tmp_path / "pa.parquet")
    pq.write_table(pa.table(cols), p, compression="NONE",
                   use_dictionary=False, data_page_version="1.0")
    host, _ = read_files_batch([p])
    dev, _ = read_files_batch_device([p], "cuda")
    for name in cols:
        assert torch.equal(host.tensor(name), dev.tensor(name).cpu()), name


def test_select_range_one_sided_sentinels():
    """Regression: one-sided ranges (>=, <) with u64-order sentinels —
    u64 0 encodes as int64 0 and u64 max as int64 -1."""
    vals = _rand_i64(500_000, lo=-10_000, hi=10_000, seed=42)
    keys = cpu_ref.normalize_key(vals)
    v = int(cpu_ref.normalize_key(torch.tensor([5000]))[0])
    got = ops.select_range_u64(keys.cuda(), v, -1, True, True).cpu()
    assert got.numel() == int((vals >= 5000).sum())
    got = ops.select_range_u64(keys.cuda(), 0, v, True, False).cpu()
    assert got.numel() == int((vals < 5000).sum())


def test_device_dictionary_decode(tmp_path):
    """K1 dictionary path: pyarrow's default dictionary-encoded pages
    decode on device (RLE/bit-packed runs + dictionary gather)."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from hyperspace_amd.sources.parquet_io import read_files_batch_device
    rng = np.random.default_rng(33)
    cols = {
        "low_card": rng.integers(0, 7, 300_000),          # bw ~3
        "mid_card": rng.integers(0, 40_000, 300_000),     # bw ~16
        "fval": np.round(rng.random(300_000) * 100, 1),   # float dict
    }
    p = str(tmp_path / "dict.parquet")
    pq.write_table(pa.table(cols), p, compression="NONE")  # dict default
    md = pq.ParquetFile(p).metadata.row_group(0)
    assert any("DICTIONARY" in e for e in md.column(0).encodings)
    dev, counts = read_files_batch_device([p], "cuda")
    ref = pq.read_table(p)
    for name in cols:
        got = dev.tensor(name).cpu().numpy()
        exp = ref.column(name).to_numpy()
        assert np.array_equal(got, exp), name


def test_device_dictionary_decode_repeat_runs(tmp_path):
    """Long repeated runs (sorted low-cardinality data -> RLE repeats)."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from hyperspace_amd.sources.parquet_io import read_files_batch_device
    vals = np.repeat(np.arange(20, dtype=np.int64), 50_000)  # 1M rows
    p = str(tmp_path / "runs.parquet")
    pq.write_table(pa.table({"v": vals}), p, compression="NONE")
    dev, _ = read_files_batch_device([p], "cuda")
    assert np.array_equal(dev.tensor("v").cpu().numpy(), vals)


def test_device_decode_multi_rowgroup_dict(tmp_path):
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from hyperspace_amd.sources.parquet_io import read_files_batch_device
    rng = np.random.default_rng(44)
    cols = {"key": rng.integers(0, 5000, 1_500_000),
            "val": rng.random(1_500_000)}
    p = str(tmp_path / "mrgd.parquet")
    pq.write_table(pa.table(cols), p, compression="NONE",
                   row_group_size=400_000)  # dict + 4 row groups
    dev, counts = read_files_batch_device([p], "cuda")
    assert counts == [1_500_000]
    ref = pq.read_table(p)
    for name in cols:
        assert np.array_equal(dev.tensor(name).cpu().numpy(),
                              ref.column(name).to_numpy()), name


def test_run_merge_perm_device_matches_cpu():
    rng = np.random.default_rng(23)
    seg_sizes = [(1000, 1000), (0, 500), (700, 0), (1, 2999)]
    seg = [0]
    split = []
    parts = []
    for na, nb in seg_sizes:
        parts.append(np.sort(rng.integers(-(10**9), 10**9, na)))
        parts.append(np.sort(rng.integers(-(10**9), 10**9, nb)))
        split.append(seg[-1] + na)
        seg.append(seg[-1] + na + nb)
    raw = torch.tensor(np.concatenate(parts), dtype=torch.int64)
    keys = ops.cpu_ref.normalize_key(raw)
    seg_t, split_t = torch.tensor(seg), torch.tensor(split)
    cpu_perm = ops.cpu_ref.run_merge_perm(keys, seg_t, split_t)
    dev_perm = ops.run_merge_perm(keys.cuda(), seg_t, split_t).cpu()
    assert torch.equal(cpu_perm, dev_perm)
