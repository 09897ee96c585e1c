"""CPU reference op tests (these same functions are the oracles for the
HIP kernels in tests/test_ops_gpu.py)."""

import numpy as np
import pytest
import torch

from hyperspace_amd.ops import cpu_ref


def test_murmur3_spark_vectors():
    # Golden values from Spark's Murmur3_x86_32 with seed 42:
    # spark.sql("select hash(0)") = 933211791, hash(1) = -559580957,
    # hash(cast(0 as bigint)) = -1670924195 — our implementation is
    # bit-exact with Spark's HashPartitioning so bucket layouts match.
    h = cpu_ref.murmur3_hash_int32(torch.tensor([0, 1, 42], dtype=torch.int32),
                                   cpu_ref.SPARK_HASH_SEED)
    signed = torch.where(h > 0x7FFFFFFF, h - (1 << 32), h)
    assert signed.tolist() == [933211791, -559580957, 29417773]
    hl = cpu_ref.murmur3_hash_int64(torch.tensor([0], dtype=torch.int64),
                                    cpu_ref.SPARK_HASH_SEED)
    sl = torch.where(hl > 0x7FFFFFFF, hl - (1 << 32), hl)
    assert sl.tolist() == [-1670924195]


def test_murmur3_bucket_range_and_determinism():
    keys = torch.arange(100000, dtype=torch.int64)
    b = cpu_ref.murmur3_bucket([keys], 200)
    assert b.dtype == torch.int32
    assert int(b.min()) >= 0 and int(b.max()) < 200
    # deterministic
    b2 = cpu_ref.murmur3_bucket([keys], 200)
    assert torch.equal(b, b2)
    # roughly uniform
    counts = torch.bincount(b.to(torch.int64), minlength=200).float()
    assert counts.min() > counts.mean() * 0.7


def test_murmur3_multi_column_fold():
    a = torch.tensor([1, 2, 3], dtype=torch.int64)
    c = torch.tensor([4, 5, 6], dtype=torch.int32)
    b1 = cpu_ref.murmur3_bucket([a, c], 100)
    b2 = cpu_ref.murmur3_bucket([a, c], 100)
    b3 = cpu_ref.murmur3_bucket([c, a], 100)
    assert torch.equal(b1, b2)
    assert not torch.equal(b1, b3)  # order matters


def test_normalize_key_order_preserving():
    for dtype in (torch.int64, torch.int32, torch.float64, torch.float32):
        if dtype.is_floating_point:
            vals = torch.tensor([-1e30, -3.5, -0.0, 0.0, 1e-9, 7.25, 1e30],
                                dtype=dtype)
        else:
            vals = torch.tensor([-2**31, -5, 0, 3, 2**31 - 1], dtype=dtype)
        norm = cpu_ref.normalize_key(vals)
        s = cpu_ref._as_unsigned_sortable(norm)
        assert torch.all(s[1:] >= s[:-1]), dtype


def test_stable_sort():
    keys = torch.tensor([3, 1, 3, 2, 1], dtype=torch.int64)
    norm = cpu_ref.normalize_key(keys)
    payload = torch.arange(5, dtype=torch.int64)
    sk, sp = cpu_ref.stable_sort_u64(norm, payload)
    assert sp.tolist() == [1, 4, 3, 0, 2]  # stability: 1@1 before 1@4


def test_merge_join_segmented():
    # two segments; only same-segment rows join
    lkeys = torch.tensor([1, 2, 2, 1, 3], dtype=torch.int64)
    rkeys = torch.tensor([2, 2, 3, 3], dtype=torch.int64)
    lseg = torch.tensor([0, 3, 5])
    rseg = torch.tensor([0, 2, 4])
    li, ri = cpu_ref.merge_join(lkeys, rkeys, lseg, rseg)
    pairs = set(zip(li.tolist(), ri.tolist()))
    # segment 0: l rows [1,2,2] vs r rows [2,2] -> (1,0),(1,1),(2,0),(2,1)
    # segment 1: l rows [1,3] vs r rows [3,3] -> (4,2),(4,3)
    assert pairs == {(1, 0), (1, 1), (2, 0), (2, 1), (4, 2), (4, 3)}


def test_merge_join_matches_numpy_join():
    rng = np.random.default_rng(1)
    lk = np.sort(rng.integers(0, 500, 2000))
    rk = np.sort(rng.integers(0, 500, 1500))
    lseg = torch.tensor([0, 2000])
    rseg = torch.tensor([0, 1500])
    li, ri = cpu_ref.merge_join(torch.from_numpy(lk), torch.from_numpy(rk),
                                lseg, rseg)
    # expected cardinality = sum over key of cnt_l*cnt_r
    import collections
    cl = collections.Counter(lk.tolist())
    cr = collections.Counter(rk.tolist())
    expected = sum(cl[k] * cr.get(k, 0) for k in cl)
    assert li.numel() == expected
    assert np.array_equal(lk[li.numpy()], rk[ri.numpy()])


def test_select_range():
    keys = torch.tensor([5, 1, 9, 3, 7], dtype=torch.int64)
    norm = cpu_ref.normalize_key(keys)
    lo = int(cpu_ref.normalize_key(torch.tensor([3], dtype=torch.int64))[0])
    hi = int(cpu_ref.normalize_key(torch.tensor([7], dtype=torch.int64))[0])
    idx = cpu_ref.select_range_u64(norm, lo, hi, True, True)
    assert sorted(keys[idx].tolist()) == [3, 5, 7]
    idx = cpu_ref.select_range_u64(norm, lo, hi, False, False)
    assert keys[idx].tolist() == [5]


def test_isin_sorted():
    vals = torch.tensor([5, 1, 9, 3, 7, 5], dtype=torch.int64)
    s = torch.tensor([3, 5], dtype=torch.int64)
    mask = cpu_ref.isin_sorted(vals, s)
    assert mask.tolist() == [True, False, False, True, False, True]
    empty = torch.empty(0, dtype=torch.int64)
    assert not cpu_ref.isin_sorted(vals, empty).any()


def test_segmented_minmax():
    vals = torch.tensor([4, 2, 9, 1, 7, 3], dtype=torch.int64)
    seg = torch.tensor([0, 3, 3, 6])
    mins, maxs = cpu_ref.segmented_minmax(vals, seg)
    assert mins.tolist() == [2, 0, 1]
    assert maxs.tolist() == [9, 0, 7]


def test_bloom_no_false_negatives():
    vals = torch.arange(0, 1000, dtype=torch.int64)
    words = cpu_ref.bloom_build(vals, m_bits=16384, k=5)
    assert cpu_ref.bloom_probe(vals, words, 16384, 5).all()
    # false positive rate low for disjoint probes
    probe = torch.arange(100000, 110000, dtype=torch.int64)
    fp = cpu_ref.bloom_probe(probe, words, 16384, 5).float().mean()
    assert fp < 0.05


def test_zorder_interleave_exact():
    # columns pre-scaled into the TOP bits (the zorder build min-max
    # scales values before interleaving)
    def top_bits(v, nbits=2):
        # place the nbits-wide value at the top of the u64 key space
        return torch.tensor(v, dtype=torch.int64) << (64 - nbits)

    x = top_bits([0b10, 0b01])
    y = top_bits([0b01, 0b11])
    z = cpu_ref.zorder_key([x, y], bits_per_col=2)
    # interleave MSB-first: x1 y1 x0 y0
    expected0 = 0b1001 << 60
    expected1 = 0b0111 << 60
    assert (z[0] & ((1 << 63) * 2 - 1)) >> 60 != 0  # sanity: top nibble set
    assert (int(z[0]) >> 60) & 0xF == expected0 >> 60
    assert (int(z[1]) >> 60) & 0xF == expected1 >> 60


def test_zorder_locality():
    # close (x,y) pairs get closer z-addresses than far pairs once values
    # are scaled into the top bits
    def scaled(vals, lo, hi, bits=16):
        t = torch.tensor(vals, dtype=torch.float64)
        s = ((t - lo) / (hi - lo) * ((1 << bits) - 1)).to(torch.int64)
        return s << (64 - bits)

    xs = scaled([100, 101, 60000], 0, 65535)
    ys = scaled([200, 200, 60000], 0, 65535)
    z = cpu_ref.zorder_key([xs, ys], bits_per_col=16)
    d01 = abs(int(z[1]) - int(z[0]))
    d02 = abs(int(z[2]) - int(z[0]))
    assert d01 < d02


def test_run_merge_perm_matches_sort():
    """K4b segmented two-run merge == stable per-segment sort."""
    import numpy as np
    from hyperspace_amd import ops
    rng = np.random.default_rng(17)
    seg_sizes = [0, 10, 1, 177, 64]
    seg = torch.tensor(np.concatenate([[0], np.cumsum(
        [a + b for a, b in zip(seg_sizes, seg_sizes[::-1])])]))
    keys_parts, split = [], []
    pos = 0
    for na, nb in zip(seg_sizes, seg_sizes[::-1]):
        keys_parts.append(np.sort(rng.integers(-50, 50, na)))
        keys_parts.append(np.sort(rng.integers(-50, 50, nb)))
        split.append(pos + na)
        pos += na + nb
    raw = torch.tensor(np.concatenate(keys_parts), dtype=torch.int64)
    keys = ops.normalize_key(raw)
    perm = ops.run_merge_perm(keys, seg, torch.tensor(split))
    merged = raw[perm]
    for s_i in range(seg.numel() - 1):
        a, b = int(seg[s_i]), int(seg[s_i + 1])
        part = merged[a:b]
        assert (part[1:] >= part[:-1]).all()
        # same multiset
        assert sorted(part.tolist()) == sorted(raw[a:b].tolist())
    # stability: A rows (idx < split) precede equal B rows
    for s_i in range(seg.numel() - 1):
        a, b = int(seg[s_i]), int(seg[s_i + 1])
        sp = split[s_i]
        p = perm[a:b]
        k = raw[p]
        for j in range(1, b - a):
            if k[j] == k[j - 1]:
                # equal keys: A-origin must not follow B-origin
                assert not (p[j] < sp <= p[j - 1]), (s_i, j)
