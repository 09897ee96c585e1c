"""CPU reference implementations of the data-plane ops.

These are the numerics oracles for the HIP kernels (tests compare the
device kernels against these) and the execution path on machines without a
GPU.  They intentionally use plain torch ops — clarity over speed.

Spark-compatible Murmur3 (seed 42) is implemented here bit-exactly so that
bucket assignment matches Spark's HashPartitioning for int/long columns
(reference data plane K2: index/covering/CoveringIndex.scala:58-61).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

MASK32 = 0xFFFFFFFF
SPARK_HASH_SEED = 42


def _rotl32(x: torch.Tensor, r: int) -> torch.Tensor:
    return ((x << r) | (x >> (32 - r))) & MASK32


def _mix_k1(k1: torch.Tensor) -> torch.Tensor:
    k1 = (k1 * 0xCC9E2D51) & MASK32
    k1 = _rotl32(k1, 15)
    k1 = (k1 * 0x1B873593) & MASK32
    return k1


def _mix_h1(h1: torch.Tensor, k1: torch.Tensor) -> torch.Tensor:
    h1 = h1 ^ k1
    h1 = _rotl32(h1, 13)
    h1 = (h1 * 5 + 0xE6546B64) & MASK32
    return h1


def _fmix(h1: torch.Tensor, length: int) -> torch.Tensor:
    h1 = h1 ^ length
    h1 = h1 ^ (h1 >> 16)
    h1 = (h1 * 0x85EBCA6B) & MASK32
    h1 = h1 ^ (h1 >> 13)
    h1 = (h1 * 0xC2B2AE35) & MASK32
    h1 = h1 ^ (h1 >> 16)
    return h1


def murmur3_hash_int32(x: torch.Tensor, seed) -> torch.Tensor:
    """Spark Murmur3_x86_32 hashInt.  x: any int tensor (uses low 32 bits).
    seed: int or int tensor.  Returns uint32 values in an int64 tensor."""
    v = x.to(torch.int64) & MASK32
    if isinstance(seed, int):
        seed = torch.full_like(v, seed)
    h1 = _mix_h1(seed & MASK32, _mix_k1(v))
    return _fmix(h1, 4)


def murmur3_hash_int64(x: torch.Tensor, seed) -> torch.Tensor:
    """Spark Murmur3 hashLong: mixes low then high 32-bit halves."""
    v = x.to(torch.int64)
    low = v & MASK32
    high = (v >> 32) & MASK32
    if isinstance(seed, int):
        seed = torch.full_like(v, seed)
    h1 = _mix_h1(seed & MASK32, _mix_k1(low))
    h1 = _mix_h1(h1, _mix_k1(high))
    return _fmix(h1, 8)


def murmur3_bucket(keys: List[torch.Tensor], num_buckets: int,
                   masks: Optional[List[Optional[torch.Tensor]]] = None
                   ) -> torch.Tensor:
    """Bucket id per row: Spark HashPartitioning semantics —
    h = seed 42 folded over columns, bucket = pmod(h, n).  A null key
    column leaves the running hash unchanged (Murmur3Hash skips null
    children, so an all-null single-key row lands in pmod(42, n)).
    ``masks[i]`` is a bool validity tensor or None.  Returns int32."""
    h: Optional[torch.Tensor] = None
    for ci, k in enumerate(keys):
        seed = SPARK_HASH_SEED if h is None else h
        if k.dtype in (torch.int64, torch.float64):
            if k.dtype == torch.float64:
                k = k.view(torch.int64)
            nh = murmur3_hash_int64(k, seed)
        else:
            if k.dtype == torch.float32:
                k = k.view(torch.int32)
            nh = murmur3_hash_int32(k, seed)
        mask = masks[ci] if masks is not None else None
        if mask is not None and mask.numel():
            seed_t = (torch.full_like(nh, SPARK_HASH_SEED)
                      if h is None else h)
            nh = torch.where(mask.to(torch.bool), nh, seed_t)
        h = nh
    assert h is not None
    signed = h.to(torch.int64)
    signed = torch.where(signed > 0x7FFFFFFF, signed - (1 << 32), signed)
    return ((signed % num_buckets + num_buckets) % num_buckets).to(torch.int32)


# ---------------------------------------------------------------------------
# Order-preserving u64 key normalization
# ---------------------------------------------------------------------------

def normalize_key(col: torch.Tensor) -> torch.Tensor:
    """Map a column to u64 (stored in int64) such that unsigned comparison
    of the result == natural ordering of the values."""
    if col.dtype in (torch.int8, torch.int16, torch.int32, torch.int64):
        return col.to(torch.int64) ^ (-0x8000000000000000)
    if col.dtype == torch.float64:
        bits = col.view(torch.int64)
        mask = torch.where(bits < 0,
                           torch.full_like(bits, -1),
                           torch.full_like(bits, -0x8000000000000000))
        return bits ^ mask
    if col.dtype == torch.float32:
        return normalize_key(col.to(torch.float64))
    if col.dtype == torch.bool:
        return col.to(torch.int64) ^ (-0x8000000000000000)
    raise ValueError(f"Unsupported dtype for sort key: {col.dtype}")


def _as_unsigned_sortable(u64: torch.Tensor) -> torch.Tensor:
    """int64-storing-u64 -> int64 whose signed order == the u64 order."""
    return u64 ^ (-0x8000000000000000)


def stable_sort_u64(keys_u64: torch.Tensor,
                    payload: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stable sort of (u64 key, payload) by key.  Returns sorted keys and
    correspondingly permuted payload."""
    sortable = _as_unsigned_sortable(keys_u64)
    _, perm = torch.sort(sortable, stable=True)
    return keys_u64[perm], payload[perm]


def sort_perm_u64(keys_u64: torch.Tensor) -> torch.Tensor:
    sortable = _as_unsigned_sortable(keys_u64)
    _, perm = torch.sort(sortable, stable=True)
    return perm


# ---------------------------------------------------------------------------
# Segmented sorted merge join (K4)
# ---------------------------------------------------------------------------

def merge_join(lkeys: torch.Tensor, rkeys: torch.Tensor,
               lseg: torch.Tensor, rseg: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Join two segment-partitioned sorted key arrays.

    ``lseg``/``rseg`` are segment offset arrays of equal length S+1; segment
    s spans [lseg[s], lseg[s+1]).  Keys are sorted within each segment.
    Returns (left_idx, right_idx) global row-index pairs of all equal-key
    matches, segment-aligned (only same-segment rows join — this is the
    zero-exchange co-bucketed join).
    """
    assert lseg.numel() == rseg.numel()
    louts, routs = [], []
    S = lseg.numel() - 1
    for s in range(S):
        l0, l1 = int(lseg[s]), int(lseg[s + 1])
        r0, r1 = int(rseg[s]), int(rseg[s + 1])
        if l1 <= l0 or r1 <= r0:
            continue
        lk = lkeys[l0:l1]
        rk = rkeys[r0:r1]
        lo = torch.searchsorted(rk, lk, side="left")
        hi = torch.searchsorted(rk, lk, side="right")
        counts = (hi - lo).clamp(min=0)
        total = int(counts.sum())
        if total == 0:
            continue
        lidx = torch.repeat_interleave(
            torch.arange(l0, l1, dtype=torch.int64), counts)
        starts = torch.repeat_interleave(lo, counts)
        offs = torch.arange(total, dtype=torch.int64) - \
            torch.repeat_interleave(torch.cumsum(counts, 0) - counts, counts)
        ridx = r0 + starts + offs
        louts.append(lidx)
        routs.append(ridx)
    if not louts:
        e = torch.empty(0, dtype=torch.int64)
        return e, e.clone()
    return torch.cat(louts), torch.cat(routs)


# ---------------------------------------------------------------------------
# Filters (K7 + filter scan)
# ---------------------------------------------------------------------------

def select_range_u64(keys_u64: torch.Tensor, lo: int, hi: int,
                     lo_incl: bool, hi_incl: bool) -> torch.Tensor:
    """Indices i where lo <?< keys[i] <?< hi under u64 ordering.
    lo/hi are normalized u64 (int64-encoded)."""
    s = _as_unsigned_sortable(keys_u64)
    slo = lo ^ -0x8000000000000000
    shi = hi ^ -0x8000000000000000
    m_lo = (s >= slo) if lo_incl else (s > slo)
    m_hi = (s <= shi) if hi_incl else (s < shi)
    return torch.nonzero(m_lo & m_hi, as_tuple=False).flatten()


def isin_sorted(values: torch.Tensor,
                sorted_set: torch.Tensor) -> torch.Tensor:
    """Boolean mask: values ∈ sorted_set (both int64)."""
    if sorted_set.numel() == 0:
        return torch.zeros_like(values, dtype=torch.bool)
    pos = torch.searchsorted(sorted_set, values)
    pos = pos.clamp(max=sorted_set.numel() - 1)
    return sorted_set[pos] == values


# ---------------------------------------------------------------------------
# Sketches (K8)
# ---------------------------------------------------------------------------

def segmented_minmax(vals: torch.Tensor, seg_off: torch.Tensor
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-segment min/max.  seg_off: S+1 offsets.  Returns (mins, maxs)."""
    S = seg_off.numel() - 1
    mins = torch.empty(S, dtype=vals.dtype)
    maxs = torch.empty(S, dtype=vals.dtype)
    for s in range(S):
        a, b = int(seg_off[s]), int(seg_off[s + 1])
        seg = vals[a:b]
        mins[s] = seg.min() if b > a else 0
        maxs[s] = seg.max() if b > a else 0
    return mins, maxs


def _bloom_hashes(vals: torch.Tensor, k: int, m_bits: int) -> torch.Tensor:
    """k bit positions per value, shape [N, k].  Double hashing with the
    Spark-style (h1 + i*h2) scheme over Murmur3."""
    h1 = murmur3_hash_int64(vals, 0)
    h2 = murmur3_hash_int64(vals, h1)
    idx = []
    for i in range(k):
        combined = (h1 + i * h2) & 0x7FFFFFFFFFFFFFFF
        idx.append(combined % m_bits)
    return torch.stack(idx, dim=1)


def bloom_build(vals: torch.Tensor, m_bits: int, k: int) -> torch.Tensor:
    """Build a bloom filter (int64 word array of ceil(m/64) words)."""
    words = torch.zeros((m_bits + 63) // 64, dtype=torch.int64)
    pos = _bloom_hashes(vals.to(torch.int64), k, m_bits).flatten()
    w = (pos // 64).long()
    b = (pos % 64).long()
    np_words = words.numpy()
    import numpy as np
    np.bitwise_or.at(np_words, w.numpy(),
                     (np.int64(1) << b.numpy().astype(np.int64)))
    return torch.from_numpy(np_words)


def bloom_probe(vals: torch.Tensor, words: torch.Tensor, m_bits: int,
                k: int) -> torch.Tensor:
    pos = _bloom_hashes(vals.to(torch.int64), k, m_bits)
    w = (pos // 64).long()
    b = pos % 64
    bits = (words[w] >> b) & 1
    return bits.all(dim=1)


# ---------------------------------------------------------------------------
# Z-order (K10)
# ---------------------------------------------------------------------------

def zorder_key(cols_u64: List[torch.Tensor], bits_per_col: int
               ) -> torch.Tensor:
    """Interleave the top ``bits_per_col`` bits of each normalized column
    into a single u64 z-address.  cols are normalize_key() outputs."""
    n_cols = len(cols_u64)
    assert n_cols * bits_per_col <= 64
    z = torch.zeros_like(cols_u64[0])
    for b in range(bits_per_col):
        # bit (63 - b) of each column, MSB-first round-robin
        for c, col in enumerate(cols_u64):
            bit = (col >> (63 - b)) & 1
            out_pos = 63 - (b * n_cols + c)
            z = z | (bit << out_pos)
    return z


def run_merge_perm(keys_u64: torch.Tensor, seg: torch.Tensor,
                   split: torch.Tensor) -> torch.Tensor:
    """Segmented two-sorted-run merge permutation (K4b reference).

    Each segment b's rows [seg[b], seg[b+1]) hold two sorted runs split
    at split[b] (== seg[b+1] when the segment is a single run); returns
    perm such that gathering by it sorts each segment, with A-rows
    stably preceding equal B-rows."""
    sortable = _as_unsigned_sortable(keys_u64)
    n = keys_u64.numel()
    perm = torch.empty(n, dtype=torch.int64)
    for b in range(seg.numel() - 1):
        a0, b1, sp = int(seg[b]), int(seg[b + 1]), int(split[b])
        ka, kb = sortable[a0:sp], sortable[sp:b1]
        pos_a = torch.arange(sp - a0) + torch.searchsorted(
            kb, ka, right=False)
        pos_b = torch.arange(b1 - sp) + torch.searchsorted(
            ka, kb, right=True)
        perm[a0 + pos_a] = torch.arange(a0, sp)
        perm[a0 + pos_b] = torch.arange(sp, b1)
    return perm
