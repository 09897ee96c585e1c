"""Data-plane ops with CPU/HIP dispatch.

Each op routes CUDA tensors to the hand-written HIP/CDNA4 kernels
(csrc/kernels/*.hip, built as hyperspace_amd._hip) and CPU tensors to the
torch reference implementation (cpu_ref.py).  On CUDA the native extension
is REQUIRED — there is no silent PyTorch fallback (KernelUnavailableError).

Op -> reference data-plane mapping:
  murmur3_bucket   K2  hash partitioning (Spark HashPartitioning-compatible,
                       null keys pass the seed through)
  sort_pairs       K3  per-bucket sort (stable LSD radix on device)
  merge_join       K4  co-bucketed sort-merge join (two-phase per-tile)
  run_merge_perm   K4b segmented two-sorted-run merge (multi-file buckets)
  select_range     K1/filter scan predicate + compaction
  isin_sorted      K7  lineage delete filter
  segmented_minmax K8  data-skipping MinMax sketch
  bloom_build/probe K8 data-skipping BloomFilter sketch
  zorder_key       K10 z-address bit interleave
  (extension-only: copy_unaligned/rle_decode/snappy_decompress — the K1
  parquet page decode surface used by sources/parquet_io.py)
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from . import cpu_ref, native
from .cpu_ref import MASK32, SPARK_HASH_SEED  # re-export

__all__ = [
    "murmur3_bucket", "normalize_key", "sort_pairs", "sort_perm",
    "merge_join", "run_merge_perm", "select_range_u64", "isin_sorted", "segmented_minmax",
    "bloom_build", "bloom_probe", "bloom_probe_many", "zorder_key", "gather_rows", "native",
]


def normalize_key(col: torch.Tensor) -> torch.Tensor:
    if col.is_cuda:
        return native.ext().normalize_key(col.contiguous())
    return cpu_ref.normalize_key(col)


def gather_rows(values: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    if values.is_cuda:
        return native.ext().gather_rows(values.contiguous(),
                                        idx.contiguous())
    return values[idx]


def _is_cuda(*tensors: torch.Tensor) -> bool:
    return any(t.is_cuda for t in tensors)


def murmur3_bucket(keys: List[torch.Tensor], num_buckets: int,
                   masks=None) -> torch.Tensor:
    """masks: optional list parallel to keys; entries are bool validity
    tensors or None (no nulls in that key column)."""
    if _is_cuda(*keys):
        if masks is not None and any(m is not None for m in masks):
            mlist = [
                (m.contiguous() if m is not None
                 else torch.empty(0, dtype=torch.bool,
                                  device=keys[0].device))
                for m in masks]
            return native.ext().murmur3_bucket(list(keys), num_buckets,
                                               mlist)
        return native.ext().murmur3_bucket(list(keys), num_buckets)
    return cpu_ref.murmur3_bucket(keys, num_buckets, masks)


def sort_pairs(keys_u64: torch.Tensor, payload: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stable sort (u64-ordered keys stored as int64, int64 payload)."""
    if _is_cuda(keys_u64):
        return native.ext().radix_sort_pairs(keys_u64, payload)
    return cpu_ref.stable_sort_u64(keys_u64, payload)


def sort_perm(keys_u64: torch.Tensor) -> torch.Tensor:
    """Permutation that stably sorts keys_u64 under u64 order."""
    n = keys_u64.numel()
    payload = torch.arange(n, dtype=torch.int64, device=keys_u64.device)
    _, perm = sort_pairs(keys_u64, payload)
    return perm


def run_merge_perm(keys_u64: torch.Tensor, seg: torch.Tensor,
                   split: torch.Tensor) -> torch.Tensor:
    """Segmented two-sorted-run merge permutation (K4b): one launch
    merges every bucket's [A-run | B-run] layout into sorted order."""
    if keys_u64.is_cuda:
        return native.ext().run_merge_perm(keys_u64.contiguous(), seg,
                                           split)
    return cpu_ref.run_merge_perm(keys_u64, seg, split)


def merge_join(lkeys: torch.Tensor, rkeys: torch.Tensor,
               lseg: torch.Tensor, rseg: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _is_cuda(lkeys, rkeys):
        return native.ext().merge_join(lkeys, rkeys, lseg, rseg)
    return cpu_ref.merge_join(lkeys, rkeys, lseg, rseg)


def select_range_u64(keys_u64: torch.Tensor, lo: int, hi: int,
                     lo_incl: bool = True, hi_incl: bool = True
                     ) -> torch.Tensor:
    if _is_cuda(keys_u64):
        return native.ext().select_range_u64(
            keys_u64, lo, hi, lo_incl, hi_incl)
    return cpu_ref.select_range_u64(keys_u64, lo, hi, lo_incl, hi_incl)


def isin_sorted(values: torch.Tensor, sorted_set: torch.Tensor
                ) -> torch.Tensor:
    if _is_cuda(values):
        return native.ext().isin_sorted(values, sorted_set)
    return cpu_ref.isin_sorted(values, sorted_set)


def segmented_minmax(vals: torch.Tensor, seg_off: torch.Tensor
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _is_cuda(vals):
        return native.ext().segmented_minmax(vals, seg_off)
    return cpu_ref.segmented_minmax(vals, seg_off)


def bloom_build(vals: torch.Tensor, m_bits: int, k: int) -> torch.Tensor:
    if _is_cuda(vals):
        return native.ext().bloom_build(vals, m_bits, k)
    return cpu_ref.bloom_build(vals, m_bits, k)


def bloom_probe(vals: torch.Tensor, words: torch.Tensor, m_bits: int, k: int
                ) -> torch.Tensor:
    if _is_cuda(vals):
        return native.ext().bloom_probe(vals, words, m_bits, k)
    return cpu_ref.bloom_probe(vals, words, m_bits, k)


def bloom_probe_many(vals: torch.Tensor, words: torch.Tensor, m_bits: int,
                     k: int) -> torch.Tensor:
    """K9 batched sketch-predicate probe: ``words`` is [n_filters,
    words_per_filter]; returns bool[n_filters] = any value may be in
    that filter (device kernel when the tensors are on GPU)."""
    if _is_cuda(vals):
        return native.ext().bloom_probe_many(vals, words, m_bits, k)
    out = torch.zeros(words.shape[0], dtype=torch.bool)
    for f in range(words.shape[0]):
        out[f] = bool(cpu_ref.bloom_probe(vals, words[f], m_bits, k)
                      .any())
    return out


def zorder_key(cols_u64: List[torch.Tensor], bits_per_col: int
               ) -> torch.Tensor:
    if _is_cuda(*cols_u64):
        return native.ext().zorder_key(list(cols_u64), bits_per_col)
    return cpu_ref.zorder_key(cols_u64, bits_per_col)
