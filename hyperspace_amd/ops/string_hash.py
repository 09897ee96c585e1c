"""Spark-style Murmur3 for UTF-8 strings + bucket-key extraction that is
consistent across batches for dictionary-encoded string columns.

Spark hashes strings with Murmur3_x86_32.hashUnsafeBytes over the UTF-8
bytes (4-byte little-endian words, then per-byte sign-extended tail,
fmix by total length).  We hash each dictionary VALUE once on the host
(dictionaries are small) and map codes through a LUT on device, so the
bucket of a row depends only on the string value — never on the
per-batch code assignment.
"""

from __future__ import annotations

from typing import List

import torch

from . import cpu_ref

MASK32 = 0xFFFFFFFF


def _mix_k1(k1: int) -> int:
    k1 = (k1 * 0xCC9E2D51) & MASK32
    k1 = ((k1 << 15) | (k1 >> 17)) & MASK32
    k1 = (k1 * 0x1B873593) & MASK32
    return k1


def _mix_h1(h1: int, k1: int) -> int:
    h1 ^= k1
    h1 = ((h1 << 13) | (h1 >> 19)) & MASK32
    h1 = (h1 * 5 + 0xE6546B64) & MASK32
    return h1


def _fmix(h1: int, length: int) -> int:
    h1 ^= length
    h1 ^= h1 >> 16
    h1 = (h1 * 0x85EBCA6B) & MASK32
    h1 ^= h1 >> 13
    h1 = (h1 * 0xC2B2AE35) & MASK32
    h1 ^= h1 >> 16
    return h1


def murmur3_string(s: str, seed: int) -> int:
    """Murmur3_x86_32.hashUnsafeBytes(utf8(s), seed) — Spark's string
    hash (4-byte LE words + sign-extended byte tail)."""
    data = s.encode("utf-8")
    n = len(data)
    h1 = seed & MASK32
    aligned = n - n % 4
    for i in range(0, aligned, 4):
        word = int.from_bytes(data[i:i + 4], "little")
        h1 = _mix_h1(h1, _mix_k1(word))
    for i in range(aligned, n):
        b = data[i]
        if b >= 128:
            b -= 256  # sign-extended byte
        h1 = _mix_h1(h1, _mix_k1(b & MASK32))
    return _fmix(h1, n)


def bucket_hash_keys(batch, columns: List[str]) -> List[torch.Tensor]:
    """Extract hashable per-row key tensors for bucket assignment.

    Numeric columns pass through; string columns become int32 tensors of
    per-VALUE murmur3 hashes (seed 42 fold handled by the caller's
    multi-column fold is NOT applicable here — string columns contribute
    their value-hash as an i32 column, which the murmur fold then mixes;
    this is self-consistent across builds and queries)."""
    from ..execution.columnar import StringColumn
    out = []
    for c in columns:
        col = batch.column(c)
        if isinstance(col, StringColumn):
            lut = torch.tensor(
                [murmur3_string(v, 42) & MASK32 for v in col.values],
                dtype=torch.int64, device=col.codes.device)
            hashed = lut[col.codes.long()].to(torch.int32)
            out.append(hashed)
        else:
            out.append(col)
    return out


def bucket_of_string_value(value: str, num_buckets: int) -> int:
    """Bucket of a string literal under the scheme above (single string
    key column): fold murmur3(value-hash as i32) like the kernel does."""
    h = murmur3_string(value, 42) & MASK32
    signed = h if h <= 0x7FFFFFFF else h - (1 << 32)
    t = torch.tensor([signed], dtype=torch.int32)
    return int(cpu_ref.murmur3_bucket([t], num_buckets)[0])
