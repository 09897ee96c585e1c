"""Hashing helpers (reference: util/HashingUtils.scala — md5Hex)."""

import hashlib


def md5_hex(s: str) -> str:
    return hashlib.md5(s.encode("utf-8")).hexdigest()
