"""Conf-keyed memo cache (reference: util/CacheWithTransform.scala —
caches a derived value, invalidated when the conf inputs it was derived
from change)."""

from __future__ import annotations

from typing import Callable, Generic, Optional, TypeVar

K = TypeVar("K")
V = TypeVar("V")


class CacheWithTransform(Generic[K, V]):
    def __init__(self, key_fn: Callable[[], K],
                 transform: Callable[[K], V]):
        self.key_fn = key_fn
        self.transform = transform
        self._key: Optional[K] = None
        self._value: Optional[V] = None

    def load(self) -> V:
        key = self.key_fn()
        if self._key != key or self._value is None:
            self._key = key
            self._value = self.transform(key)
        return self._value

    def clear(self):
        self._key = None
        self._value = None
