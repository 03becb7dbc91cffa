"""Generic LRU + multi-layer cache (reference common/cacheService.ts)."""

from __future__ import annotations

import time
from collections import OrderedDict
from typing import Generic, Hashable, Optional, TypeVar

K = TypeVar("K", bound=Hashable)
V = TypeVar("V")


class LRUCache(Generic[K, V]):
    def __init__(self, capacity: int = 128, ttl_ms: Optional[float] = None) -> None:
        self.capacity = capacity
        self.ttl_ms = ttl_ms
        self._data: "OrderedDict[K, tuple]" = OrderedDict()
        self.hits = 0
        self.misses = 0

    def get(self, key: K, default: Optional[V] = None) -> Optional[V]:
        item = self._data.get(key)
        if item is None:
            self.misses += 1
            return default
        value, stored_at = item
        if self.ttl_ms is not None and (time.time() * 1000 - stored_at) > self.ttl_ms:
            del self._data[key]
            self.misses += 1
            return default
        self._data.move_to_end(key)
        self.hits += 1
        return value

    def put(self, key: K, value: V) -> None:
        if key in self._data:
            self._data.move_to_end(key)
        self._data[key] = (value, time.time() * 1000)
        while len(self._data) > self.capacity:
            self._data.popitem(last=False)

    def invalidate(self, key: K) -> None:
        self._data.pop(key, None)

    def clear(self) -> None:
        self._data.clear()

    def __len__(self) -> int:
        return len(self._data)

    @property
    def hit_rate(self) -> Optional[float]:
        total = self.hits + self.misses
        return self.hits / total if total else None


class MultiLayerCache:
    """Two-layer (hot/warm) cache: small fast layer over a larger TTL layer."""

    def __init__(self, hot_capacity: int = 32, warm_capacity: int = 256,
                 warm_ttl_ms: float = 300_000) -> None:
        self.hot: LRUCache = LRUCache(hot_capacity)
        self.warm: LRUCache = LRUCache(warm_capacity, ttl_ms=warm_ttl_ms)

    def get(self, key, default=None):
        v = self.hot.get(key)
        if v is not None:
            return v
        v = self.warm.get(key)
        if v is not None:
            self.hot.put(key, v)
            return v
        return default

    def put(self, key, value) -> None:
        self.hot.put(key, value)
        self.warm.put(key, value)
