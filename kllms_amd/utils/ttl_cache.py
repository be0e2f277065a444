"""Minimal TTL-bounded LRU cache.

Standalone replacement for the ``cachetools.TTLCache`` the reference uses for
its similarity/embedding memoization (reference: k_llms/utils/consensus_utils.py:620-623).
Not thread-safe by itself — callers guard it with a Lock, exactly as the
reference does (consensus_utils.py:780-794).
"""

from __future__ import annotations

import time
from collections import OrderedDict
from typing import Any, Hashable


class TTLCache:
    def __init__(self, maxsize: int = 1024, ttl: float = 300.0, timer=time.monotonic):
        self.maxsize = maxsize
        self.ttl = ttl
        self._timer = timer
        self._data: "OrderedDict[Hashable, tuple[float, Any]]" = OrderedDict()

    def _expire(self) -> None:
        now = self._timer()
        dead = [k for k, (exp, _) in self._data.items() if exp <= now]
        for k in dead:
            del self._data[k]

    def __contains__(self, key: Hashable) -> bool:
        try:
            self[key]
            return True
        except KeyError:
            return False

    def __getitem__(self, key: Hashable) -> Any:
        exp, value = self._data[key]
        if exp <= self._timer():
            del self._data[key]
            raise KeyError(key)
        return value

    def __setitem__(self, key: Hashable, value: Any) -> None:
        self._expire()
        if key in self._data:
            del self._data[key]
        elif len(self._data) >= self.maxsize:
            self._data.popitem(last=False)
        self._data[key] = (self._timer() + self.ttl, value)

    def get(self, key: Hashable, default: Any = None) -> Any:
        try:
            return self[key]
        except KeyError:
            return default

    def __len__(self) -> int:
        self._expire()
        return len(self._data)

    def clear(self) -> None:
        self._data.clear()
