"""TimedStorage — key/value store with per-entry expiration (SURVEY.md §2
"TimedStorage / expiring cache"): the local storage behind every DHT node and
its cache. DHT time is plain wall-clock (peers on one node / LAN share it)."""
from __future__ import annotations

import heapq
import time
from typing import Any, Dict, Iterator, Optional, Tuple

DHTExpiration = float


def get_dht_time() -> DHTExpiration:
    return time.time()


class TimedStorage:
    """dict-like storage where each key carries an expiration timestamp.
    `store` keeps the freshest (max-expiration) value per key; expired entries
    vanish lazily on access and on `remove_outdated`."""

    def __init__(self, maxsize: Optional[int] = None):
        self.maxsize = maxsize
        self._data: Dict[Any, Tuple[Any, DHTExpiration]] = {}
        self._expiration_heap: list = []  # (expiration, key)

    def store(self, key: Any, value: Any, expiration_time: DHTExpiration) -> bool:
        """Store value unless a fresher value (later expiration) is present."""
        if expiration_time < get_dht_time():
            return False
        current = self._data.get(key)
        if current is not None and current[1] > expiration_time:
            return False
        self._data[key] = (value, expiration_time)
        heapq.heappush(self._expiration_heap, (expiration_time, key))
        if self.maxsize is not None and len(self._data) > self.maxsize:
            self._evict_one()
        return True

    def _evict_one(self) -> None:
        # evict the soonest-to-expire live entry
        while self._expiration_heap:
            expiration, key = heapq.heappop(self._expiration_heap)
            entry = self._data.get(key)
            if entry is not None and entry[1] == expiration:
                del self._data[key]
                return

    def get(self, key: Any) -> Tuple[Optional[Any], Optional[DHTExpiration]]:
        entry = self._data.get(key)
        if entry is None:
            return None, None
        value, expiration = entry
        if expiration < get_dht_time():
            del self._data[key]
            return None, None
        return value, expiration

    def remove_outdated(self) -> None:
        now = get_dht_time()
        while self._expiration_heap and self._expiration_heap[0][0] < now:
            expiration, key = heapq.heappop(self._expiration_heap)
            entry = self._data.get(key)
            if entry is not None and entry[1] == expiration:
                del self._data[key]

    def items(self) -> Iterator[Tuple[Any, Any, DHTExpiration]]:
        self.remove_outdated()
        for key, (value, expiration) in list(self._data.items()):
            yield key, value, expiration

    def __contains__(self, key: Any) -> bool:
        return self.get(key)[0] is not None or key in self._data and self._data[key][1] >= get_dht_time()

    def __len__(self) -> int:
        self.remove_outdated()
        return len(self._data)

    def top(self) -> Tuple[Optional[Any], Optional[Any], Optional[DHTExpiration]]:
        """Return (key, value, expiration) of the soonest-to-expire live entry."""
        self.remove_outdated()
        while self._expiration_heap:
            expiration, key = self._expiration_heap[0]
            entry = self._data.get(key)
            if entry is not None and entry[1] == expiration:
                return key, entry[0], expiration
            heapq.heappop(self._expiration_heap)
        return None, None, None
