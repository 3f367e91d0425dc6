"""Expiring key-value storage with heap-based eviction.

Parity target: reference ``hivemind/utils/timed_storage.py:14-143``
(``TimedStorage``, ``get_dht_time``, ``MAX_DHT_TIME_DISCREPANCY_SECONDS``).
"""

from __future__ import annotations

import heapq
import time
from contextlib import contextmanager
from typing import Any, Dict, Generic, Iterator, List, NamedTuple, Optional, Tuple, TypeVar

KeyType = TypeVar("KeyType")
ValueType = TypeVar("ValueType")

DHTExpiration = float
MAX_DHT_TIME_DISCREPANCY_SECONDS = 3.0  # same tolerance as reference (timed_storage.py:15)
ROOT = 0


def get_dht_time() -> DHTExpiration:
    """Global time used for all expirations; wall clock, NTP-synced in deployment."""
    return time.time()


class ValueWithExpiration(NamedTuple):
    value: Any
    expiration_time: DHTExpiration

    def __eq__(self, other):
        if isinstance(other, ValueWithExpiration):
            return self.value == other.value and self.expiration_time == other.expiration_time
        if isinstance(other, tuple):
            return tuple.__eq__(self, other)
        return False

    def __ne__(self, other):
        return not self.__eq__(other)


class HeapEntry(NamedTuple):
    expiration_time: DHTExpiration
    key: Any


class TimedStorage(Generic[KeyType, ValueType]):
    """A dict that stores (value, expiration_time) and evicts expired entries lazily.

    ``store`` keeps the *latest-expiring* value for a key; storing with an
    earlier expiration than the current entry is a no-op (returns False).
    """

    frozen = False  # set True to preserve entries regardless of expiration (for tests)

    def __init__(self, maxsize: Optional[int] = None):
        self.maxsize = maxsize or float("inf")
        self.data: Dict[KeyType, ValueWithExpiration[ValueType]] = {}
        self.expiration_heap: List[HeapEntry] = []
        self.key_to_heap: Dict[KeyType, HeapEntry[KeyType]] = {}

    def _remove_outdated(self):
        while (
            not self.frozen
            and self.expiration_heap
            and (
                self.expiration_heap[ROOT].expiration_time < get_dht_time()
                or len(self.expiration_heap) > len(self.data) * 2 + 16
            )
        ):
            entry = heapq.heappop(self.expiration_heap)
            if self.key_to_heap.get(entry.key) == entry:
                if entry.expiration_time < get_dht_time():
                    del self.data[entry.key], self.key_to_heap[entry.key]
                else:
                    heapq.heappush(self.expiration_heap, entry)
                    break

    def store(self, key: KeyType, value: ValueType, expiration_time: DHTExpiration) -> bool:
        if expiration_time < get_dht_time() and not self.frozen:
            return False
        self.key_to_heap[key] = HeapEntry(expiration_time, key)
        heapq.heappush(self.expiration_heap, self.key_to_heap[key])
        if key in self.data:
            if self.data[key].expiration_time < expiration_time:
                self.data[key] = ValueWithExpiration(value, expiration_time)
                return True
            return False
        self.data[key] = ValueWithExpiration(value, expiration_time)
        self._remove_outdated()
        if len(self.data) > self.maxsize:
            for entry in sorted(self.key_to_heap.values()):
                if entry.key in self.data:
                    del self.data[entry.key], self.key_to_heap[entry.key]
                    break
        return True

    def get(self, key: KeyType) -> Optional[ValueWithExpiration]:
        self._remove_outdated()
        if key in self.data and (self.frozen or self.data[key].expiration_time >= get_dht_time()):
            return self.data[key]
        return None

    def items(self) -> Iterator[Tuple[KeyType, ValueWithExpiration[ValueType]]]:
        self._remove_outdated()
        now = get_dht_time()
        return ((k, v) for k, v in self.data.items() if self.frozen or v.expiration_time >= now)

    def top(self) -> Tuple[Optional[KeyType], Optional[ValueWithExpiration]]:
        """Return the entry with the soonest expiration."""
        self._remove_outdated()
        while self.expiration_heap:
            entry = self.expiration_heap[ROOT]
            if self.key_to_heap.get(entry.key) == entry:
                return entry.key, self.data[entry.key]
            heapq.heappop(self.expiration_heap)
        return None, None

    def __contains__(self, key: KeyType) -> bool:
        self._remove_outdated()
        return key in self.data and (self.frozen or self.data[key].expiration_time >= get_dht_time())

    def __len__(self) -> int:
        self._remove_outdated()
        return len(self.data)

    def __delitem__(self, key: KeyType):
        if key in self.key_to_heap:
            del self.data[key], self.key_to_heap[key]

    def __bool__(self) -> bool:
        return bool(self.data)

    @contextmanager
    def freeze(self):
        """Context where entries are not evicted (used by DHT protocol during iteration)."""
        prev, self.frozen = self.frozen, True
        try:
            yield self
        finally:
            self.frozen = prev
