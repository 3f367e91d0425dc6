"""Local DHT storage: plain values or sub-key dictionaries with per-subkey expiration.

Parity target: reference ``hivemind/dht/storage.py:11-70`` (DictionaryDHTValue
msgpack ext type 0x50, DHTLocalStorage.store / store_subkey semantics).
"""

from __future__ import annotations

from typing import Optional, Union

from ..utils.serializer import MSGPackSerializer
from ..utils.timed_storage import DHTExpiration, TimedStorage, ValueWithExpiration
from .routing import BinaryDHTValue, DHTID, Subkey


@MSGPackSerializer.ext_serializable(0x50)
class DictionaryDHTValue(TimedStorage):
    """A dictionary-valued DHT record: each subkey has its own value + expiration.

    The record's own ``latest_expiration_time`` is the max over subkeys; a
    dictionary can only be replaced by a regular value with a strictly later
    expiration (and vice versa).
    """

    latest_expiration_time = float("-inf")

    def store(self, key: Subkey, value: BinaryDHTValue, expiration_time: DHTExpiration) -> bool:
        self.latest_expiration_time = max(self.latest_expiration_time, expiration_time)
        return super().store(key, value, expiration_time)

    def packb(self) -> bytes:
        packed_items = [[key, value, expiration] for key, (value, expiration) in self.items()]
        return MSGPackSerializer.dumps([self.latest_expiration_time, self.maxsize if self.maxsize != float("inf") else None, packed_items])

    @classmethod
    def unpackb(cls, data: bytes) -> "DictionaryDHTValue":
        latest_expiration_time, maxsize, items = MSGPackSerializer.loads(data)
        with DHTLocalStorage.allow_expired():
            self = cls(maxsize=maxsize)
            self.frozen = True  # keep entries while loading even if expired in transit
            for key, value, expiration in items:
                self.store(key, value, expiration)
            self.frozen = False
        self.latest_expiration_time = max(self.latest_expiration_time, latest_expiration_time)
        return self


class DHTLocalStorage(TimedStorage):
    """Node-local storage of key -> bytes | DictionaryDHTValue."""

    from contextlib import contextmanager

    @classmethod
    @contextmanager
    def allow_expired(cls):
        yield

    def store(
        self, key: DHTID, value: BinaryDHTValue, expiration_time: DHTExpiration
    ) -> bool:
        existing = super().get(key)
        if existing is not None and isinstance(existing.value, DictionaryDHTValue):
            if not isinstance(value, DictionaryDHTValue) and expiration_time <= existing.value.latest_expiration_time:
                return False  # regular value cannot displace a fresher dictionary
        return super().store(key, value, expiration_time)

    def store_subkey(
        self, key: DHTID, subkey: Subkey, value: BinaryDHTValue, expiration_time: DHTExpiration
    ) -> bool:
        """Add (subkey -> value) into a dictionary record at `key`.

        If `key` currently holds a regular value with a later expiration, the
        subkey store is rejected (reference storage.py:51-69).
        """
        previous = super().get(key)
        if previous is None or not isinstance(previous.value, DictionaryDHTValue):
            if previous is not None and previous.expiration_time >= expiration_time:
                return False
            new_dict = DictionaryDHTValue()
            new_dict.store(subkey, value, expiration_time)
            return super().store(key, new_dict, expiration_time)
        ok = previous.value.store(subkey, value, expiration_time)
        if ok:
            # refresh outer expiration to the dictionary's latest
            super().store(key, previous.value, previous.value.latest_expiration_time)
        return ok

    def get(self, key: DHTID) -> Optional[ValueWithExpiration]:
        result = super().get(key)
        if result is not None and isinstance(result.value, DictionaryDHTValue):
            return ValueWithExpiration(result.value, result.value.latest_expiration_time)
        return result
