"""Group keys: where peers advertise themselves for matchmaking.

Parity target: reference ``hivemind/averaging/key_manager.py:22-119``.
Group key = ``{prefix}.0b{group_bits}``; after every successful round the
suffix is re-hashed with an RNG seeded by the group id, so small groups mix
globally in O(log N) rounds (Moshpit SGD). On one 8-GPU MI355X node the
default ``target_group_size=8`` makes a single group -- the rehash only
matters for multi-node swarms.
"""

from __future__ import annotations

import asyncio
import random
import re
from typing import List, Optional, Tuple

from ..dht import DHT
from ..p2p import PeerID
from ..utils.logging import get_logger
from ..utils.timed_storage import DHTExpiration, ValueWithExpiration, get_dht_time
from .group_info import GroupInfo

logger = get_logger(__name__)

GroupKey = str
GROUP_PATTERN = re.compile(r"^(([^.])+)[.]0b[01]*$")  # e.g. bert_exp4_averaging.0b01001101


def is_valid_group(maybe_group: str) -> bool:
    return bool(GROUP_PATTERN.fullmatch(maybe_group))


class GroupKeyManager:
    """Declares averagers under group keys and fetches current group membership."""

    RESERVED_KEY_FOR_NBITS = "nbits"

    def __init__(
        self,
        dht: DHT,
        prefix: str,
        initial_group_bits: str = "",
        target_group_size: Optional[int] = None,
        p2p=None,
    ):
        assert all(bit in "01" for bit in initial_group_bits)
        if target_group_size is not None and not (target_group_size & (target_group_size - 1) == 0):
            logger.warning("target_group_size is not a power of two, which may cause uneven group sizes")
        self.dht, self.prefix, self.group_bits = dht, prefix, initial_group_bits
        self.target_group_size = target_group_size
        self.peer_id = dht.peer_id
        self.p2p = p2p if p2p is not None else dht.replicate_p2p()

    @property
    def current_key(self) -> GroupKey:
        return f"{self.prefix}.0b{self.group_bits}"

    async def declare_averager(
        self, group_key: GroupKey, peer_id: PeerID, expiration_time: DHTExpiration, looking_for_group: bool = True
    ) -> bool:
        """Publish (peer_id -> looking_for_group) under the group key (reference key_manager.py:46)."""
        expiration = expiration_time if looking_for_group else float(get_dht_time())
        endpoint = self.p2p.endpoint if self.p2p is not None else ""
        return await asyncio.wrap_future(
            self.dht.store(
                key=group_key,
                subkey=peer_id.to_base58(),
                value=[looking_for_group, endpoint],
                expiration_time=expiration,
                return_future=True,
            )
        )

    async def get_averagers(self, group_key: GroupKey, only_active: bool = True) -> List[Tuple[PeerID, DHTExpiration]]:
        """Find all averagers declared under this key (reference key_manager.py:70)."""
        assert is_valid_group(group_key), f"invalid group key: {group_key}"
        result = await asyncio.wrap_future(self.dht.get(group_key, latest=True, return_future=True))
        if result is None or not isinstance(result.value, dict):
            return []
        averagers = []
        for key, entry in result.value.items():
            if key == self.RESERVED_KEY_FOR_NBITS:
                continue
            try:
                value = entry.value
                if isinstance(value, (list, tuple)) and len(value) == 2:
                    looking_for_group, endpoint = value
                else:
                    looking_for_group, endpoint = bool(value), ""
                if only_active and not looking_for_group:
                    continue
                peer_id = PeerID.from_base58(key)
                if endpoint and self.p2p is not None and peer_id != self.peer_id:
                    self.p2p.learn_endpoint(peer_id, endpoint)
                averagers.append((peer_id, entry.expiration_time))
            except Exception as e:
                logger.debug(f"skipping malformed averager record {key}: {e}")
        return averagers

    async def update_key_on_group_assembled(self, group_info: GroupInfo):
        """Re-hash into a new group-bits suffix seeded by the group id
        (reference key_manager.py:94: Moshpit-style mixing)."""
        if self.target_group_size is None or not self.group_bits:
            return
        rng = random.Random(group_info.group_id)
        index = group_info.peer_ids.index(self.peer_id)
        generalized_index = rng.sample(range(self.target_group_size), group_info.group_size)[index]
        nbits = max(1, (self.target_group_size - 1).bit_length())
        new_bits = bin(generalized_index)[2:].rjust(nbits, "0")
        self.group_bits = (self.group_bits + new_bits)[-len(self.group_bits) :]
        logger.debug(f"{self.peer_id}: regrouped, new group bits = {self.group_bits}")

    async def update_key_on_not_enough_peers(self):
        """Shrink the keyspace when the current bucket is too empty (reference key_manager.py:107)."""
        if self.group_bits:
            self.group_bits = self.group_bits[1:]
