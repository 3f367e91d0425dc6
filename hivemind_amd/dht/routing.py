"""Kademlia routing: 256-bit DHTIDs, k-buckets with splitting, k-nearest queries.

Parity target: reference ``hivemind/dht/routing.py:20-303`` (RoutingTable,
KBucket with replacement nodes, DHTID with XOR distance). Differences by
design: IDs are SHA-256 (256-bit) rather than SHA-1, and bucket residents map
DHTID -> PeerInfo-endpoint strings instead of multiaddrs.
"""

from __future__ import annotations

import hashlib
import heapq
import os
import random
from itertools import chain
from typing import Any, Dict, Iterator, List, Optional, Sequence, Tuple, Union

from ..p2p import PeerID
from ..utils.serializer import MSGPackSerializer

DHTKey = Any
Subkey = Any
BinaryDHTValue = bytes

ID_BITS = 256


class DHTID(int):
    MIN, MAX = 0, 2**ID_BITS - 1

    def __new__(cls, value: int):
        assert cls.MIN <= value <= cls.MAX, "DHTID out of range"
        return super().__new__(cls, value)

    @classmethod
    def generate(cls, source: Optional[Any] = None, nbits: int = 255) -> "DHTID":
        """Hash `source` (or random bytes) into a DHTID."""
        source = os.urandom(32) if source is None else source
        if not isinstance(source, bytes):
            source = MSGPackSerializer.dumps(source)
        return cls(int.from_bytes(hashlib.sha256(source).digest(), "big"))

    def xor_distance(self, other: Union["DHTID", Sequence["DHTID"]]):
        if isinstance(other, (int,)):
            return int(self) ^ int(other)
        return [int(self) ^ int(x) for x in other]

    @classmethod
    def longest_common_prefix_length(cls, *ids: "DHTID") -> int:
        ids_bits = [bin(uid)[2:].rjust(ID_BITS, "0") for uid in ids]
        for i in range(min(map(len, ids_bits))):
            if len(set(bits[i] for bits in ids_bits)) > 1:
                return i
        return min(map(len, ids_bits))

    def to_bytes(self, length: int = 32, byteorder: str = "big", *, signed: bool = False) -> bytes:
        # routing-table DHTIDs get re-serialized into every FindResponse that
        # lists them as neighbors: cache the canonical 32-byte form
        if length == 32 and byteorder == "big" and not signed:
            b = getattr(self, "_b32", None)
            if b is None:
                b = self._b32 = int.to_bytes(self, 32, "big")
            return b
        return int.to_bytes(self, length, byteorder, signed=signed)

    # the same ~swarm-size node ids get re-parsed from every FindResponse:
    # intern by raw bytes so a repeat costs one dict lookup instead of an int
    # parse + object alloc (cap bounds memory in long-running processes)
    _INTERN: dict = {}
    _INTERN_CAP = 65536

    @classmethod
    def from_bytes(cls, raw: bytes, byteorder: str = "big", *, signed: bool = False) -> "DHTID":
        if byteorder == "big" and not signed and cls is DHTID:
            v = cls._INTERN.get(raw)
            if v is None:
                if len(cls._INTERN) >= cls._INTERN_CAP:
                    cls._INTERN.clear()
                # a 32-byte input is in range by construction: bypass __new__'s assert
                v = cls._INTERN[raw] = int.__new__(cls, int.from_bytes(raw, "big"))
                if len(raw) == 32:
                    v._b32 = raw
            return v
        return int.__new__(cls, int.from_bytes(raw, byteorder, signed=signed))

    def __repr__(self):
        return f"{self.__class__.__name__}({hex(self)[:10]}…)"

    def __bytes__(self):
        return self.to_bytes()


class RoutingTable:
    """K-bucket routing table with bucket splitting and LRU replacement.

    Maps known DHTIDs to their peers (PeerID + endpoint string).
    """

    def __init__(self, node_id: DHTID, bucket_size: int = 20, depth_modulo: int = 5):
        self.node_id, self.bucket_size, self.depth_modulo = node_id, bucket_size, depth_modulo
        self.buckets = [KBucket(DHTID.MIN, DHTID.MAX + 1, bucket_size)]
        self.peer_id_to_uid: Dict[PeerID, DHTID] = {}
        self.uid_to_peer_id: Dict[DHTID, PeerID] = {}

    def get_bucket_index(self, node_id: DHTID) -> int:
        lo, hi = 0, len(self.buckets)
        while hi - lo > 1:
            mid = (lo + hi) // 2
            if self.buckets[mid].lower <= node_id:
                lo = mid
            else:
                hi = mid
        return lo

    def add_or_update_node(self, node_id: DHTID, peer_id: PeerID, endpoint: str) -> Optional[Tuple[DHTID, PeerID]]:
        """Update routing table after an incoming request or response.

        Returns None on success; if the bucket is full, returns the node to ping
        (the least-recently-updated resident) -- caller decides eviction.
        """
        bucket_index = self.get_bucket_index(node_id)
        bucket = self.buckets[bucket_index]
        store_success = bucket.add_or_update_node(node_id, peer_id, endpoint)
        if node_id in bucket.nodes_to_peers or node_id in bucket.replacement_nodes:
            self.uid_to_peer_id[node_id] = peer_id
            self.peer_id_to_uid[peer_id] = node_id
        if not store_success:
            # either split the bucket (if it covers our own id or is shallow) or request eviction check
            if bucket.has_in_range(self.node_id) or bucket.depth % self.depth_modulo != 0:
                self.split_bucket(bucket_index)
                return self.add_or_update_node(node_id, peer_id, endpoint)
            lru_id, lru_peer = bucket.request_ping_node()
            return (lru_id, lru_peer)
        return None

    def split_bucket(self, index: int):
        first, second = self.buckets[index].split()
        self.buckets[index : index + 1] = [first, second]

    def get(self, *, node_id: Optional[DHTID] = None, peer_id: Optional[PeerID] = None):
        """Look up endpoint by node id, or node id by peer id."""
        if node_id is not None:
            bucket = self.buckets[self.get_bucket_index(node_id)]
            return bucket.nodes_to_peers.get(node_id) or bucket.replacement_nodes.get(node_id)
        if peer_id is not None:
            return self.peer_id_to_uid.get(peer_id)
        return None

    def __contains__(self, node_id: DHTID) -> bool:
        bucket = self.buckets[self.get_bucket_index(node_id)]
        return node_id in bucket.nodes_to_peers or node_id in bucket.replacement_nodes

    def __delitem__(self, node_id: DHTID):
        bucket = self.buckets[self.get_bucket_index(node_id)]
        peer_entry = bucket.nodes_to_peers.get(node_id)
        del bucket[node_id]
        if peer_entry is not None:
            peer_id = peer_entry[0]
            if self.peer_id_to_uid.get(peer_id) == node_id:
                del self.peer_id_to_uid[peer_id]
            self.uid_to_peer_id.pop(node_id, None)

    def get_nearest_neighbors(
        self, query_id: DHTID, k: int, exclude: Optional[DHTID] = None
    ) -> List[Tuple[DHTID, Tuple[PeerID, str]]]:
        """k nodes nearest to query_id by XOR metric, excluding `exclude`."""
        # hot path: called for every incoming rpc_find; DHTID is an int, so
        # XOR directly and tuple-sort (distances are unique -- XOR metric with
        # distinct uids -- so the comparison never falls through to field 2)
        q = int(query_id)
        candidates: List[Tuple[int, DHTID, Tuple[PeerID, str]]] = []
        append = candidates.append
        for bucket in self.buckets:
            for uid, peer in bucket.nodes_to_peers.items():
                if uid != exclude:
                    append((q ^ uid, uid, peer))
        candidates.sort()
        return [(uid, peer) for _, uid, peer in candidates[:k]]

    def __repr__(self):
        total = sum(len(b.nodes_to_peers) for b in self.buckets)
        return f"RoutingTable({total} nodes, {len(self.buckets)} buckets)"


class KBucket:
    """Bucket of up to `size` nodes in [lower, upper) with replacement candidates."""

    def __init__(self, lower: int, upper: int, size: int, depth: int = 0):
        assert upper - lower == 2 ** (ID_BITS - depth)
        self.lower, self.upper, self.size, self.depth = lower, upper, size, depth
        self.nodes_to_peers: Dict[DHTID, Tuple[PeerID, str]] = {}
        self.replacement_nodes: Dict[DHTID, Tuple[PeerID, str]] = {}
        self.nodes_requested_for_ping: set = set()
        self.last_updated = 0.0

    def has_in_range(self, node_id: DHTID) -> bool:
        return self.lower <= node_id < self.upper

    def add_or_update_node(self, node_id: DHTID, peer_id: PeerID, endpoint: str) -> bool:
        import time as _time

        self.last_updated = _time.time()
        if node_id in self.nodes_requested_for_ping:
            self.nodes_requested_for_ping.discard(node_id)
        if node_id in self.nodes_to_peers:
            # move to the end (most recently seen)
            del self.nodes_to_peers[node_id]
            self.nodes_to_peers[node_id] = (peer_id, endpoint)
            return True
        if len(self.nodes_to_peers) < self.size:
            self.nodes_to_peers[node_id] = (peer_id, endpoint)
            return True
        self.replacement_nodes[node_id] = (peer_id, endpoint)
        return False

    def request_ping_node(self) -> Tuple[DHTID, Tuple[PeerID, str]]:
        for uid, peer in self.nodes_to_peers.items():
            if uid not in self.nodes_requested_for_ping:
                self.nodes_requested_for_ping.add(uid)
                return uid, peer
        # all requested already: return the LRU anyway
        uid = next(iter(self.nodes_to_peers))
        return uid, self.nodes_to_peers[uid]

    def __getitem__(self, node_id: DHTID) -> Tuple[PeerID, str]:
        return self.nodes_to_peers[node_id] if node_id in self.nodes_to_peers else self.replacement_nodes[node_id]

    def __delitem__(self, node_id: DHTID):
        if node_id in self.nodes_to_peers:
            del self.nodes_to_peers[node_id]
            if self.replacement_nodes:
                newnode_id, newnode = self.replacement_nodes.popitem()
                self.nodes_to_peers[newnode_id] = newnode
        self.replacement_nodes.pop(node_id, None)
        self.nodes_requested_for_ping.discard(node_id)

    def split(self) -> Tuple["KBucket", "KBucket"]:
        midpoint = (self.lower + self.upper) // 2
        left = KBucket(self.lower, midpoint, self.size, depth=self.depth + 1)
        right = KBucket(midpoint, self.upper, self.size, depth=self.depth + 1)
        for node_id, peer in chain(self.nodes_to_peers.items(), self.replacement_nodes.items()):
            bucket = left if node_id < midpoint else right
            bucket.add_or_update_node(node_id, *peer)
        return left, right

    def __repr__(self):
        return (
            f"KBucket({hex(self.lower)[:8]}…{hex(self.upper)[:8]}, depth={self.depth}, "
            f"{len(self.nodes_to_peers)} nodes, {len(self.replacement_nodes)} replacements)"
        )
