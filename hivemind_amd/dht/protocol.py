"""DHT wire protocol: rpc_ping / rpc_store / rpc_find over the TCP RPC layer.

Parity target: reference ``hivemind/dht/protocol.py:25-430`` -- the same three
RPCs with the same semantics: ping exchanges node ids and validates clock skew
(±3 s); store accepts bulk keys/subkeys/values/expirations with an in_cache
flag; find returns the stored value (regular or dictionary) plus the k nearest
nodes from the routing table; every request updates the routing table.
"""

from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field
from typing import Collection, Dict, List, Optional, Sequence, Tuple, Union

from ..p2p import P2P, PeerID, RpcContext, RpcMessage, ServicerBase
from ..utils.logging import get_logger
from ..utils.serializer import MSGPackSerializer
from ..utils.timed_storage import (
    DHTExpiration,
    MAX_DHT_TIME_DISCREPANCY_SECONDS,
    ValueWithExpiration,
    get_dht_time,
)
from .routing import BinaryDHTValue, DHTID, RoutingTable, Subkey
from .storage import DHTLocalStorage, DictionaryDHTValue
from .validation import DHTRecord, RecordValidatorBase

logger = get_logger(__name__)


@dataclass
class NodeInfo(RpcMessage):
    node_id: bytes = b""
    endpoint: str = ""


@dataclass
class PingRequest(RpcMessage):
    peer: Optional[NodeInfo] = None
    validate: bool = False


@dataclass
class PingResponse(RpcMessage):
    peer: Optional[NodeInfo] = None
    sender_endpoint: str = ""
    dht_time: float = 0.0
    available: bool = False


@dataclass
class StoreRequest(RpcMessage):
    keys: List[bytes] = field(default_factory=list)
    subkeys: List[bytes] = field(default_factory=list)  # b"" = plain value
    values: List[bytes] = field(default_factory=list)
    expiration_time: List[float] = field(default_factory=list)
    in_cache: List[bool] = field(default_factory=list)
    peer: Optional[NodeInfo] = None


@dataclass
class StoreResponse(RpcMessage):
    store_ok: List[bool] = field(default_factory=list)
    peer: Optional[NodeInfo] = None


@dataclass
class FindRequest(RpcMessage):
    keys: List[bytes] = field(default_factory=list)
    peer: Optional[NodeInfo] = None


NOT_FOUND, FOUND_REGULAR, FOUND_DICTIONARY = 0, 1, 2


@dataclass
class FindResult(RpcMessage):
    type: int = NOT_FOUND
    value: bytes = b""
    expiration_time: float = -float("inf")
    # nearest nodes: parallel lists
    nearest_node_ids: List[bytes] = field(default_factory=list)
    nearest_peer_ids: List[bytes] = field(default_factory=list)
    nearest_endpoints: List[str] = field(default_factory=list)


@dataclass
class FindResponse(RpcMessage):
    results: List[FindResult] = field(default_factory=list)
    peer: Optional[NodeInfo] = None


class ValidationError(Exception):
    """A record or response failed validation."""


class DHTProtocol(ServicerBase):
    serializer = MSGPackSerializer

    def __init__(self):
        self.node_id: Optional[DHTID] = None
        self.p2p: Optional[P2P] = None
        self.routing_table: Optional[RoutingTable] = None
        self.storage = DHTLocalStorage()
        self.cache = DHTLocalStorage()
        self.bucket_size = 20
        self.num_replicas = 5
        self.wait_timeout = 5.0
        self.record_validator: Optional[RecordValidatorBase] = None
        self.client_mode = False
        self.rpc_stats: dict = {}

    @classmethod
    async def create(
        cls,
        p2p: P2P,
        node_id: DHTID,
        bucket_size: int = 20,
        depth_modulo: int = 5,
        num_replicas: int = 5,
        wait_timeout: float = 5.0,
        cache_size: Optional[int] = None,
        client_mode: bool = False,
        record_validator: Optional[RecordValidatorBase] = None,
    ) -> "DHTProtocol":
        self = cls()
        self.p2p, self.node_id = p2p, node_id
        self.bucket_size, self.num_replicas, self.wait_timeout = bucket_size, num_replicas, wait_timeout
        # per-method client-side in-flight time sums/counts (observability)
        self.rpc_stats: dict = {}
        self.routing_table = RoutingTable(node_id, bucket_size, depth_modulo)
        self.cache = DHTLocalStorage(maxsize=cache_size)
        self.record_validator = record_validator
        self.client_mode = client_mode
        if not client_mode:
            await self.add_p2p_handlers(p2p)
        return self

    async def shutdown(self):
        if not self.client_mode and self.p2p is not None:
            self.remove_p2p_handlers(self.p2p)

    # -------------------------------------------------------------- identity

    def _my_info(self) -> NodeInfo:
        endpoint = self.p2p.endpoint if not self.client_mode else ""
        return NodeInfo(node_id=self.node_id.to_bytes(), endpoint=endpoint)

    def _update_routing(self, peer_info: Optional[NodeInfo], remote_peer_id: PeerID):
        """Register the requester/responder in our routing table (reference protocol.py:371)."""
        if peer_info is None or not peer_info.node_id or not peer_info.endpoint:
            return  # client-mode peers are not routable
        node_id = DHTID.from_bytes(peer_info.node_id)
        self.p2p.learn_endpoint(remote_peer_id, peer_info.endpoint)
        maybe_ping = self.routing_table.add_or_update_node(node_id, remote_peer_id, peer_info.endpoint)
        if maybe_ping is not None:
            # bucket full: ping the LRU resident in the background; evict if dead
            asyncio.get_event_loop().create_task(self._ping_and_maybe_evict(*maybe_ping))

    async def _ping_and_maybe_evict(self, node_id: DHTID, peer: Tuple[PeerID, str]):
        peer_id, endpoint = peer
        from ..p2p import PeerInfo

        response = await self.call_ping(PeerInfo(peer_id, (endpoint,)))
        if response is None and node_id in self.routing_table:
            del self.routing_table[node_id]

    # ------------------------------------------------------------------ ping

    async def call_ping(self, peer, validate: bool = False) -> Optional[DHTID]:
        """Ping a peer; returns its DHTID or None if unreachable."""
        try:
            request = PingRequest(peer=self._my_info(), validate=validate)
            time_requested = get_dht_time()
            _t0 = time.monotonic()
            raw = await self._get_stub(peer).rpc_ping(request, timeout=self.wait_timeout)
            _dt = time.monotonic() - _t0
            st = self.rpc_stats
            st["rpc_ping_n"] = st.get("rpc_ping_n", 0) + 1
            st["rpc_ping_s"] = st.get("rpc_ping_s", 0.0) + _dt
            st["rpc_ping_max_s"] = max(st.get("rpc_ping_max_s", 0.0), _dt)
            response = PingResponse.loads(raw)
            time_responded = get_dht_time()
            if validate:
                if not self.client_mode and not response.available:
                    raise ValidationError("peer can't access this node")
                expected = (time_requested + time_responded) / 2
                if abs(response.dht_time - expected) > MAX_DHT_TIME_DISCREPANCY_SECONDS + (time_responded - time_requested):
                    raise ValidationError(
                        f"clock skew with {peer}: local {expected:.3f} vs remote {response.dht_time:.3f}"
                    )
            if response.peer is not None and response.peer.node_id:
                peer_id = peer.peer_id if hasattr(peer, "peer_id") else peer
                self._update_routing(response.peer, peer_id)
                return DHTID.from_bytes(response.peer.node_id)
        except ValidationError:
            raise
        except Exception as e:
            logger.debug(f"call_ping to {peer} failed: {e!r}")
            self._on_peer_failure(peer)
        return None

    async def rpc_ping(self, request: PingRequest, context: RpcContext) -> PingResponse:
        response = PingResponse(peer=self._my_info(), dht_time=get_dht_time(), available=False)
        if request.peer is not None and request.peer.node_id:
            self._update_routing(request.peer, context.remote_id)
            if request.validate and request.peer.endpoint:
                from ..p2p import PeerInfo

                response.available = await self.p2p.ping(PeerInfo(context.remote_id, (request.peer.endpoint,)))
                response.sender_endpoint = request.peer.endpoint
        return response

    # ----------------------------------------------------------------- store

    async def call_store(
        self,
        peer,
        keys: Sequence[DHTID],
        values: Sequence[Union[BinaryDHTValue, DictionaryDHTValue]],
        expiration_time: Union[DHTExpiration, Sequence[DHTExpiration]],
        subkeys: Optional[Sequence[Optional[Subkey]]] = None,
        in_cache: Optional[Union[bool, Sequence[bool]]] = None,
    ) -> Optional[List[bool]]:
        """Bulk-store keys on one peer; returns per-key success or None if unreachable."""
        if isinstance(expiration_time, (int, float)):
            expiration_time = [expiration_time] * len(keys)
        if subkeys is None:
            subkeys = [None] * len(keys)
        if in_cache is None:
            in_cache = [False] * len(keys)
        elif isinstance(in_cache, bool):
            in_cache = [in_cache] * len(keys)
        keys, subkeys, values = list(keys), list(subkeys), list(values)
        assert len(keys) == len(subkeys) == len(values) == len(expiration_time) == len(in_cache)
        wire_keys, wire_subkeys, wire_values = [], [], []
        for key, subkey, value, expiration in zip(keys, subkeys, values, expiration_time):
            wire_keys.append(key.to_bytes())
            if isinstance(value, DictionaryDHTValue):
                assert subkey is None, "can't set subkey when storing a whole dictionary"
                wire_subkeys.append(b"\x00DICT")
                wire_values.append(self.serializer.dumps(value))
            else:
                assert isinstance(value, bytes), f"value must be bytes, got {type(value)}"
                packed_subkey = self.serializer.dumps(subkey) if subkey is not None else b""
                wire_subkeys.append(packed_subkey)
                if self.record_validator is not None:
                    value = self.record_validator.sign_value(
                        DHTRecord(key.to_bytes(), packed_subkey, value, expiration)
                    )
                wire_values.append(value)
        try:
            request = StoreRequest(
                keys=wire_keys,
                subkeys=wire_subkeys,
                values=wire_values,
                expiration_time=list(expiration_time),
                in_cache=list(in_cache),
                peer=self._my_info(),
            )
            _t0 = time.monotonic()
            raw = await self._get_stub(peer).rpc_store(request, timeout=self.wait_timeout)
            _dt = time.monotonic() - _t0
            st = self.rpc_stats
            st["rpc_store_n"] = st.get("rpc_store_n", 0) + 1
            st["rpc_store_s"] = st.get("rpc_store_s", 0.0) + _dt
            st["rpc_store_max_s"] = max(st.get("rpc_store_max_s", 0.0), _dt)
            response = StoreResponse.loads(raw)
            if response.peer is not None and response.peer.node_id:
                peer_id = peer.peer_id if hasattr(peer, "peer_id") else peer
                self._update_routing(response.peer, peer_id)
            return list(response.store_ok)
        except Exception as e:
            logger.debug(f"call_store to {peer} failed: {e!r}")
            self._on_peer_failure(peer)
            return None

    async def rpc_store(self, request: StoreRequest, context: RpcContext) -> StoreResponse:
        if request.peer is not None:
            self._update_routing(request.peer, context.remote_id)
        response = StoreResponse(store_ok=[], peer=self._my_info())
        for key, subkey, value, expiration, in_cache in zip(
            request.keys, request.subkeys, request.values, request.expiration_time, request.in_cache
        ):
            storage = self.cache if in_cache else self.storage
            key_id = DHTID.from_bytes(key)
            if subkey == b"\x00DICT":
                dictionary = self.serializer.loads(value)
                assert isinstance(dictionary, DictionaryDHTValue)
                ok = True
                for sub, (sub_value, sub_expiration) in dictionary.items():
                    packed_sub = self.serializer.dumps(sub)
                    if not self._validate_record(key, packed_sub, sub_value, sub_expiration):
                        ok = False
                        continue
                    ok = storage.store_subkey(key_id, sub, sub_value, sub_expiration) and ok
                response.store_ok.append(ok)
            elif not subkey:
                if not self._validate_record(key, b"", value, expiration):
                    response.store_ok.append(False)
                    continue
                response.store_ok.append(storage.store(key_id, value, expiration))
            else:
                if not self._validate_record(key, subkey, value, expiration):
                    response.store_ok.append(False)
                    continue
                sub = self.serializer.loads(subkey)
                response.store_ok.append(storage.store_subkey(key_id, sub, value, expiration))
        return response

    def _validate_record(self, key: bytes, subkey: bytes, value: bytes, expiration: float) -> bool:
        if self.record_validator is None:
            return True
        return self.record_validator.validate(DHTRecord(key, subkey, value, expiration))

    # ------------------------------------------------------------------ find

    async def call_find(
        self, peer, keys: Collection[DHTID]
    ) -> Optional[Dict[DHTID, Tuple[Optional[ValueWithExpiration], Dict[DHTID, Tuple[PeerID, str]]]]]:
        """Request values & nearest neighbors for keys from one peer.

        Returns {key: ((value, expiration) | None, {neighbor_id: (peer_id, endpoint)})}.
        """
        keys = list(keys)
        try:
            request = FindRequest(keys=[k.to_bytes() for k in keys], peer=self._my_info())
            _t0 = time.monotonic()
            raw = await self._get_stub(peer).rpc_find(request, timeout=self.wait_timeout)
            _dt = time.monotonic() - _t0
            st = self.rpc_stats
            st["rpc_find_n"] = st.get("rpc_find_n", 0) + 1
            st["rpc_find_s"] = st.get("rpc_find_s", 0.0) + _dt
            st["rpc_find_max_s"] = max(st.get("rpc_find_max_s", 0.0), _dt)
            response = FindResponse.loads(raw)
            if response.peer is not None and response.peer.node_id:
                peer_id = peer.peer_id if hasattr(peer, "peer_id") else peer
                self._update_routing(response.peer, peer_id)
            assert len(response.results) == len(keys)
            output = {}
            _from_bytes = DHTID.from_bytes
            for key_id, result in zip(keys, response.results):
                nearest = {
                    _from_bytes(nid): (PeerID(pid), ep)
                    for nid, pid, ep in zip(result.nearest_node_ids, result.nearest_peer_ids, result.nearest_endpoints)
                }
                if result.type == NOT_FOUND:
                    output[key_id] = (None, nearest)
                elif result.type == FOUND_REGULAR:
                    value = result.value
                    if self.record_validator is not None:
                        record = DHTRecord(key_id.to_bytes(), b"", value, result.expiration_time)
                        if not self.record_validator.validate(record):
                            output[key_id] = (None, nearest)
                            continue
                        value = self.record_validator.strip_value(record)
                    output[key_id] = (ValueWithExpiration(value, result.expiration_time), nearest)
                else:  # dictionary
                    dictionary: DictionaryDHTValue = self.serializer.loads(result.value)
                    if self.record_validator is not None:
                        filtered = DictionaryDHTValue()
                        for sub, (sub_value, sub_expiration) in dictionary.items():
                            packed_sub = self.serializer.dumps(sub)
                            record = DHTRecord(key_id.to_bytes(), packed_sub, sub_value, sub_expiration)
                            if self.record_validator.validate(record):
                                filtered.store(sub, self.record_validator.strip_value(record), sub_expiration)
                        dictionary = filtered
                    output[key_id] = (
                        ValueWithExpiration(dictionary, dictionary.latest_expiration_time),
                        nearest,
                    )
            return output
        except Exception as e:
            logger.debug(f"call_find to {peer} failed: {e!r}")
            self._on_peer_failure(peer)
            return None

    async def rpc_find(self, request: FindRequest, context: RpcContext) -> FindResponse:
        if request.peer is not None:
            self._update_routing(request.peer, context.remote_id)
        response = FindResponse(results=[], peer=self._my_info())
        exclude = DHTID.from_bytes(request.peer.node_id) if request.peer and request.peer.node_id else None
        for key_bytes in request.keys:
            key_id = DHTID.from_bytes(key_bytes)
            result = FindResult()
            maybe_item = self.storage.get(key_id)
            cached_item = self.cache.get(key_id)
            if cached_item is not None and (maybe_item is None or cached_item.expiration_time > maybe_item.expiration_time):
                maybe_item = cached_item
            if maybe_item is not None:
                if isinstance(maybe_item.value, DictionaryDHTValue):
                    result.type = FOUND_DICTIONARY
                    result.value = self.serializer.dumps(maybe_item.value)
                else:
                    result.type = FOUND_REGULAR
                    result.value = maybe_item.value
                result.expiration_time = maybe_item.expiration_time
            nn = self.routing_table.get_nearest_neighbors(key_id, k=self.bucket_size, exclude=exclude)
            result.nearest_node_ids = [node_id.to_bytes() for node_id, _ in nn]
            result.nearest_peer_ids = [peer[0].to_bytes() for _, peer in nn]
            result.nearest_endpoints = [peer[1] for _, peer in nn]
            response.results.append(result)
        return response

    # --------------------------------------------------------------- helpers

    def _get_stub(self, peer):
        return DHTProtocol.get_stub(self.p2p, peer)

    def _on_peer_failure(self, peer):
        """Evict an unresponsive peer from the routing table (reference protocol.py:403-405)."""
        peer_id = peer.peer_id if hasattr(peer, "peer_id") else peer
        if isinstance(peer_id, PeerID):
            node_id = self.routing_table.get(peer_id=peer_id)
            if node_id is not None and node_id in self.routing_table:
                del self.routing_table[node_id]
