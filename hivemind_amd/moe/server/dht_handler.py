"""Expert discovery via the DHT.

Parity target: reference ``hivemind/moe/server/dht_handler.py:22-108``: a
thread re-declares every hosted expert UID *and all its grid prefixes* each
``update_period`` (expiration-based liveness); ``get_experts`` resolves UIDs
to ``ExpertInfo``. Record layout:

* full uid:   key = uid,        value = [peer_id_b58, endpoint]
* prefix:     key = prefix,     subkey = next_coord, value = [uid, peer_id_b58, endpoint]
  (e.g. expert "ffn.3.5" writes key "ffn.3" subkey 5 and key "ffn" subkey 3,
   which is exactly what the client's beam search walks left-to-right).
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional, Sequence, Tuple

from ...dht import DHT
from ...p2p import PeerID
from ...utils.logging import get_logger
from ...utils.timed_storage import DHTExpiration, get_dht_time
from ..expert_uid import UID_DELIMITER, ExpertInfo, ExpertPrefix, ExpertUID, is_valid_uid, split_uid

logger = get_logger(__name__)


class DHTHandlerThread(threading.Thread):
    def __init__(
        self,
        module_backends,
        dht: DHT,
        update_period: float = 30.0,
        expiration: Optional[float] = None,
        endpoint: str = "",
        uid_to_peer: Optional[Dict[ExpertUID, Tuple[str, str]]] = None,
    ):
        super().__init__(name="moe-dht-handler", daemon=True)
        if expiration is None:
            expiration = max(2 * update_period, 60.0)
        self.module_backends = module_backends
        self.dht, self.update_period, self.expiration = dht, update_period, expiration
        self.endpoint = endpoint
        self.uid_to_peer = uid_to_peer  # uid -> (peer_id_b58, endpoint): balanced handlers
        self.stop_event = threading.Event()

    def run(self) -> None:
        while True:
            try:
                declare_experts(
                    self.dht,
                    list(self.module_backends.keys()),
                    expiration_time=get_dht_time() + self.expiration,
                    endpoint=self.endpoint,
                    uid_to_peer=self.uid_to_peer,
                )
            except Exception as e:
                logger.warning(f"expert declaration failed: {e!r}")
            if self.stop_event.wait(self.update_period):
                break

    def shutdown(self):
        self.stop_event.set()


def declare_experts(
    dht: DHT,
    uids: Sequence[ExpertUID],
    expiration_time: DHTExpiration,
    endpoint: str = "",
    uid_to_peer: Optional[Dict[ExpertUID, Tuple[str, str]]] = None,
) -> Dict:
    """Declare experts + grid prefixes to the DHT (reference dht_handler.py:41-79).

    ``uid_to_peer`` maps uids to (peer_id_b58, endpoint) when experts are served
    by several balanced connection-handler listeners (reference's ``balanced``
    p2pd flag, expressed as expert sharding across handler loops).
    """
    for uid in uids:
        assert is_valid_uid(uid), f"invalid expert uid: {uid}"
    default_peer = (dht.peer_id.to_base58(), endpoint or dht.endpoint)

    async def _declare(dht_obj, node):
        from ...utils.serializer import MSGPackSerializer

        keys, subkeys, values = [], [], []
        for uid in uids:
            peer_b58, ep = (uid_to_peer or {}).get(uid, default_peer)
            keys.append(uid)
            subkeys.append(None)
            values.append(MSGPackSerializer.dumps([peer_b58, ep]))
            # all proper prefixes: "ffn.3.5" -> ("ffn.3", 5), ("ffn", 3)
            prefix = uid
            while UID_DELIMITER in prefix:
                parent, coord = split_uid(prefix)
                parent = parent.rstrip(UID_DELIMITER)
                keys.append(parent)
                subkeys.append(coord)
                values.append(MSGPackSerializer.dumps([uid, peer_b58, ep]))
                prefix = parent
        return await node.store_many(keys, values, expiration_time, subkeys=subkeys)

    return dht.run_coroutine(_declare)


def get_experts(
    dht: DHT, uids: Sequence[ExpertUID], expiration_time: Optional[DHTExpiration] = None
) -> List[Optional["RemoteExpert"]]:
    """Resolve expert UIDs into RemoteExpert handles (reference dht_handler.py:81-108)."""
    from ..client.expert import RemoteExpert, create_remote_experts

    infos = get_expert_infos(dht, uids, expiration_time)
    return create_remote_experts(infos, dht)


def get_expert_infos(
    dht: DHT, uids: Sequence[ExpertUID], expiration_time: Optional[DHTExpiration] = None
) -> List[Optional[ExpertInfo]]:
    async def _get(dht_obj, node):
        from ...utils.serializer import MSGPackSerializer

        found = await node.get_many(list(uids))
        results: List[Optional[ExpertInfo]] = []
        for uid in uids:
            entry = found.get(uid)
            if entry is None or entry.value is None:
                results.append(None)
                continue
            try:
                peer_b58, endpoint = MSGPackSerializer.loads(entry.value)
                results.append(ExpertInfo(uid, PeerID.from_base58(peer_b58), endpoint))
            except Exception:
                results.append(None)
        return results

    return dht.run_coroutine(_get)
