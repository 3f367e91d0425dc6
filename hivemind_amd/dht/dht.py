"""DHT facade: a DHTNode on a background event-loop thread.

Parity target: reference ``hivemind/dht/dht.py:22-337``. The reference runs
the node in a forked ``mp.Process`` with pipe+MPFuture RPC because its peers
are CPU processes; this framework is one-process-per-GPU, so the DHT runs as
an asyncio loop on a daemon *thread* of the same process -- no pickling, no
fork, and the GPU-owning main thread is never blocked (SURVEY.md §7 "hard
parts": avoid the 3-process architecture). The public API is preserved:
``DHT(initial_peers, start=True)``, ``get``/``store``/``run_coroutine``
(each with ``return_future=False|True``), ``get_visible_maddrs``.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
from functools import partial
from typing import Any, Awaitable, Callable, Dict, Iterable, List, Optional, Sequence, TypeVar, Union

from ..p2p import P2P, PeerID, PeerInfo
from ..utils.logging import get_logger
from ..utils.asyncio_utils import EventLoopThread
from ..utils.serializer import MSGPackSerializer
from ..utils.timed_storage import DHTExpiration, ValueWithExpiration, get_dht_time
from .node import DHTNode
from .routing import DHTID, DHTKey, Subkey
from .storage import DictionaryDHTValue
from .validation import CompositeValidator, RecordValidatorBase

logger = get_logger(__name__)

ReturnType = TypeVar("ReturnType")


class DHT:
    """High-level DHT interface used by averaging, optim and MoE layers."""

    def __init__(
        self,
        initial_peers: Sequence[Union[str, PeerInfo]] = (),
        *,
        start: bool = False,
        p2p: Optional[P2P] = None,
        daemon: bool = True,
        num_workers: int = 4,
        record_validators: Iterable[RecordValidatorBase] = (),
        shutdown_timeout: float = 3.0,
        await_ready: bool = True,
        **kwargs,
    ):
        self.initial_peers = list(initial_peers)
        self.kwargs = kwargs
        self.num_workers = num_workers
        self._record_validator = CompositeValidator(record_validators)
        self.shutdown_timeout = shutdown_timeout
        self._p2p_external = p2p
        self._loop_thread: Optional[EventLoopThread] = None
        self._node: Optional[DHTNode] = None
        self._ready = concurrent.futures.Future()
        if start:
            self.run_in_background(await_ready=await_ready)

    # ------------------------------------------------------------- lifecycle

    def run_in_background(self, await_ready: bool = True, timeout: Optional[float] = 30.0):
        if self._loop_thread is not None:
            raise RuntimeError("DHT is already running")
        self._loop_thread = EventLoopThread(name="hivemind-dht")
        self._loop_thread.start_and_wait()

        async def _startup():
            try:
                self._node = await DHTNode.create(
                    p2p=self._p2p_external,
                    initial_peers=self.initial_peers,
                    num_workers=self.num_workers,
                    record_validator=self._record_validator,
                    **self.kwargs,
                )
                self._ready.set_result(None)
            except Exception as e:
                logger.exception("DHT failed to start")
                self._ready.set_exception(e)

        self._loop_thread.run_coroutine_async(_startup())
        if await_ready:
            self.wait_until_ready(timeout)
        return self

    start = run_in_background  # reference-compatible alias

    def wait_until_ready(self, timeout: Optional[float] = None):
        self._ready.result(timeout)

    @property
    def is_alive(self) -> bool:
        return self._loop_thread is not None and self._loop_thread.is_alive()

    def shutdown(self):
        if self._loop_thread is None:
            return
        try:
            if self._node is not None:
                self._loop_thread.run_coroutine(self._node.shutdown(), timeout=self.shutdown_timeout)
        except Exception:
            pass
        self._loop_thread.shutdown(self.shutdown_timeout)
        self._loop_thread = None

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass

    # ------------------------------------------------------------ properties

    @property
    def peer_id(self) -> PeerID:
        self.wait_until_ready()
        return self._node.p2p.peer_id

    @property
    def node_id(self) -> DHTID:
        self.wait_until_ready()
        return self._node.node_id

    @property
    def endpoint(self) -> str:
        self.wait_until_ready()
        return self._node.p2p.endpoint

    def get_visible_maddrs(self, latest: bool = False) -> List[PeerInfo]:
        """Addresses other peers can use as ``initial_peers``."""
        self.wait_until_ready()
        return [self._node.p2p.peer_info]

    @property
    def peer_info(self) -> PeerInfo:
        self.wait_until_ready()
        return self._node.p2p.peer_info

    # ------------------------------------------------------------- get/store

    def get(
        self, key: DHTKey, latest: bool = False, return_future: bool = False, **kwargs
    ) -> Union[Optional[ValueWithExpiration], concurrent.futures.Future]:
        """Find the freshest value for ``key``; deserializes stored python values."""
        future = self._run(self._get(key, latest, **kwargs))
        return future if return_future else future.result()

    async def _get(self, key: DHTKey, latest: bool, **kwargs) -> Optional[ValueWithExpiration]:
        result = await self._node.get(key, latest=latest, **kwargs)
        return _deserialize_result(result, self._record_validator)

    def store(
        self,
        key: DHTKey,
        value: Any,
        expiration_time: DHTExpiration,
        subkey: Optional[Subkey] = None,
        return_future: bool = False,
        **kwargs,
    ) -> Union[bool, concurrent.futures.Future]:
        future = self._run(self._store(key, value, expiration_time, subkey, **kwargs))
        return future if return_future else future.result()

    async def _store(self, key, value, expiration_time, subkey, **kwargs) -> bool:
        return await self._node.store(
            key, MSGPackSerializer.dumps(value), expiration_time, subkey=subkey, **kwargs
        )

    def run_coroutine(
        self,
        coro: Callable[["DHT", DHTNode], Awaitable[ReturnType]],
        return_future: bool = False,
    ) -> Union[ReturnType, concurrent.futures.Future]:
        """Execute ``coro(self, node)`` on the DHT event loop (reference dht.py:240)."""
        future = self._run(self._wrap_coroutine(coro))
        return future if return_future else future.result()

    async def _wrap_coroutine(self, coro):
        return await coro(self, self._node)

    def _run(self, coro) -> concurrent.futures.Future:
        self.wait_until_ready()
        return self._loop_thread.run_coroutine_async(coro)

    # -------------------------------------------------------------- plumbing

    def replicate_p2p(self) -> P2P:
        """Share this DHT's transport with other components in the process
        (reference dht.py:320: other components reuse the p2pd daemon)."""
        self.wait_until_ready()
        return self._node.p2p

    @property
    def loop(self):
        self.wait_until_ready()
        return self._loop_thread.loop

    def add_validators(self, record_validators: Iterable[RecordValidatorBase]):
        self.wait_until_ready()
        self._record_validator.extend(record_validators)


def _deserialize_result(
    result: Optional[ValueWithExpiration], validator: Optional[RecordValidatorBase] = None
) -> Optional[ValueWithExpiration]:
    """Record validators sign stored values in place; strip signatures before
    deserializing (values served from local storage arrive signed)."""
    from .validation import DHTRecord

    def _strip(raw: bytes, expiration: float) -> bytes:
        if validator is None or not isinstance(raw, bytes):
            return raw
        try:
            return validator.strip_value(DHTRecord(b"", b"", raw, expiration))
        except Exception:
            return raw

    if result is None:
        return None
    value, expiration = result
    if isinstance(value, DictionaryDHTValue):
        out: Dict[Any, ValueWithExpiration] = {}
        for subkey, (sub_value, sub_expiration) in value.items():
            sub_value = _strip(sub_value, sub_expiration)
            try:
                out[subkey] = ValueWithExpiration(MSGPackSerializer.loads(sub_value), sub_expiration)
            except Exception:
                out[subkey] = ValueWithExpiration(sub_value, sub_expiration)
        return ValueWithExpiration(out, expiration)
    value = _strip(value, expiration)
    try:
        return ValueWithExpiration(MSGPackSerializer.loads(value), expiration)
    except Exception:
        return ValueWithExpiration(value, expiration)
