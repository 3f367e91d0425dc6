"""Asyncio TCP transport: persistent multiplexed peer connections with
unary and bidirectional-streaming RPC.

This replaces the reference's out-of-tree Go libp2p daemon (``p2pd``) and its
unix-socket control protocol (``hivemind/p2p/p2p_daemon.py``,
``p2p_daemon_bindings/control.py``). Design differences, on purpose:

* One asyncio TCP server per ``P2P`` instance, run on the caller's event loop
  (no subprocess): tensor bytes never travel through this layer on the GPU
  node -- RCCL over xGMI moves them -- so the control plane stays simple.
* Persistent connections are multiplexed by 8-byte call ids; both directions
  of a single TCP connection carry calls (like libp2p stream reuse).
* Frames are length-prefixed msgpack envelopes; payloads are raw bytes.

Public surface mirrors the reference semantics: ``P2P.create``,
``add_unary_handler`` / ``add_stream_handler`` (reference
``add_protobuf_handler`` with stream flags), ``call_unary`` /
``call_stream`` (reference ``call_protobuf_handler`` /
``iterate_protobuf_handler``), ``P2PHandlerError``, ``P2PDaemonError``.
"""

from __future__ import annotations

import asyncio
import os
import struct
from typing import AsyncIterator, Awaitable, Callable, Dict, List, Optional, Tuple, Union

from ..utils.crypto import PrivateKey, PublicKey
from ..utils.logging import get_logger
from ..utils.networking import LOCALHOST, make_endpoint, split_endpoint
from ..utils.serializer import MSGPackSerializer
from .peer_id import PeerID, PeerInfo

logger = get_logger(__name__)

# frame types
_T_HELLO = 0
_T_REQUEST = 1  # unary request
_T_RESPONSE = 2  # unary response
_T_STREAM_OPEN = 3  # open a streaming call
_T_STREAM_ITEM = 4  # item caller->callee or callee->caller
_T_STREAM_HALF_CLOSE = 5  # caller finished sending
_T_STREAM_END = 6  # callee finished (stream done)
_T_ERROR = 7
_T_CANCEL = 8
_T_PING = 9
# circuit relay (the reference's libp2p relay/NAT-traversal, re-expressed for
# this transport): a NATed peer keeps an OUTBOUND connection to a public relay
# peer and registers; a third party dials the relay, which asks the NATed peer
# to dial back a fresh socket and then splices the two sockets byte-for-byte.
# After the splice both ends run the ordinary HELLO handshake -- the RPC layer
# never knows the connection is relayed.
_T_RELAY_CONNECT = 10  # caller -> relay: payload = target peer id
_T_RELAY_OK = 11  # relay -> caller: target dialed back, pipe is live
_T_RELAY_OPEN = 12  # relay -> target (over its registered conn): call_id = channel
_T_RELAY_ACCEPT = 13  # target -> relay (on a fresh socket): call_id = channel
_T_AUTH = 14  # handshake: Ed25519 signature over the remote HELLO's nonce

# Domain separator for handshake signatures; a signature produced here can
# never be replayed as a DHT record signature (dht/crypto.py) or vice versa.
_AUTH_TAG = b"hivemind-amd-handshake-v1:"

RELAY_SCHEME = "relay://"
UNIX_SCHEME = "unix:"
RELAY_REGISTER_HANDLER = "__relay__.register"

MAX_FRAME_SIZE = 256 * 1024 * 1024  # control plane sanity bound
STREAM_BUFFER_LIMIT = 64 * 1024 * 1024  # asyncio StreamReader high-water mark:
# the default 64 KiB throttles multi-MB tensor frames with hundreds of
# flow-control round-trips per read
STREAM_CHUNK_SIZE = 512 * 1024  # aligns with averaging part size

HandlerType = Union[
    Callable[[bytes, "RpcContext"], Awaitable[bytes]],
    Callable[[AsyncIterator[bytes], "RpcContext"], AsyncIterator[bytes]],
]


class P2PHandlerError(Exception):
    """Raised on the caller side when the remote handler raised."""


class P2PDaemonError(Exception):
    """Transport-level failure (peer unreachable, connection lost)."""


class RpcContext:
    """Information about the remote side of an in-flight RPC."""

    __slots__ = ("remote_id", "local_id")

    def __init__(self, remote_id: PeerID, local_id: PeerID):
        self.remote_id = remote_id
        self.local_id = local_id


def _tune_socket_buffers(writer: asyncio.StreamWriter, size: int = 4 * 1024 * 1024):
    """Unix sockets default to ~200 KB buffers; multi-MB tensor frames then pay
    many extra wakeups. Loopback TCP autotunes, AF_UNIX does not -- set both
    directions explicitly (best effort). TCP also gets TCP_NODELAY: the RPC
    protocol is small header+payload frame pairs, and Nagle + delayed-ACK
    stalls sequential DHT lookup chains."""
    try:
        import socket as _socket

        sock = writer.get_extra_info("socket")
        if sock is None:
            return
        if sock.family == _socket.AF_UNIX:
            sock.setsockopt(_socket.SOL_SOCKET, _socket.SO_SNDBUF, size)
            sock.setsockopt(_socket.SOL_SOCKET, _socket.SO_RCVBUF, size)
        elif sock.family in (_socket.AF_INET, _socket.AF_INET6):
            sock.setsockopt(_socket.IPPROTO_TCP, _socket.TCP_NODELAY, 1)
    except Exception:
        pass


class _Connection:
    """One TCP connection to a peer; carries multiplexed calls in both directions."""

    def __init__(self, p2p: "P2P", reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self.p2p = p2p
        self.reader = reader
        self.writer = writer
        _tune_socket_buffers(writer)
        self.remote_id: Optional[PeerID] = None
        self.send_lock = asyncio.Lock()
        # caller-side state for calls we initiated over this connection
        self.pending_unary: Dict[bytes, asyncio.Future] = {}
        self.inbound_streams: Dict[bytes, asyncio.Queue] = {}  # call_id -> queue of items/end
        # callee-side state for calls the remote initiated
        self.serving_streams: Dict[bytes, asyncio.Queue] = {}
        self.serving_tasks: Dict[bytes, asyncio.Task] = {}
        self.reader_task: Optional[asyncio.Task] = None
        self.closed = asyncio.Event()
        import time as _time

        self.last_used = _time.monotonic()  # for LRU eviction (connection manager)

    def is_busy(self) -> bool:
        """True if calls are in flight (must not be evicted)."""
        return bool(self.pending_unary or self.inbound_streams or self.serving_streams or self.serving_tasks)

    async def send_frame(self, ftype: int, call_id: bytes, handler: str = "", payload: bytes = b""):
        # header and payload travel as separate buffers: a multi-MB tensor part
        # is never re-wrapped through msgpack (zero extra copies on the hot path)
        import time as _time

        self.last_used = _time.monotonic()
        header = MSGPackSerializer.dumps((ftype, call_id, handler))
        async with self.send_lock:
            if len(payload) < 65536:
                # one write -> one send syscall for control-plane frames
                self.writer.write(struct.pack(">II", len(header), len(payload)) + header + payload)
            else:
                # keep multi-MB tensor payloads zero-copy
                self.writer.write(struct.pack(">II", len(header), len(payload)))
                self.writer.write(header)
                self.writer.write(payload)
            await self.writer.drain()

    async def recv_frame(self) -> Tuple[int, bytes, str, bytes]:
        lengths = await self.reader.readexactly(8)
        hlen, plen = struct.unpack(">II", lengths)
        if hlen + plen > MAX_FRAME_SIZE:
            raise P2PDaemonError(f"frame too large: {hlen + plen}")
        header = await self.reader.readexactly(hlen)
        payload = await self.reader.readexactly(plen) if plen else b""
        ftype, call_id, handler = MSGPackSerializer.loads(header)
        return ftype, call_id, handler, payload

    def fail_all(self, exc: Exception):
        for fut in self.pending_unary.values():
            if not fut.done():
                fut.set_exception(exc)
        self.pending_unary.clear()
        for queue in self.inbound_streams.values():
            queue.put_nowait(exc)
        for task in self.serving_tasks.values():
            task.cancel()
        self.closed.set()

    async def close(self):
        try:
            self.writer.close()
            await self.writer.wait_closed()
        except Exception:
            pass
        self.fail_all(P2PDaemonError("connection closed"))


class P2P:
    """A peer: TCP listener + outgoing connection cache + RPC handler registry.

    Reference counterpart: ``P2P.create`` (p2p_daemon.py:83). ``replicate`` in
    the reference shares one daemon between processes; here components in one
    process simply share the P2P object.
    """

    def __init__(self):
        self._identity: Optional[PrivateKey] = None
        self.peer_id: Optional[PeerID] = None
        self._server: Optional[asyncio.AbstractServer] = None
        self._listen_host: str = LOCALHOST
        self._port: int = 0
        self._handlers: Dict[str, Tuple[HandlerType, bool, bool]] = {}
        self._balanced: Dict[str, List[Tuple[HandlerType, bool, bool]]] = {}
        self._balanced_rr: Dict[str, int] = {}
        self._connections: Dict[PeerID, _Connection] = {}
        # per-peer dial locks: dedup concurrent dials to the SAME peer without
        # serializing dials to different peers (a global lock here made every
        # first-contact batch pay ~#peers x (connect + handshake) sequentially
        # -- the dominant latency at the 1024-peer DHT benchmark config)
        self._dial_locks: dict = {}  # peer_id -> [asyncio.Lock, refcount]
        # cap concurrent outbound dials: full per-peer parallelism for normal
        # fan-out, but no thundering herd against remote accept queues
        self._dial_semaphore = asyncio.Semaphore(
            int(os.environ.get("HIVEMIND_MAX_CONCURRENT_DIALS", "64"))
        )
        # lightweight transport observability (always on; ~ns per event):
        # counters + wall-time sums keyed by event name, read via stats()
        self.transport_stats: Dict[str, float] = {}
        self._endpoint_book: Dict[PeerID, str] = {}  # last known endpoint per peer
        self._alive = True
        self._listen = True
        # relay state: as a RELAY -- which peers registered + sockets waiting to
        # be spliced; as a NATED PEER -- the relay we are reachable through
        self._relay_served: Dict[PeerID, float] = {}
        self._relay_waiting: Dict[bytes, asyncio.Future] = {}
        self._relay_endpoint: Optional[str] = None
        self._relay_splices: set = set()
        self._uds_path: Optional[str] = None
        self._uds_server: Optional[asyncio.AbstractServer] = None

    # ------------------------------------------------------------------ setup

    @classmethod
    async def create(
        cls,
        listen_host: str = LOCALHOST,
        port: int = 0,
        identity: Optional[PrivateKey] = None,
        listen: bool = True,
        relay_endpoint: Optional[str] = None,
        max_connections: Optional[int] = None,
    ) -> "P2P":
        if max_connections is None:
            # default matches go-libp2p's connection manager high-water mark (896):
            # a cap below the swarm's working set makes every RPC pay a fresh
            # dial+handshake (measured: 1024-peer DHT store 36 -> 17 ms/key
            # when the cap stopped evicting the working set)
            max_connections = int(os.environ.get("HIVEMIND_MAX_CONNECTIONS", 896))
        self = cls()
        self._identity = identity if identity is not None else PrivateKey()
        self.peer_id = PeerID.from_identity(self._identity)
        # connection manager (reference: p2pd's connManager high-water mark):
        # beyond this many cached connections, idle LRU ones are closed --
        # without it a large DHT swarm exhausts file descriptors
        self._max_connections = max_connections
        self._listen_host = listen_host
        self._listen = listen
        if listen:
            # backlog: asyncio's default of 100 drops SYNs under the dial
            # bursts concurrent per-peer dialing produces in large swarms;
            # dropped SYNs retransmit after 1-3 s and poison handshake latency
            self._server = await asyncio.start_server(
                self._on_accept, listen_host, port, limit=STREAM_BUFFER_LIMIT, backlog=128
            )
            self._port = self._server.sockets[0].getsockname()[1]
            # any listening peer can serve as a circuit relay for NATed peers
            self.add_unary_handler(RELAY_REGISTER_HANDLER, self._rpc_relay_register)
            # same-host fast path: a unix-domain socket advertised alongside the
            # TCP endpoint; remote peers fail the unix dial instantly and fall
            # back to TCP, local peers skip the loopback TCP stack
            try:
                if os.environ.get("HIVEMIND_NO_UDS"):
                    raise RuntimeError("disabled via HIVEMIND_NO_UDS")
                import tempfile

                self._uds_path = os.path.join(
                    tempfile.gettempdir(), f"hivemind_p2p_{os.getpid()}_{self._port}.sock"
                )
                self._uds_server = await asyncio.start_unix_server(
                    self._on_accept, self._uds_path, limit=STREAM_BUFFER_LIMIT, backlog=128
                )
            except Exception as e:
                logger.debug(f"no unix-domain listener: {e!r}")
                self._uds_path, self._uds_server = None, None
        if relay_endpoint is not None:
            await self.register_with_relay(relay_endpoint)
        # libp2p-connmgr-style grace sweep: connections idle beyond the
        # timeout are closed even below the cap, so long-lived processes
        # (and many-peers-per-process benchmark hosts) do not accumulate
        # one-shot connections toward the fd ceiling. The LRU cap handles
        # bursts; the sweeper handles slow accumulation.
        self._idle_timeout = float(os.environ.get("HIVEMIND_IDLE_CONN_TIMEOUT", "120"))
        self._sweeper_task = (
            asyncio.get_event_loop().create_task(self._idle_sweeper()) if self._idle_timeout > 0 else None
        )
        return self

    async def _idle_sweeper(self):
        import time as _time

        interval = min(15.0, max(1.0, self._idle_timeout / 4))
        while self._alive:
            await asyncio.sleep(interval)
            now = _time.monotonic()
            stale = [
                c
                for c in self._connections.values()
                if not c.is_busy() and not c.closed.is_set() and now - c.last_used > self._idle_timeout
            ]
            for conn in stale:
                self._connections.pop(conn.remote_id, None)
                self.transport_stats["idle_closed"] = self.transport_stats.get("idle_closed", 0) + 1
                try:
                    await conn.close()
                except Exception:
                    pass

    @property
    def tcp_endpoint(self) -> str:
        return make_endpoint(self._listen_host, self._port)

    @property
    def endpoint(self) -> str:
        """Advertised address: "unix:<path>,host:port" when a same-host socket
        exists (dialers try components in order, unix fails instantly off-host),
        "relay://<relay>/<peer>" for NATed peers, else plain host:port."""
        if not self._listen and self._relay_endpoint is not None:
            return f"{RELAY_SCHEME}{self._relay_endpoint}/{self.peer_id.to_base58()}"
        if self._uds_path is not None:
            return f"{UNIX_SCHEME}{self._uds_path},{self.tcp_endpoint}"
        return make_endpoint(self._listen_host, self._port)

    @property
    def peer_info(self) -> PeerInfo:
        if self._listen or self._relay_endpoint is not None:
            return PeerInfo(self.peer_id, (self.endpoint,))
        return PeerInfo(self.peer_id, ())

    def get_visible_maddrs(self) -> List[str]:
        """API-compat: list of announce addresses (host:port or relay:// strings)."""
        if self._listen or self._relay_endpoint is not None:
            return [self.endpoint]
        return []

    # ----------------------------------------------------------------- relay

    async def register_with_relay(self, relay_endpoint: str, keepalive: float = 30.0) -> None:
        """NATed side: keep an outbound connection to a public relay peer and
        become reachable at ``relay://<relay_endpoint>/<our peer id>``. A
        background keepalive re-registers periodically so a relay restart or a
        dropped connection (or an expiring NAT mapping) self-heals."""
        info = await self.connect_endpoint(relay_endpoint)
        await self.call_unary(info.peer_id, RELAY_REGISTER_HANDLER, b"")
        tcp_parts = [p for p in relay_endpoint.split(",") if not p.startswith(UNIX_SCHEME)]
        self._relay_endpoint = tcp_parts[0] if tcp_parts else relay_endpoint

        async def _keepalive():
            while self._alive:
                await asyncio.sleep(keepalive)
                try:
                    peer = await self.connect_endpoint(relay_endpoint)
                    await self.call_unary(peer.peer_id, RELAY_REGISTER_HANDLER, b"", timeout=10)
                except Exception as e:
                    logger.debug(f"relay keepalive failed (will retry): {e!r}")

        task = asyncio.create_task(_keepalive())
        self._relay_splices.add(task)  # anchored + cancelled on shutdown
        task.add_done_callback(self._relay_splices.discard)

    async def _rpc_relay_register(self, _payload: bytes, ctx: "RpcContext") -> bytes:
        import time as _time

        self._relay_served[ctx.remote_id] = _time.monotonic()
        logger.debug(f"serving as relay for {ctx.remote_id}")
        return b""

    async def _relay_dial_back(self, channel: bytes):
        """NATed side: the relay asked us to open the second leg of a circuit."""
        try:
            host, port = split_endpoint(self._relay_endpoint)
            reader, writer = await asyncio.wait_for(
                asyncio.open_connection(host, port, limit=STREAM_BUFFER_LIMIT), timeout=10
            )
            conn = _Connection(self, reader, writer)
            await conn.recv_frame()  # absorb the relay's eager HELLO
            await conn.send_frame(_T_RELAY_ACCEPT, channel)
            # from here the socket is spliced to the caller: ordinary
            # authenticated handshake (the relay is only a byte pipe)
            await self._handshake_outbound(conn)
            self._register_connection(conn)
            conn.reader_task = asyncio.create_task(self._connection_loop(conn))
        except Exception as e:
            logger.warning(f"relay dial-back failed: {e!r}")

    async def _dial_via_relay(self, relay_url: str) -> _Connection:
        """Caller side: connect to ``relay://<host:port>/<target b58>``."""
        rest = relay_url[len(RELAY_SCHEME):]
        relay_ep, _, target_b58 = rest.rpartition("/")
        if not relay_ep or not target_b58:
            raise P2PDaemonError(f"malformed relay endpoint: {relay_url}")
        target = PeerID.from_base58(target_b58)
        host, port = split_endpoint(relay_ep)
        reader, writer = await asyncio.wait_for(
            asyncio.open_connection(host, port, limit=STREAM_BUFFER_LIMIT), timeout=10
        )
        conn = _Connection(self, reader, writer)
        try:
            await conn.recv_frame()  # absorb the relay's eager HELLO
            await conn.send_frame(_T_RELAY_CONNECT, b"", "", target.to_bytes())
            ftype, _, _, payload = await conn.recv_frame()
            if ftype == _T_ERROR:
                raise P2PDaemonError(payload.decode(errors="replace"))
            if ftype != _T_RELAY_OK:
                raise P2PDaemonError(f"unexpected relay response type {ftype}")
            # circuit is live: ordinary authenticated handshake with the target
            await self._handshake_outbound(conn)
            if conn.remote_id != target:
                raise P2PDaemonError(f"relayed peer proved identity {conn.remote_id}, expected {target}")
            return conn
        except Exception:
            try:
                writer.close()
            except Exception:
                pass
            raise

    async def _relay_splice(self, conn_a: _Connection, conn_b: _Connection):
        """Relay side: forward raw bytes between the two circuit legs."""

        async def pump(src: _Connection, dst: _Connection):
            try:
                while True:
                    chunk = await src.reader.read(STREAM_CHUNK_SIZE)
                    if not chunk:
                        break
                    dst.writer.write(chunk)
                    await dst.writer.drain()
            except Exception:
                pass
            finally:
                for c in (conn_a, conn_b):
                    try:
                        c.writer.close()
                    except Exception:
                        pass

        await asyncio.gather(pump(conn_a, conn_b), pump(conn_b, conn_a))

    # -------------------------------------------------------------- handlers

    def add_unary_handler(self, name: str, handler, balanced: bool = False):
        self._add_handler(name, handler, False, False, balanced)

    def add_stream_handler(self, name: str, handler, stream_input: bool = True, stream_output: bool = True, balanced: bool = False):
        self._add_handler(name, handler, stream_input, stream_output, balanced)

    def _add_handler(self, name: str, handler, stream_input: bool, stream_output: bool, balanced: bool):
        entry = (handler, stream_input, stream_output)
        if balanced:
            self._balanced.setdefault(name, []).append(entry)
            self._handlers.setdefault(name, entry)
        elif name in self._handlers and name not in self._balanced:
            raise ValueError(f"handler {name} already registered")
        else:
            self._handlers[name] = entry

    def remove_handler(self, name: str):
        self._handlers.pop(name, None)
        self._balanced.pop(name, None)

    def _resolve_handler(self, name: str) -> Tuple[HandlerType, bool, bool]:
        if name in self._balanced:
            entries = self._balanced[name]
            idx = self._balanced_rr.get(name, 0) % len(entries)
            self._balanced_rr[name] = idx + 1
            return entries[idx]
        if name not in self._handlers:
            raise KeyError(f"no handler named {name}")
        return self._handlers[name]

    # ----------------------------------------------------------- connections

    def _hello_payload(self) -> Tuple[bytes, bytes]:
        """(payload, nonce): HELLO announces PeerID + Ed25519 pubkey + a fresh
        nonce the remote must sign to prove it holds the key the PeerID hashes.

        The reference gets authenticated identity from libp2p's secure channel
        (hivemind/p2p/p2p_daemon_bindings/datastructures.py:66-88: PeerID =
        multihash(pubkey) verified by TLS); this transport proves it with an
        explicit challenge-response."""
        nonce = os.urandom(16)
        payload = MSGPackSerializer.dumps(
            [self.peer_id.to_bytes(), self._identity.get_public_key().to_bytes(), nonce]
        )
        return payload, nonce

    async def _finish_auth(self, conn: _Connection, remote_hello: bytes, my_nonce: bytes) -> PeerID:
        """Verify the remote HELLO and exchange AUTH signatures. Returns the
        PROVEN remote PeerID; raises (and the caller closes the socket) if the
        claimed identity does not hash from the presented key or the signature
        over our nonce fails."""
        try:
            claimed_id_bytes, pubkey_bytes, remote_nonce = MSGPackSerializer.loads(remote_hello)
            pubkey = PublicKey.from_bytes(pubkey_bytes)
        except Exception as e:
            raise P2PDaemonError(f"malformed HELLO: {e!r}") from e
        claimed_id = PeerID(claimed_id_bytes)
        if PeerID.from_public_key(pubkey) != claimed_id:
            raise P2PDaemonError(f"peer claims id {claimed_id} but its public key hashes differently")
        my_pub = self._identity.get_public_key().to_bytes()
        signature = self._identity.sign(_AUTH_TAG + remote_nonce + my_pub)
        await conn.send_frame(_T_AUTH, b"", "", signature)
        ftype, _, _, remote_sig = await conn.recv_frame()
        if ftype != _T_AUTH:
            raise P2PDaemonError(f"expected AUTH frame, got type {ftype}")
        if not pubkey.verify(_AUTH_TAG + my_nonce + pubkey_bytes, remote_sig):
            raise P2PDaemonError(f"peer {claimed_id} failed the handshake signature check")
        return claimed_id

    async def _handshake_outbound(self, conn: _Connection) -> None:
        """Client side of the handshake: send HELLO, read the server's HELLO,
        run the mutual AUTH exchange. Sets conn.remote_id to the proven id."""
        hello, nonce = self._hello_payload()
        await conn.send_frame(_T_HELLO, b"", "", hello)
        ftype, _, _, payload = await conn.recv_frame()
        if ftype != _T_HELLO:
            raise P2PDaemonError("expected HELLO")
        conn.remote_id = await self._finish_auth(conn, payload, nonce)

    async def _on_accept(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        conn = _Connection(self, reader, writer)
        try:
            hello, nonce = self._hello_payload()
            await conn.send_frame(_T_HELLO, b"", "", hello)
            ftype, call_id, _, payload = await conn.recv_frame()
            if ftype == _T_RELAY_CONNECT:
                await self._serve_relay_connect(conn, PeerID(payload))
                return
            if ftype == _T_RELAY_ACCEPT:
                fut = self._relay_waiting.pop(call_id, None)
                if fut is not None and not fut.done():
                    fut.set_result(conn)  # the splice task owns the socket now
                else:
                    await conn.close()
                return
            if ftype != _T_HELLO:
                raise P2PDaemonError("expected HELLO")
            conn.remote_id = await self._finish_auth(conn, payload, nonce)
        except Exception as e:
            logger.debug(f"inbound handshake failed: {e!r}")
            writer.close()
            return
        self._register_connection(conn)
        conn.reader_task = asyncio.create_task(self._connection_loop(conn))

    async def _serve_relay_connect(self, caller_conn: _Connection, target: PeerID):
        """Relay side: a caller asked to be piped to a registered NATed peer."""
        target_conn = self._connections.get(target)
        if target not in self._relay_served or target_conn is None or target_conn.closed.is_set():
            try:
                await caller_conn.send_frame(_T_ERROR, b"", "", f"no registered relay target {target}".encode())
            finally:
                await caller_conn.close()
            return
        channel = os.urandom(8)
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        self._relay_waiting[channel] = fut
        try:
            await target_conn.send_frame(_T_RELAY_OPEN, channel)
            accept_conn: _Connection = await asyncio.wait_for(fut, timeout=15)
        except Exception as e:
            self._relay_waiting.pop(channel, None)
            try:
                await caller_conn.send_frame(_T_ERROR, b"", "", f"relay target did not dial back: {e!r}".encode())
            finally:
                await caller_conn.close()
            return
        await caller_conn.send_frame(_T_RELAY_OK, b"")
        task = asyncio.current_task()
        self._relay_splices.add(task)
        try:
            await self._relay_splice(caller_conn, accept_conn)
        finally:
            self._relay_splices.discard(task)

    async def connect_endpoint(self, endpoint: str) -> PeerInfo:
        """Dial a bare endpoint and learn the peer's identity (bootstrap helper)."""
        parts = [p for p in endpoint.split(",") if p]
        last_exc: Optional[Exception] = None
        for part in parts:
            try:
                return await self._connect_endpoint_single(part)
            except Exception as e:
                last_exc = e
        raise P2PDaemonError(f"could not connect to any of {parts}: {last_exc!r}")

    async def _connect_endpoint_single(self, endpoint: str) -> PeerInfo:
        if endpoint.startswith(UNIX_SCHEME):
            reader, writer = await asyncio.wait_for(
                asyncio.open_unix_connection(endpoint[len(UNIX_SCHEME):], limit=STREAM_BUFFER_LIMIT), timeout=10
            )
        else:
            host, port = split_endpoint(endpoint)
            reader, writer = await asyncio.wait_for(asyncio.open_connection(host, port, limit=STREAM_BUFFER_LIMIT), timeout=10)
        conn = _Connection(self, reader, writer)
        try:
            await self._handshake_outbound(conn)
        except Exception:
            try:
                writer.close()
            except Exception:
                pass
            raise
        self._endpoint_book[conn.remote_id] = endpoint
        existing = self._connections.get(conn.remote_id)
        if existing is not None and not existing.closed.is_set():
            await conn.close()
            return PeerInfo(existing.remote_id, (endpoint,))
        self._register_connection(conn)
        conn.reader_task = asyncio.create_task(self._connection_loop(conn))
        return PeerInfo(conn.remote_id, (endpoint,))

    def _register_connection(self, conn: _Connection):
        existing = self._connections.get(conn.remote_id)
        if existing is None or existing.closed.is_set():
            self._connections[conn.remote_id] = conn
        self._enforce_connection_limit()

    def _enforce_connection_limit(self):
        """Close idle least-recently-used connections beyond the high-water
        mark (the reference's libp2p connection manager)."""
        if len(self._connections) <= self._max_connections:
            return
        evictable = sorted(
            (c for c in self._connections.values() if not c.is_busy() and not c.closed.is_set()),
            key=lambda c: c.last_used,
        )
        excess = len(self._connections) - self._max_connections
        st = self.transport_stats
        st["evictions"] = st.get("evictions", 0) + min(excess, len(evictable))
        for conn in evictable[:excess]:
            peer = conn.remote_id
            logger.debug(f"connection manager: evicting idle connection to {peer}")
            self._connections.pop(peer, None)
            task = asyncio.ensure_future(conn.close())
            self._relay_splices.add(task)  # anchor until done
            task.add_done_callback(self._relay_splices.discard)
        # drop closed leftovers from the table while we're here
        for peer in [p for p, c in self._connections.items() if c.closed.is_set()]:
            self._connections.pop(peer, None)

    async def _connect(self, peer: Union[PeerInfo, PeerID], endpoint: Optional[str] = None) -> _Connection:
        if isinstance(peer, PeerInfo):
            peer_id, endpoints = peer.peer_id, list(peer.endpoints)
        else:
            peer_id, endpoints = peer, []
        conn = self._connections.get(peer_id)
        if conn is not None and not conn.closed.is_set():
            return conn
        if endpoint is not None:
            endpoints.insert(0, endpoint)
        if peer_id in self._endpoint_book:
            endpoints.append(self._endpoint_book[peer_id])
        # expand "unix:/path,host:port" advertisements into individual dials
        endpoints = [part for ep in endpoints for part in ep.split(",") if part]
        entry = self._dial_locks.get(peer_id)
        if entry is None:
            entry = self._dial_locks[peer_id] = [asyncio.Lock(), 0]
        entry[1] += 1
        try:
            async with entry[0]:
                return await self._connect_locked(peer_id, endpoints)
        finally:
            entry[1] -= 1
            if entry[1] == 0:
                self._dial_locks.pop(peer_id, None)

    async def _connect_locked(self, peer_id, endpoints) -> "_Connection":
        conn = self._connections.get(peer_id)
        if conn is not None and not conn.closed.is_set():
            return conn
        last_exc: Optional[Exception] = None
        import time as _time

        _t_sem = _time.monotonic()
        async with self._dial_semaphore:
            _t_dial = _time.monotonic()
            st = self.transport_stats
            st["dial_sem_wait_s"] = st.get("dial_sem_wait_s", 0.0) + (_t_dial - _t_sem)
            for ep in endpoints:
                try:
                    if ep.startswith(RELAY_SCHEME):
                        conn = await self._dial_via_relay(ep)
                    elif ep.startswith(UNIX_SCHEME):
                        reader, writer = await asyncio.wait_for(
                            asyncio.open_unix_connection(ep[len(UNIX_SCHEME):], limit=STREAM_BUFFER_LIMIT), timeout=10
                        )
                        conn = _Connection(self, reader, writer)
                        await self._handshake_outbound(conn)
                    else:
                        host, port = split_endpoint(ep)
                        reader, writer = await asyncio.wait_for(asyncio.open_connection(host, port, limit=STREAM_BUFFER_LIMIT), timeout=10)
                        conn = _Connection(self, reader, writer)
                        await self._handshake_outbound(conn)
                    if conn.remote_id != peer_id:
                        # authenticated identity != who we meant to dial: refuse
                        # (stale endpoint book / DHT poisoning / MITM attempt)
                        await conn.close()
                        raise P2PDaemonError(f"peer at {ep} proved identity {conn.remote_id}, expected {peer_id}")
                    self._endpoint_book[conn.remote_id] = ep
                    self._register_connection(conn)
                    conn.reader_task = asyncio.create_task(self._connection_loop(conn))
                    st["dials_ok"] = st.get("dials_ok", 0) + 1
                    st["dial_time_s"] = st.get("dial_time_s", 0.0) + (_time.monotonic() - _t_dial)
                    st["dial_time_max_s"] = max(st.get("dial_time_max_s", 0.0), _time.monotonic() - _t_dial)
                    return conn
                except Exception as e:
                    st["dials_failed"] = st.get("dials_failed", 0) + 1
                    last_exc = e
                    if conn is not None:  # failed dial/handshake: release the fd
                        try:
                            conn.writer.close()
                        except Exception:
                            pass
                        conn = None
                    continue
        raise P2PDaemonError(f"could not connect to {peer_id} via {endpoints}: {last_exc}")

    async def _connection_loop(self, conn: _Connection):
        try:
            while True:
                ftype, call_id, handler, payload = await conn.recv_frame()
                if ftype == _T_REQUEST:
                    task = asyncio.create_task(self._serve_unary(conn, call_id, handler, payload))
                    conn.serving_tasks[call_id] = task
                    task.add_done_callback(lambda _t, c=call_id: conn.serving_tasks.pop(c, None))
                elif ftype == _T_RESPONSE:
                    fut = conn.pending_unary.pop(call_id, None)
                    if fut is not None and not fut.done():
                        fut.set_result(payload)
                elif ftype == _T_STREAM_OPEN:
                    input_queue: asyncio.Queue = asyncio.Queue()
                    conn.serving_streams[call_id] = input_queue
                    task = asyncio.create_task(self._serve_stream(conn, call_id, handler, input_queue))
                    conn.serving_tasks[call_id] = task
                    task.add_done_callback(lambda _t, c=call_id: conn.serving_tasks.pop(c, None))
                elif ftype == _T_STREAM_ITEM:
                    if call_id in conn.serving_streams:
                        conn.serving_streams[call_id].put_nowait(payload)
                    elif call_id in conn.inbound_streams:
                        conn.inbound_streams[call_id].put_nowait(payload)
                elif ftype == _T_STREAM_HALF_CLOSE:
                    if call_id in conn.serving_streams:
                        conn.serving_streams[call_id].put_nowait(StopAsyncIteration)
                elif ftype == _T_STREAM_END:
                    if call_id in conn.inbound_streams:
                        conn.inbound_streams[call_id].put_nowait(StopAsyncIteration)
                elif ftype == _T_ERROR:
                    err = P2PHandlerError(payload.decode("utf-8", "replace"))
                    fut = conn.pending_unary.pop(call_id, None)
                    if fut is not None and not fut.done():
                        fut.set_exception(err)
                    if call_id in conn.inbound_streams:
                        conn.inbound_streams[call_id].put_nowait(err)
                    if call_id in conn.serving_streams:
                        conn.serving_streams[call_id].put_nowait(err)
                elif ftype == _T_CANCEL:
                    task = conn.serving_tasks.pop(call_id, None)
                    if task is not None:
                        task.cancel()
                    conn.serving_streams.pop(call_id, None)
                elif ftype == _T_RELAY_OPEN:
                    task = asyncio.create_task(self._relay_dial_back(call_id))
                    self._relay_splices.add(task)  # strong ref until done (tasks are weakly held)
                    task.add_done_callback(self._relay_splices.discard)
                elif ftype == _T_PING:
                    pass
        except (asyncio.IncompleteReadError, ConnectionError, OSError):
            pass
        except asyncio.CancelledError:
            raise
        except Exception as e:
            logger.debug(f"connection loop error: {e!r}")
        finally:
            if self._connections.get(conn.remote_id) is conn:
                self._connections.pop(conn.remote_id, None)
            conn.fail_all(P2PDaemonError(f"connection to {conn.remote_id} lost"))
            try:
                conn.writer.close()  # fail_all only fails futures; release the fd
            except Exception:
                pass

    # -------------------------------------------------------------- serving

    async def _serve_unary(self, conn: _Connection, call_id: bytes, handler_name: str, payload: bytes):
        try:
            handler, stream_in, stream_out = self._resolve_handler(handler_name)
            assert not (stream_in or stream_out), f"{handler_name} is a streaming handler"
            ctx = RpcContext(conn.remote_id, self.peer_id)
            result = await handler(payload, ctx)
            await conn.send_frame(_T_RESPONSE, call_id, "", result)
        except asyncio.CancelledError:
            raise
        except Exception as e:
            logger.debug(f"handler {handler_name} failed: {e!r}")
            try:
                await conn.send_frame(_T_ERROR, call_id, "", f"{type(e).__name__}: {e}".encode())
            except Exception:
                pass

    async def _serve_stream(self, conn: _Connection, call_id: bytes, handler_name: str, input_queue: asyncio.Queue):
        async def _input_aiter() -> AsyncIterator[bytes]:
            while True:
                item = await input_queue.get()
                if item is StopAsyncIteration:
                    break
                if isinstance(item, Exception):
                    raise item
                yield item

        try:
            handler, stream_in, stream_out = self._resolve_handler(handler_name)
            ctx = RpcContext(conn.remote_id, self.peer_id)
            if stream_in and stream_out:
                aiter_out = handler(_input_aiter(), ctx)
            elif stream_in and not stream_out:
                async def _single_out():
                    yield await handler(_input_aiter(), ctx)
                aiter_out = _single_out()
            else:  # unary in, stream out
                first = await _input_aiter().__anext__()
                aiter_out = handler(first, ctx)
            async for item in aiter_out:
                await conn.send_frame(_T_STREAM_ITEM, call_id, "", item)
            await conn.send_frame(_T_STREAM_END, call_id, "", b"")
        except asyncio.CancelledError:
            raise
        except Exception as e:
            logger.debug(f"stream handler {handler_name} failed: {e!r}")
            try:
                await conn.send_frame(_T_ERROR, call_id, "", f"{type(e).__name__}: {e}".encode())
            except Exception:
                pass
        finally:
            conn.serving_streams.pop(call_id, None)

    # --------------------------------------------------------------- calling

    async def call_unary(
        self,
        peer: Union[PeerInfo, PeerID],
        handler_name: str,
        payload: bytes,
        timeout: Optional[float] = None,
    ) -> bytes:
        conn = await self._connect(peer)
        call_id = os.urandom(8)
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        conn.pending_unary[call_id] = fut
        try:
            await conn.send_frame(_T_REQUEST, call_id, handler_name, payload)
            return await asyncio.wait_for(fut, timeout)
        except (asyncio.TimeoutError, asyncio.CancelledError):
            conn.pending_unary.pop(call_id, None)
            try:
                await conn.send_frame(_T_CANCEL, call_id, "", b"")
            except Exception:
                pass
            raise

    async def call_stream(
        self,
        peer: Union[PeerInfo, PeerID],
        handler_name: str,
        input_aiter: AsyncIterator[bytes],
    ) -> AsyncIterator[bytes]:
        """Bidirectional stream: feed `input_aiter` to the remote handler, yield its output."""
        conn = await self._connect(peer)
        call_id = os.urandom(8)
        out_queue: asyncio.Queue = asyncio.Queue()
        conn.inbound_streams[call_id] = out_queue
        await conn.send_frame(_T_STREAM_OPEN, call_id, handler_name, b"")

        async def _feeder():
            try:
                async for item in input_aiter:
                    await conn.send_frame(_T_STREAM_ITEM, call_id, "", item)
                await conn.send_frame(_T_STREAM_HALF_CLOSE, call_id, "", b"")
            except asyncio.CancelledError:
                raise
            except Exception as e:
                logger.debug(f"stream feeder failed: {e!r}")

        feeder = asyncio.create_task(_feeder())
        try:
            while True:
                item = await out_queue.get()
                if item is StopAsyncIteration:
                    break
                if isinstance(item, Exception):
                    raise item
                yield item
        except (GeneratorExit, asyncio.CancelledError):
            try:
                await conn.send_frame(_T_CANCEL, call_id, "", b"")
            except Exception:
                pass
            raise
        finally:
            feeder.cancel()
            conn.inbound_streams.pop(call_id, None)

    # ------------------------------------------------------------- lifecycle

    def learn_endpoint(self, peer_id: PeerID, endpoint: str):
        """Remember where a peer can be reached (filled from DHT records)."""
        self._endpoint_book[peer_id] = endpoint

    async def ping(self, peer: Union[PeerInfo, PeerID]) -> bool:
        try:
            conn = await self._connect(peer)
            await conn.send_frame(_T_PING, b"", "", b"")
            return True
        except Exception:
            return False

    async def list_peers(self) -> List[PeerInfo]:
        return [
            PeerInfo(pid, (self._endpoint_book.get(pid, ""),))
            for pid, conn in self._connections.items()
            if not conn.closed.is_set()
        ]

    async def disconnect(self, peer_id: PeerID):
        conn = self._connections.pop(peer_id, None)
        if conn is not None:
            await conn.close()

    async def shutdown(self):
        self._alive = False
        if getattr(self, "_sweeper_task", None) is not None:
            self._sweeper_task.cancel()
        for task in list(self._relay_splices):
            task.cancel()
        self._relay_splices.clear()
        if self._server is not None:
            self._server.close()
            try:
                await self._server.wait_closed()
            except Exception:
                pass
        if self._uds_server is not None:
            self._uds_server.close()
            try:
                await self._uds_server.wait_closed()
            except Exception:
                pass
            try:
                os.unlink(self._uds_path)
            except OSError:
                pass
        for conn in list(self._connections.values()):
            await conn.close()
        self._connections.clear()

    def __repr__(self):
        return f"P2P({self.peer_id}, {self.endpoint if self._listen else 'client-only'})"
