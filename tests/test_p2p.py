"""Transport-layer tests (reference shape: tests/test_p2p_daemon*.py)."""

import asyncio
from dataclasses import dataclass
from typing import AsyncIterator

import pytest

from hivemind_amd.p2p import P2P, P2PHandlerError, PeerID, RpcMessage, ServicerBase
from hivemind_amd.utils.crypto import PrivateKey


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_peer_id_roundtrip():
    key = PrivateKey()
    pid = PeerID.from_identity(key)
    assert PeerID.from_base58(pid.to_base58()) == pid
    assert pid == PeerID(pid.to_bytes())
    assert len({pid, PeerID.from_identity(key)}) == 1


def test_ed25519_sign_verify():
    key = PrivateKey()
    pub = key.get_public_key()
    sig = key.sign(b"hello world")
    assert pub.verify(b"hello world", sig)
    assert not pub.verify(b"hello worlds", sig)
    assert not pub.verify(b"hello world", sig[:-1] + bytes([sig[-1] ^ 1]))
    # round-trip through bytes
    from hivemind_amd.utils.crypto import PublicKey

    assert PublicKey.from_bytes(pub.to_bytes()).verify(b"hello world", sig)


def test_unary_call():
    async def main():
        server = await P2P.create()
        client = await P2P.create()

        async def echo(payload: bytes, ctx) -> bytes:
            return b"echo:" + payload

        server.add_unary_handler("echo", echo)
        result = await client.call_unary(server.peer_info, "echo", b"hi", timeout=5)
        assert result == b"echo:hi"
        # error propagation
        async def boom(payload: bytes, ctx) -> bytes:
            raise ValueError("nope")

        server.add_unary_handler("boom", boom)
        with pytest.raises(P2PHandlerError):
            await client.call_unary(server.peer_info, "boom", b"", timeout=5)
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_stream_call():
    async def main():
        server = await P2P.create()
        client = await P2P.create()

        async def doubler(input_aiter, ctx):
            async for item in input_aiter:
                yield item * 2

        server.add_stream_handler("doubler", doubler)

        async def inputs():
            for i in range(5):
                yield bytes([i])

        received = []
        async for item in client.call_stream(server.peer_info, "doubler", inputs()):
            received.append(item)
        assert received == [bytes([i, i]) for i in range(5)]
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_bidirectional_reuse():
    """Both sides can call each other over one connection."""

    async def main():
        a = await P2P.create()
        b = await P2P.create()

        async def name_handler(payload, ctx):
            return b"a"

        async def name_handler_b(payload, ctx):
            return b"b"

        a.add_unary_handler("whoami", name_handler)
        b.add_unary_handler("whoami", name_handler_b)
        assert await a.call_unary(b.peer_info, "whoami", b"", timeout=5) == b"b"
        # b now calls back over the same TCP connection (a's PeerID only)
        assert await b.call_unary(a.peer_id, "whoami", b"", timeout=5) == b"a"
        await a.shutdown()
        await b.shutdown()

    run(main())


@dataclass
class Ping(RpcMessage):
    text: str = ""
    number: int = 0


@dataclass
class Pong(RpcMessage):
    reply: str = ""


class PingServicer(ServicerBase):
    async def rpc_ping(self, request: Ping, context) -> Pong:
        return Pong(reply=f"{request.text}/{request.number}")

    async def rpc_count(self, request: Ping, context) -> AsyncIterator[Pong]:
        for i in range(request.number):
            yield Pong(reply=str(i))


def test_servicer_stub():
    async def main():
        server = await P2P.create()
        client = await P2P.create()
        servicer = PingServicer()
        await servicer.add_p2p_handlers(server)
        stub = PingServicer.get_stub(client, server.peer_info)
        raw = await stub.rpc_ping(Ping(text="x", number=7), timeout=5)
        assert Pong.loads(raw).reply == "x/7"
        replies = []
        async for payload in stub.rpc_count(Ping(number=3)):
            replies.append(Pong.loads(payload).reply)
        assert replies == ["0", "1", "2"]
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_circuit_relay():
    """A peer with NO listener (NATed) registers with a public relay peer and
    becomes reachable at relay://<relay>/<peer>: unary and streaming RPCs run
    through the spliced circuit (the reference's libp2p relay capability)."""

    async def main():
        relay = await P2P.create()
        # NATed peer: no inbound socket at all -- only reachable via the relay
        nated = await P2P.create(listen=False, relay_endpoint=relay.endpoint)
        assert nated.endpoint.startswith("relay://")
        assert nated.endpoint.endswith(nated.peer_id.to_base58())

        async def echo(payload: bytes, ctx) -> bytes:
            return b"via-relay:" + payload

        async def doubler(items, ctx):
            async for item in items:
                yield item * 2

        nated.add_unary_handler("echo", echo)
        nated.add_stream_handler("doubler", doubler)

        caller = await P2P.create()
        info = nated.peer_info
        assert info.endpoints and info.endpoints[0].startswith("relay://")
        result = await caller.call_unary(info, "echo", b"hello", timeout=10)
        assert result == b"via-relay:hello"

        async def inputs():
            yield b"ab"
            yield b"cd"

        out = [item async for item in caller.call_stream(info, "doubler", inputs())]
        assert out == [b"abab", b"cdcd"]

        # a second caller reuses the same advertised endpoint
        caller2 = await P2P.create(listen=False)
        result = await caller2.call_unary(info, "echo", b"x", timeout=10)
        assert result == b"via-relay:x"

        # dialing an unregistered target through the relay fails cleanly
        from hivemind_amd.p2p.transport import P2PDaemonError
        from hivemind_amd.p2p.peer_id import PeerInfo
        from hivemind_amd.utils.crypto import PrivateKey as _PK
        bogus_id = PeerID.from_identity(_PK())
        bogus = PeerInfo(bogus_id, (f"relay://{relay.endpoint}/{bogus_id.to_base58()}",))
        with pytest.raises(P2PDaemonError):
            await caller.call_unary(bogus, "echo", b"", timeout=10)

        for p in (caller, caller2, nated, relay):
            await p.shutdown()

    run(main())


def test_relay_reregisters_after_relay_connection_drop():
    """The NATed peer's keepalive re-registers after its relay connection dies
    (relay restart / NAT timeout self-healing)."""

    async def main():
        relay = await P2P.create()
        nated = await P2P.create(listen=False)
        await nated.register_with_relay(relay.endpoint, keepalive=0.3)

        async def echo(payload: bytes, ctx) -> bytes:
            return b"ok:" + payload

        nated.add_unary_handler("echo", echo)
        caller = await P2P.create()
        assert await caller.call_unary(nated.peer_info, "echo", b"1", timeout=10) == b"ok:1"

        # sever the nated<->relay control connection (simulates relay-side drop)
        conn = relay._connections.get(nated.peer_id)
        assert conn is not None
        await conn.close()
        nated_conn = nated._connections.get(relay.peer_id)
        if nated_conn is not None:
            await nated_conn.close()
        await asyncio.sleep(1.0)  # keepalive re-registers

        caller2 = await P2P.create(listen=False)
        assert await caller2.call_unary(nated.peer_info, "echo", b"2", timeout=10) == b"ok:2"
        for p in (caller, caller2, nated, relay):
            await p.shutdown()

    run(main())


def test_handshake_rejects_impersonator():
    """A peer that claims a PeerID it does not hold the key for must be
    rejected during the handshake (unauthenticated HELLO let
    any peer claim any identity). Reference gets this from libp2p TLS
    (p2p_daemon_bindings/datastructures.py:66-88)."""

    async def main():
        server = await P2P.create()
        honest = await P2P.create()
        impostor = await P2P.create()
        # the impostor claims the honest peer's identity in its HELLO but
        # cannot sign with the honest peer's key
        impostor.peer_id = honest.peer_id

        async def echo(payload: bytes, ctx) -> bytes:
            return payload

        server.add_unary_handler("echo", echo)
        with pytest.raises(Exception):
            await impostor.call_unary(server.peer_info, "echo", b"x", timeout=5)
        # the server must not have registered a connection under the forged id
        assert honest.peer_id not in server._connections
        # an honest client still works
        assert await honest.call_unary(server.peer_info, "echo", b"ok", timeout=5) == b"ok"
        await honest.shutdown()
        await impostor.shutdown()
        await server.shutdown()

    run(main())


def test_dial_by_id_rejects_wrong_peer():
    """Dialing PeerInfo(id=X, endpoint) where the endpoint's peer proves a
    different identity must fail (mismatched dials
    were only warned about)."""
    from hivemind_amd.p2p.peer_id import PeerInfo
    from hivemind_amd.p2p.transport import P2PDaemonError

    async def main():
        server = await P2P.create()
        client = await P2P.create()
        other_id = PeerID.from_identity(PrivateKey())
        forged = PeerInfo(other_id, server.peer_info.endpoints)

        async def echo(payload: bytes, ctx) -> bytes:
            return payload

        server.add_unary_handler("echo", echo)
        with pytest.raises(Exception):
            await client.call_unary(forged, "echo", b"x", timeout=5)
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_ed25519_backends_interop():
    """The libcrypto fast path and the pure-Python reference implementation
    must produce interchangeable keys and signatures (deterministic Ed25519:
    identical bytes)."""
    from hivemind_amd.utils import crypto as c

    secret = bytes(range(32))
    backend = c._LibCrypto.get()
    if backend is None:
        pytest.skip("libcrypto unavailable; pure-Python backend active")
    assert backend.public_key(secret) == c.ed25519_public_key(secret)
    msg = b"interop-check"
    assert backend.sign(secret, msg) == c.ed25519_sign(secret, msg)
    sig = backend.sign(secret, msg)
    pub = backend.public_key(secret)
    assert c.ed25519_verify(pub, msg, sig)
    assert backend.verify(pub, msg, c.ed25519_sign(secret, msg))
    assert not backend.verify(pub, msg + b"x", sig)


def test_connection_manager_evicts_idle_lru():
    """Beyond max_connections, idle least-recently-used connections close
    (reference p2pd connManager; a 1024-peer swarm exhausted fds without it)."""

    async def main():
        server = await P2P.create(max_connections=3)

        async def echo(payload: bytes, ctx) -> bytes:
            return payload

        server.add_unary_handler("echo", echo)
        clients = [await P2P.create() for _ in range(6)]
        for i, c in enumerate(clients):
            assert await c.call_unary(server.peer_info, "echo", bytes([i]), timeout=5) == bytes([i])
        await asyncio.sleep(0.2)
        live = [c for c in server._connections.values() if not c.closed.is_set()]
        assert len(live) <= 3, f"{len(live)} live connections cached, cap is 3"
        # evicted clients can still call again (fresh dial + handshake)
        assert await clients[0].call_unary(server.peer_info, "echo", b"again", timeout=5) == b"again"
        for c in clients:
            await c.shutdown()
        await server.shutdown()

    run(main())


def test_compiled_wire_codec_roundtrips():
    """The compiled per-type converters must agree with the dynamic path for
    every field shape the framework's messages use (primitives, lists of
    primitives, Optional[nested dataclass], tuples, PeerID, missing fields)."""
    from dataclasses import dataclass, field
    from typing import Dict, List, Optional, Tuple

    from hivemind_amd.p2p.servicer import RpcMessage, _from_wire, _to_wire
    from hivemind_amd.utils.serializer import MSGPackSerializer

    @dataclass
    class Inner(RpcMessage):
        blob: bytes = b""
        sizes: List[int] = field(default_factory=list)

    @dataclass
    class Outer(RpcMessage):
        name: str = ""
        flag: bool = False
        ratio: float = 0.0
        raw: bytes = b""
        ids: List[bytes] = field(default_factory=list)
        inner: Optional[Inner] = None
        inners: List[Inner] = field(default_factory=list)
        pair: Tuple[int, int] = (0, 0)
        who: Optional[PeerID] = None
        table: Dict[str, int] = field(default_factory=dict)

    msg = Outer(
        name="x", flag=True, ratio=2.5, raw=b"\x00\xff",
        ids=[b"a", b"bb"], inner=Inner(b"deep", [1, 2]),
        inners=[Inner(b"p", [3]), Inner(b"q", [])],
        pair=(7, 9), who=PeerID(b"\x12 " + b"k" * 32), table={"n": 4},
    )
    out = Outer.loads(msg.dumps())
    assert out == msg
    assert isinstance(out.inner, Inner) and isinstance(out.inners[0], Inner)
    assert isinstance(out.pair, tuple) and out.pair == (7, 9)
    assert isinstance(out.who, PeerID) and out.who == msg.who

    # None-valued Optionals survive
    empty = Outer()
    assert Outer.loads(empty.dumps()) == empty

    # cross-compat with the dynamic reference converter in both directions
    # (byte forms may differ: the serializer ext-packs tuples, the dynamic
    # path listifies them -- both must decode to the same message)
    assert Outer.loads(MSGPackSerializer.dumps(_to_wire(msg))) == msg
    assert _from_wire(Outer, MSGPackSerializer.loads(msg.dumps())) == msg

    # forward-compat: unknown wire fields are ignored, missing ones default
    wire = MSGPackSerializer.loads(msg.dumps())
    wire["__future_field__"] = 123
    del wire["table"]
    partial = Outer.loads(MSGPackSerializer.dumps(wire))
    assert partial.table == {} and partial.name == "x"


def test_dhtid_interning_and_bytes_cache():
    from hivemind_amd.dht.routing import DHTID

    a = DHTID.generate(b"seed")
    raw = a.to_bytes()
    b1, b2 = DHTID.from_bytes(raw), DHTID.from_bytes(raw)
    assert b1 is b2, "interned parse must return the same object"
    assert b1 == a and b1.to_bytes() == raw
    # non-canonical forms bypass the intern table but stay correct
    little = DHTID.from_bytes(raw[::-1], byteorder="little")
    assert little == a
    assert DHTID.from_bytes((2**255).to_bytes(32, "big")) == 2**255


def test_concurrent_dials_same_peer_share_one_connection():
    """Per-peer dial locks must dedup concurrent dials to one peer (one
    physical connection, one handshake) while not serializing dials to
    DIFFERENT peers (the old global dial lock did, and made large-swarm
    first-contact batches pay #peers x handshake sequentially)."""
    async def main():
        servers = [await P2P.create() for _ in range(6)]
        client = await P2P.create()

        async def echo(payload: bytes, ctx) -> bytes:
            return payload

        for s in servers:
            s.add_unary_handler("echo", echo)

        # burst of calls to the SAME fresh peer: exactly one dial
        await asyncio.gather(*(client.call_unary(servers[0].peer_info, "echo", b"x", timeout=10) for _ in range(8)))
        assert client.transport_stats.get("dials_ok", 0) == 1

        # burst across DIFFERENT fresh peers: one dial each, all succeed
        await asyncio.gather(
            *(client.call_unary(s.peer_info, "echo", b"y", timeout=10) for s in servers[1:] for _ in range(3))
        )
        assert client.transport_stats.get("dials_ok", 0) == len(servers)
        assert client.transport_stats.get("dials_failed", 0) == 0
        assert client.transport_stats.get("dial_time_s", 0) > 0

        await client.shutdown()
        for s in servers:
            await s.shutdown()

    run(main())


def test_idle_connection_sweeper():
    """Connections idle beyond HIVEMIND_IDLE_CONN_TIMEOUT close even below the
    LRU cap (libp2p connmgr grace sweep): long-lived processes must not
    accumulate one-shot connections toward the fd ceiling."""
    import os

    async def main():
        os.environ["HIVEMIND_IDLE_CONN_TIMEOUT"] = "1"
        try:
            server = await P2P.create()
            client = await P2P.create()
        finally:
            del os.environ["HIVEMIND_IDLE_CONN_TIMEOUT"]

        async def echo(payload: bytes, ctx) -> bytes:
            return payload

        server.add_unary_handler("echo", echo)
        assert await client.call_unary(server.peer_info, "echo", b"x", timeout=5) == b"x"
        assert server.peer_id in client._connections
        for _ in range(40):
            await asyncio.sleep(0.1)
            if server.peer_id not in client._connections:
                break
        assert server.peer_id not in client._connections, "idle connection not swept"
        assert client.transport_stats.get("idle_closed", 0) >= 1
        # the path still works after the sweep (transparent re-dial)
        assert await client.call_unary(server.peer_info, "echo", b"y", timeout=5) == b"y"
        await client.shutdown()
        await server.shutdown()

    run(main())
