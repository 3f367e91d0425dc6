"""Peer identity: Ed25519 pubkey -> sha256 multihash -> base58 PeerID.

Parity target: reference ``hivemind/p2p/p2p_daemon_bindings/datastructures.py:66-134``
(PeerID with base58 text form derived from the identity key, PeerInfo).
base58 is implemented inline (the pip package is unavailable).
"""

from __future__ import annotations

import hashlib
from dataclasses import dataclass, field
from typing import Sequence

from ..utils.crypto import PrivateKey, PublicKey

_B58_ALPHABET = "123456789ABCDEFGHJKLMNPQRSTUVWXYZabcdefghijkmnopqrstuvwxyz"
_B58_INDEX = {c: i for i, c in enumerate(_B58_ALPHABET)}


def b58encode(data: bytes) -> str:
    n = int.from_bytes(data, "big")
    out = []
    while n > 0:
        n, rem = divmod(n, 58)
        out.append(_B58_ALPHABET[rem])
    pad = 0
    for byte in data:
        if byte == 0:
            pad += 1
        else:
            break
    return "1" * pad + "".join(reversed(out))


def b58decode(text: str) -> bytes:
    n = 0
    for char in text:
        n = n * 58 + _B58_INDEX[char]
    raw = n.to_bytes((n.bit_length() + 7) // 8, "big")
    pad = 0
    for char in text:
        if char == "1":
            pad += 1
        else:
            break
    return b"\x00" * pad + raw


class PeerID:
    """sha256 multihash of the peer's public key, printed in base58.

    The base58 form is computed lazily: DHT traversal deserializes thousands
    of PeerIDs per lookup and an eager b58encode was ~16% of store latency."""

    __slots__ = ("_bytes", "_b58_cached")

    def __init__(self, peer_id_bytes: bytes):
        self._bytes = peer_id_bytes
        self._b58_cached = None

    @property
    def _b58(self) -> str:
        if self._b58_cached is None:
            self._b58_cached = b58encode(self._bytes)
        return self._b58_cached

    @classmethod
    def from_identity(cls, private_key: PrivateKey) -> "PeerID":
        return cls.from_public_key(private_key.get_public_key())

    @classmethod
    def from_public_key(cls, public_key: PublicKey) -> "PeerID":
        digest = hashlib.sha256(public_key.to_bytes()).digest()
        # multihash prefix: 0x12 = sha2-256, 0x20 = 32 bytes
        return cls(b"\x12\x20" + digest)

    @classmethod
    def from_base58(cls, b58: str) -> "PeerID":
        return cls(b58decode(b58))

    def to_bytes(self) -> bytes:
        return self._bytes

    def to_base58(self) -> str:
        return self._b58

    def __repr__(self) -> str:
        return f"<PeerID {self._b58[:12]}…>"

    def __str__(self) -> str:
        return self._b58

    def __eq__(self, other) -> bool:
        if isinstance(other, PeerID):
            return self._bytes == other._bytes
        if isinstance(other, bytes):
            return self._bytes == other
        if isinstance(other, str):
            return self._b58 == other
        return False

    def __hash__(self) -> int:
        return hash(self._bytes)

    def __lt__(self, other: "PeerID") -> bool:
        return self._bytes < other._bytes


@dataclass(frozen=True)
class PeerInfo:
    """A peer's identity plus the endpoints ("host:port") where it listens."""

    peer_id: PeerID
    endpoints: Sequence[str] = field(default_factory=tuple)

    def __repr__(self):
        return f"PeerInfo({self.peer_id}, {list(self.endpoints)})"
