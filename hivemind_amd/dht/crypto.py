"""Signature-protected DHT records.

Parity target: reference ``hivemind/dht/crypto.py:12-91``
(``RSASignatureValidator``: keys/subkeys containing an ``[owner:...]`` marker
may only be updated by the marker's owner; signatures appended in a
``[signature:...]`` envelope). This build signs with Ed25519
(``hivemind_amd.utils.crypto``) -- same protocol, smaller keys.
"""

from __future__ import annotations

import base64
import dataclasses
import re
from typing import Optional

from ..utils.crypto import PrivateKey, PublicKey
from ..utils.logging import get_logger
from .validation import DHTRecord, RecordValidatorBase

logger = get_logger(__name__)


class SignatureValidator(RecordValidatorBase):
    """Ed25519 ownership validator for protected DHT keys.

    A record whose key or subkey contains ``[owner:<base64 pubkey>]`` is valid
    only if its value ends with ``[signature:<sig over key|subkey|value|expiration>]``
    verifiable with that public key.
    """

    PUBLIC_KEY_FORMAT = b"[owner:_key_]"
    SIGNATURE_FORMAT = b"[signature:_value_]"
    PUBLIC_KEY_RE = re.compile(rb"\[owner:(.+?)\]")
    SIGNATURE_RE = re.compile(rb"\[signature:(.*?)\]$")

    def __init__(self, private_key: Optional[PrivateKey] = None):
        if private_key is None:
            private_key = PrivateKey.process_wide()
        self._private_key = private_key
        serialized_key = base64.b64encode(private_key.get_public_key().to_bytes())
        self._local_public_key = self.PUBLIC_KEY_FORMAT.replace(b"_key_", serialized_key)

    @property
    def local_public_key(self) -> bytes:
        """Marker to embed in owned keys/subkeys, e.g. ``b"epoch" + validator.local_public_key``."""
        return self._local_public_key

    def validate(self, record: DHTRecord) -> bool:
        public_keys = self.PUBLIC_KEY_RE.findall(record.key)
        if record.subkey:
            public_keys += self.PUBLIC_KEY_RE.findall(record.subkey)
        if not public_keys:
            return True  # unprotected record
        if len(set(public_keys)) > 1:
            logger.debug("record has conflicting owner markers")
            return False
        try:
            public_key = PublicKey.from_bytes(base64.b64decode(public_keys[0]))
        except Exception:
            return False
        signature_match = self.SIGNATURE_RE.search(record.value)
        if signature_match is None:
            logger.debug("protected record has no signature")
            return False
        try:
            signature = base64.b64decode(signature_match.group(1))
            stripped = dataclasses.replace(record, value=self.SIGNATURE_RE.sub(b"", record.value))
            return public_key.verify(self._serialize_record(stripped), signature)
        except Exception:
            # a malformed/hostile envelope must reject this record only, not
            # blow up the whole batched rpc_store/rpc_find
            logger.debug("malformed signature envelope on protected record")
            return False

    def sign_value(self, record: DHTRecord) -> bytes:
        if self._local_public_key not in record.key and (not record.subkey or self._local_public_key not in record.subkey):
            return record.value
        signature = self._private_key.sign(self._serialize_record(record))
        return record.value + self.SIGNATURE_FORMAT.replace(b"_value_", base64.b64encode(signature))

    def strip_value(self, record: DHTRecord) -> bytes:
        return self.SIGNATURE_RE.sub(b"", record.value)

    def _serialize_record(self, record: DHTRecord) -> bytes:
        return b"|".join([record.key, record.subkey or b"", record.value, str(record.expiration_time).encode()])

    @property
    def priority(self) -> int:
        return 10  # verify signatures before stripping by other validators

    def merge_with(self, other: RecordValidatorBase) -> bool:
        if not isinstance(other, SignatureValidator):
            return False
        return True  # all signature validators share behavior; keep ours


# Backward-compatible name mirroring the reference's class
Ed25519SignatureValidator = SignatureValidator
