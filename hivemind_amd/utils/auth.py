"""Token-based RPC authorization.

Parity target: reference ``hivemind/utils/auth.py:49-211``: an authority
issues signed access tokens (username, client public key, expiration); each
request/response carries a token + nonce + signature; receivers verify the
authority signature, the client signature, the clock window (+-60 s by
default) and reject replayed nonces. ``AuthRPCWrapper`` intercepts ``rpc_*``
methods of a servicer to enforce this. Signatures are Ed25519
(hivemind_amd.utils.crypto).
"""

from __future__ import annotations

import base64
import dataclasses
import functools
import os
import time
from enum import Enum
from typing import Optional, Set

from .crypto import PrivateKey, PublicKey
from .logging import get_logger
from .serializer import MSGPackSerializer
from .timed_storage import TimedStorage, get_dht_time

logger = get_logger(__name__)

MAX_CLIENT_SERVICER_TIME_DISCREPANCY = 60.0  # seconds, reference auth.py


@dataclasses.dataclass
class AccessToken:
    username: str = ""
    public_key: bytes = b""
    expiration_time: float = 0.0
    signature: bytes = b""

    def serialized_fields(self) -> bytes:
        return b"|".join([self.username.encode(), self.public_key, str(self.expiration_time).encode()])


class AuthorizerBase:
    async def sign_request(self, request: dict, service_public_key: Optional[PublicKey]) -> dict:
        raise NotImplementedError

    async def validate_request(self, request: dict) -> bool:
        raise NotImplementedError

    async def sign_response(self, response: dict, request: dict) -> dict:
        raise NotImplementedError

    async def validate_response(self, response: dict, request: dict) -> bool:
        raise NotImplementedError


class TokenAuthorizerBase(AuthorizerBase):
    """Authorizer backed by authority-issued access tokens (reference auth.py:49)."""

    def __init__(self, local_private_key: Optional[PrivateKey] = None):
        self._local_private_key = local_private_key or PrivateKey.process_wide()
        self._local_public_key = self._local_private_key.get_public_key()
        self._local_access_token: Optional[AccessToken] = None
        self._recent_nonces: TimedStorage = TimedStorage()

    async def get_token(self) -> AccessToken:
        """Obtain an access token from the authority (override in deployments)."""
        raise NotImplementedError

    def get_local_authority_public_key(self) -> PublicKey:
        """Public key used to verify token signatures (override)."""
        raise NotImplementedError

    def is_token_valid(self, token: AccessToken) -> bool:
        if token.expiration_time < get_dht_time():
            return False
        authority_key = self.get_local_authority_public_key()
        return authority_key.verify(token.serialized_fields(), token.signature)

    async def refresh_token_if_needed(self):
        if self._local_access_token is None or self._local_access_token.expiration_time < get_dht_time() + 10:
            self._local_access_token = await self.get_token()

    @property
    def local_public_key(self) -> PublicKey:
        return self._local_public_key

    # ---------------------------------------------------------- wire helpers

    async def sign_request(self, request: dict, service_public_key: Optional[PublicKey]) -> dict:
        await self.refresh_token_if_needed()
        auth = {
            "token": dataclasses.asdict(self._local_access_token),
            "nonce": os.urandom(16),
            "time": get_dht_time(),
            "service_public_key": service_public_key.to_bytes() if service_public_key else b"",
        }
        payload = MSGPackSerializer.dumps([request, auth])
        auth["signature"] = self._local_private_key.sign(payload)
        request = dict(request)
        request["__auth__"] = auth
        return request

    async def validate_request(self, request: dict) -> bool:
        auth = request.get("__auth__")
        if auth is None:
            return False
        token = AccessToken(**auth["token"])
        if not self.is_token_valid(token):
            logger.debug("rejected request: invalid access token")
            return False
        if abs(auth["time"] - get_dht_time()) > MAX_CLIENT_SERVICER_TIME_DISCREPANCY:
            logger.debug("rejected request: clock discrepancy")
            return False
        nonce = auth["nonce"]
        if nonce in self._recent_nonces:
            logger.debug("rejected request: replayed nonce")
            return False
        signature = auth.pop("signature")
        stripped = {k: v for k, v in request.items() if k != "__auth__"}
        payload = MSGPackSerializer.dumps([stripped, auth])
        client_key = PublicKey.from_bytes(token.public_key)
        if not client_key.verify(payload, signature):
            logger.debug("rejected request: bad client signature")
            return False
        self._recent_nonces.store(nonce, True, get_dht_time() + 2 * MAX_CLIENT_SERVICER_TIME_DISCREPANCY)
        return True

    async def sign_response(self, response: dict, request: dict) -> dict:
        auth = {"nonce": request.get("__auth__", {}).get("nonce", b""), "time": get_dht_time()}
        payload = MSGPackSerializer.dumps([response, auth])
        auth["signature"] = self._local_private_key.sign(payload)
        response = dict(response)
        response["__auth__"] = auth
        return response

    async def validate_response(self, response: dict, request: dict) -> bool:
        auth = response.get("__auth__")
        if auth is None:
            return False
        if auth.get("nonce") != request.get("__auth__", {}).get("nonce"):
            logger.debug("rejected response: nonce mismatch")
            return False
        signature = dict(auth)
        sig = signature.pop("signature")
        stripped = {k: v for k, v in response.items() if k != "__auth__"}
        payload = MSGPackSerializer.dumps([stripped, signature])
        service_key_bytes = request.get("__auth__", {}).get("service_public_key", b"")
        if service_key_bytes:
            service_key = PublicKey.from_bytes(service_key_bytes)
            return service_key.verify(payload, sig)
        return True


class SelfSignedAuthorizer(TokenAuthorizerBase):
    """Test/standalone authorizer: each peer acts as its own authority
    (the reference uses this shape in tests/test_auth.py)."""

    def __init__(self, authority_private_key: Optional[PrivateKey] = None, username: str = "peer", **kwargs):
        super().__init__(**kwargs)
        self._authority_key = authority_private_key or PrivateKey.process_wide()
        self.username = username

    def get_local_authority_public_key(self) -> PublicKey:
        return self._authority_key.get_public_key()

    async def get_token(self) -> AccessToken:
        token = AccessToken(
            username=self.username,
            public_key=self.local_public_key.to_bytes(),
            expiration_time=get_dht_time() + 300.0,
        )
        token.signature = self._authority_key.sign(token.serialized_fields())
        return token


class AuthRole(Enum):
    CLIENT = 0
    SERVICER = 1


class AuthRPCWrapper:
    """Wraps a servicer so every rpc_* call validates request auth and signs
    responses (reference auth.py:172-211). The stub side uses sign_request /
    validate_response around its calls."""

    def __init__(self, stub_or_servicer, role: AuthRole, authorizer: AuthorizerBase, service_public_key: Optional[PublicKey] = None):
        self._inner = stub_or_servicer
        self._role = role
        self._authorizer = authorizer
        self._service_public_key = service_public_key

    def __getattr__(self, name: str):
        attr = getattr(self._inner, name)
        if not name.startswith("rpc_") or not callable(attr):
            return attr
        if self._role == AuthRole.SERVICER:

            @functools.wraps(attr)
            async def servicer_method(request: dict, context):
                if self._authorizer is not None:
                    if not await self._authorizer.validate_request(request):
                        raise PermissionError("request failed authorization")
                response = await attr(request, context)
                if self._authorizer is not None:
                    response = await self._authorizer.sign_response(response, request)
                return response

            return servicer_method
        else:

            @functools.wraps(attr)
            async def client_method(request: dict, *args, **kwargs):
                if self._authorizer is not None:
                    request = await self._authorizer.sign_request(request, self._service_public_key)
                response = await attr(request, *args, **kwargs)
                if self._authorizer is not None:
                    if not await self._authorizer.validate_response(response, request):
                        raise PermissionError("response failed authorization")
                return response

            return client_method
