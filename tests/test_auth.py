"""Auth protocol tests (reference tests/test_auth.py shape)."""

import asyncio

import pytest

from hivemind_amd.utils.auth import AccessToken, AuthRole, AuthRPCWrapper, SelfSignedAuthorizer
from hivemind_amd.utils.crypto import PrivateKey
from hivemind_amd.utils.timed_storage import get_dht_time


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_valid_request_and_response():
    async def main():
        authority = PrivateKey()
        client = SelfSignedAuthorizer(authority, username="alice", local_private_key=PrivateKey())
        service = SelfSignedAuthorizer(authority, username="bob", local_private_key=PrivateKey())

        request = {"field": 1}
        signed = await client.sign_request(request, service.local_public_key)
        assert await service.validate_request(signed)

        response = {"result": 2}
        signed_response = await service.sign_response(response, signed)
        assert await client.validate_response(signed_response, signed)

    run(main())


def test_replayed_nonce_rejected():
    async def main():
        authority = PrivateKey()
        client = SelfSignedAuthorizer(authority, local_private_key=PrivateKey())
        service = SelfSignedAuthorizer(authority, local_private_key=PrivateKey())
        signed = await client.sign_request({"x": 1}, service.local_public_key)
        assert await service.validate_request(dict(signed))
        assert not await service.validate_request(dict(signed))  # replay

    run(main())


def test_tampered_request_rejected():
    async def main():
        authority = PrivateKey()
        client = SelfSignedAuthorizer(authority, local_private_key=PrivateKey())
        service = SelfSignedAuthorizer(authority, local_private_key=PrivateKey())
        signed = await client.sign_request({"x": 1}, service.local_public_key)
        signed["x"] = 999  # tamper
        assert not await service.validate_request(signed)

    run(main())


def test_foreign_authority_rejected():
    async def main():
        client = SelfSignedAuthorizer(PrivateKey(), local_private_key=PrivateKey())
        service = SelfSignedAuthorizer(PrivateKey(), local_private_key=PrivateKey())  # different authority
        signed = await client.sign_request({"x": 1}, service.local_public_key)
        assert not await service.validate_request(signed)

    run(main())


def test_auth_rpc_wrapper():
    async def main():
        authority = PrivateKey()
        client_auth = SelfSignedAuthorizer(authority, local_private_key=PrivateKey())
        service_auth = SelfSignedAuthorizer(authority, local_private_key=PrivateKey())

        class Servicer:
            async def rpc_increment(self, request: dict, context) -> dict:
                return {"result": request["value"] + 1}

        servicer = AuthRPCWrapper(Servicer(), AuthRole.SERVICER, service_auth)

        class Stub:
            async def rpc_increment(self, request: dict) -> dict:
                return await servicer.rpc_increment(request, None)

        stub = AuthRPCWrapper(Stub(), AuthRole.CLIENT, client_auth, service_auth.local_public_key)
        response = await stub.rpc_increment({"value": 41})
        assert response["result"] == 42

        # unauthorized caller is rejected
        with pytest.raises(PermissionError):
            await servicer.rpc_increment({"value": 1}, None)

    run(main())
