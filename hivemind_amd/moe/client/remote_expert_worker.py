"""Bridge from sync autograd code to async RPC (reference
hivemind/moe/client/remote_expert_worker.py:26): a process-wide event loop on
which all client-side expert RPCs run. Here it simply reuses the DHT's loop
(the P2P transport is bound to it) instead of spawning another thread."""

from __future__ import annotations

import asyncio
import concurrent.futures
from typing import Awaitable, Optional, TypeVar

T = TypeVar("T")


class RemoteExpertWorker:
    """Runs coroutines on a designated event loop from synchronous code."""

    _default_loop: Optional[asyncio.AbstractEventLoop] = None

    @classmethod
    def set_default_loop(cls, loop: asyncio.AbstractEventLoop):
        cls._default_loop = loop

    @classmethod
    def run_coroutine(cls, coro: Awaitable[T], return_future: bool = False, loop: Optional[asyncio.AbstractEventLoop] = None):
        loop = loop or cls._default_loop
        assert loop is not None, "no event loop configured; pass a DHT-backed loop"
        future = asyncio.run_coroutine_threadsafe(coro, loop)
        return future if return_future else future.result()
