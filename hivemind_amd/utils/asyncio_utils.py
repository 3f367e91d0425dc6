"""Asyncio helpers: background event loops, async iterator combinators.

Parity target: reference ``hivemind/utils/asyncio.py`` (switch_to_uvloop,
azip/achain/aiter_with_timeout/amap_in_executor/aenumerate, attach_event_on_finished,
enter_asynchronously, cancel_and_wait). uvloop is unavailable in this image, so
loops are plain asyncio -- adequate for a control plane whose tensor bytes move
over RCCL, not TCP.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import AsyncIterator, Awaitable, Callable, Iterable, Optional, Tuple, TypeVar

T = TypeVar("T")


def create_event_loop() -> asyncio.AbstractEventLoop:
    """New event loop for a dedicated thread (reference switches to uvloop here)."""
    return asyncio.new_event_loop()


switch_to_uvloop = create_event_loop  # API-compat alias


class EventLoopThread(threading.Thread):
    """A daemon thread running an asyncio loop; the spine of every component's
    control plane (replaces the reference's per-component mp.Process + pipe)."""

    def __init__(self, name: str = "hivemind-loop"):
        super().__init__(name=name, daemon=True)
        self._inflight: set = set()
        self.loop = create_event_loop()
        self._started = threading.Event()

    def run(self):
        asyncio.set_event_loop(self.loop)
        self.loop.call_soon(self._started.set)
        self.loop.run_forever()

    def start_and_wait(self, timeout: Optional[float] = 15.0):
        self.start()
        if not self._started.wait(timeout):
            raise TimeoutError("event loop thread failed to start")

    def run_coroutine(self, coro: Awaitable[T], timeout: Optional[float] = None) -> T:
        """Run a coroutine on this loop from another thread, wait for the result."""
        future = asyncio.run_coroutine_threadsafe(coro, self.loop)
        return future.result(timeout)

    def run_coroutine_async(self, coro: Awaitable[T]) -> concurrent.futures.Future:
        """Schedule a coroutine and return its future. The future is ALSO kept
        in a strong-reference registry until it completes: asyncio tasks are
        only weakly referenced by the loop, and a task whose entire await graph
        is an unreferenced cycle gets garbage-collected MID-AWAIT (the
        coroutine receives GeneratorExit) -- observed as a flaky DHT startup
        hang when the caller discarded this future."""
        future = asyncio.run_coroutine_threadsafe(coro, self.loop)
        self._inflight.add(future)
        future.add_done_callback(self._inflight.discard)
        return future

    def shutdown(self, timeout: float = 5.0):
        async def _stop():
            tasks = [t for t in asyncio.all_tasks(self.loop) if t is not asyncio.current_task()]
            for t in tasks:
                t.cancel()
            self.loop.stop()

        if self.loop.is_running():
            asyncio.run_coroutine_threadsafe(_stop(), self.loop)
            self.join(timeout)


async def anext_impl(aiter: AsyncIterator[T]) -> T:
    return await aiter.__anext__()


async def aiter(*args: T) -> AsyncIterator[T]:
    for arg in args:
        yield arg


async def azip(*iterables: AsyncIterator) -> AsyncIterator[Tuple]:
    """Async analog of zip -- stops at the shortest iterator."""
    iterators = [it.__aiter__() for it in iterables]
    while True:
        try:
            yield tuple(await asyncio.gather(*(it.__anext__() for it in iterators)))
        except StopAsyncIteration:
            break


async def achain(*iterables: AsyncIterator[T]) -> AsyncIterator[T]:
    for it in iterables:
        async for item in it:
            yield item


async def aenumerate(aiterable: AsyncIterator[T]) -> AsyncIterator[Tuple[int, T]]:
    index = 0
    async for item in aiterable:
        yield index, item
        index += 1


async def asingle(aiterable: AsyncIterator[T]) -> T:
    """Expect exactly one item."""
    count = 0
    result = None
    async for item in aiterable:
        count += 1
        if count > 1:
            raise ValueError("asingle: iterable yielded more than one item")
        result = item
    if count == 0:
        raise ValueError("asingle: iterable yielded no items")
    return result  # type: ignore[return-value]


async def afirst(aiterable: AsyncIterator[T], default: Optional[T] = None) -> Optional[T]:
    async for item in aiterable:
        return item
    return default


async def await_cancelled(awaitable: Awaitable) -> bool:
    try:
        await awaitable
        return False
    except (asyncio.CancelledError, concurrent.futures.CancelledError):
        return True
    except BaseException:
        return False


async def cancel_and_wait(task: asyncio.Task) -> bool:
    """Cancel a task and wait for cancellation to complete (reference asyncio.py:191)."""
    task.cancel()
    try:
        await task
        return False
    except asyncio.CancelledError:
        return True
    except BaseException:
        return False


async def aiter_with_timeout(iterable: AsyncIterator[T], timeout: Optional[float]) -> AsyncIterator[T]:
    """Iterate, raising asyncio.TimeoutError if the next item takes > timeout seconds."""
    iterator = iterable.__aiter__()
    while True:
        try:
            item = await asyncio.wait_for(iterator.__anext__(), timeout=timeout)
        except StopAsyncIteration:
            break
        yield item


async def amap_in_executor(
    func: Callable[..., T],
    *iterables: AsyncIterator,
    max_prefetch: int = 1,
    executor: Optional[ThreadPoolExecutor] = None,
) -> AsyncIterator[T]:
    """Map a blocking function over async iterables in a background executor with
    bounded prefetch -- the compression pipeline's workhorse (reference asyncio.py:149)."""
    loop = asyncio.get_event_loop()
    queue: asyncio.Queue = asyncio.Queue(max_prefetch)

    async def _producer():
        try:
            async for args in azip(*iterables):
                await queue.put(loop.run_in_executor(executor, func, *args))
        finally:
            await queue.put(None)

    producer = asyncio.create_task(_producer())
    try:
        while True:
            future = await queue.get()
            if future is None:
                break
            yield await future
        await producer
    finally:
        if not producer.done():
            producer.cancel()


async def attach_event_on_finished(iterable: AsyncIterator[T], event: asyncio.Event) -> AsyncIterator[T]:
    try:
        async for item in iterable:
            yield item
    finally:
        event.set()


class _AsyncContext:
    def __init__(self, lock, executor: Optional[ThreadPoolExecutor] = None):
        self.lock, self.executor = lock, executor

    async def __aenter__(self):
        loop = asyncio.get_event_loop()
        await loop.run_in_executor(self.executor, self.lock.__enter__)
        return self.lock

    async def __aexit__(self, *args):
        self.lock.__exit__(*args)


def enter_asynchronously(lock, executor: Optional[ThreadPoolExecutor] = None) -> _AsyncContext:
    """Acquire a blocking (threading) lock without stalling the event loop."""
    return _AsyncContext(lock, executor)


async def as_aiter(*items: T) -> AsyncIterator[T]:
    for item in items:
        yield item
