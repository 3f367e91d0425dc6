"""Unit tests for pure-logic utility modules (reference test_util_modules.py):
serializer, nested structures, timed storage, performance EMA, tensor
descriptors, async iterator helpers, and the event-loop thread."""

import asyncio
import time

import pytest
import torch

from hivemind_amd.utils.asyncio_utils import (
    EventLoopThread,
    achain,
    aenumerate,
    afirst,
    aiter_with_timeout,
    amap_in_executor,
    as_aiter,
    asingle,
    azip,
)
from hivemind_amd.utils.nested import nested_compare, nested_flatten, nested_map, nested_pack
from hivemind_amd.utils.performance_ema import PerformanceEMA
from hivemind_amd.utils.serializer import MSGPackSerializer
from hivemind_amd.utils.tensor_descr import BatchTensorDescriptor, TensorDescriptor
from hivemind_amd.utils.timed_storage import TimedStorage, ValueWithExpiration, get_dht_time


def test_msgpack_roundtrips():
    cases = [
        42,
        -(2**62),
        3.5,
        "строка",
        b"\x00\xff bytes",
        None,
        True,
        [1, [2, [3]]],
        (1, 2, (3, 4)),  # tuples survive (ext type), unlike plain msgpack
        {"a": 1, "b": [2, 3], "c": {"d": (5,)}},
        {1: "int keys", 2.5: "float keys"},
        [(), [], {}],
    ]
    for obj in cases:
        assert MSGPackSerializer.loads(MSGPackSerializer.dumps(obj)) == obj, obj
    # tuple identity: lists and tuples must NOT collapse into each other
    out = MSGPackSerializer.loads(MSGPackSerializer.dumps([(1, 2), [1, 2]]))
    assert isinstance(out[0], tuple) and isinstance(out[1], list)


def test_nested_structures():
    structure = {"a": [1, 2, (3, 4)], "b": {"c": 5}}
    flat = list(nested_flatten(structure))
    assert flat == [1, 2, 3, 4, 5]
    rebuilt = nested_pack([x * 10 for x in flat], structure)
    assert rebuilt == {"a": [10, 20, (30, 40)], "b": {"c": 50}}
    assert nested_compare(structure, rebuilt)
    assert not nested_compare(structure, {"a": [1], "b": {}})
    doubled = nested_map(lambda x: x * 2, structure)
    assert doubled["a"][2] == (6, 8)


def test_timed_storage():
    storage = TimedStorage(maxsize=3)
    now = get_dht_time()
    assert storage.store("a", 1, now + 30)
    assert storage.store("b", 2, now + 60)
    assert storage.get("a") == ValueWithExpiration(1, now + 30)
    # storing with an EARLIER expiration must not overwrite
    assert not storage.store("a", 99, now + 10)
    assert storage.get("a").value == 1
    # later expiration wins
    assert storage.store("a", 7, now + 90)
    assert storage.get("a").value == 7
    # expired entries vanish
    storage.store("gone", 3, now - 1)
    assert storage.get("gone") is None
    # maxsize evicts the soonest-to-expire entry
    storage.store("c", 3, now + 50)
    storage.store("d", 4, now + 70)
    assert len(storage) <= 3
    assert storage.get("a") is not None  # latest expiration survives
    top_key, top_value = storage.top()
    assert top_key is not None and top_value.expiration_time <= min(
        entry.expiration_time for _, entry in storage.items()
    )


def test_performance_ema():
    ema = PerformanceEMA(alpha=0.5)
    for _ in range(5):
        ema.update(task_size=10, interval=1.0)
    assert ema.samples_per_second == pytest.approx(10.0, rel=0.05)
    with ema.pause():
        time.sleep(0.05)  # paused time must not count
    ema.update(task_size=10, interval=1.0)
    assert ema.samples_per_second == pytest.approx(10.0, rel=0.05)


def test_tensor_descriptor_roundtrip():
    t = torch.randn(3, 5, dtype=torch.float32)
    descr = TensorDescriptor.from_tensor(t)
    z = descr.make_zeros()
    assert z.shape == t.shape and z.dtype == t.dtype
    bt = BatchTensorDescriptor.from_tensor(torch.randn(7, 4, 2))
    assert bt.make_zeros(5).shape == (5, 4, 2)


def test_async_iterator_helpers():
    async def run():
        assert [x async for x in as_aiter(1, 2, 3)] == [1, 2, 3]
        assert [x async for x in achain(as_aiter(1), as_aiter(2, 3))] == [1, 2, 3]
        assert [x async for x in azip(as_aiter(1, 2), as_aiter("a", "b"))] == [(1, "a"), (2, "b")]
        assert [x async for x in aenumerate(as_aiter("x", "y"))] == [(0, "x"), (1, "y")]
        assert await asingle(as_aiter(9)) == 9
        with pytest.raises(ValueError):
            await asingle(as_aiter(1, 2))
        assert await afirst(as_aiter()) is None
        assert await afirst(as_aiter(5, 6)) == 5
        mapped = [x async for x in amap_in_executor(lambda a: a * 2, as_aiter(1, 2, 3))]
        assert mapped == [2, 4, 6]

        async def slow():
            yield 1
            await asyncio.sleep(10)
            yield 2

        with pytest.raises(asyncio.TimeoutError):
            _ = [x async for x in aiter_with_timeout(slow(), timeout=0.1)]

    asyncio.run(run())


def test_event_loop_thread():
    loop = EventLoopThread()
    loop.start_and_wait()
    try:
        async def coro():
            await asyncio.sleep(0.01)
            return 123

        assert loop.run_coroutine(coro(), timeout=5) == 123
        fut = loop.run_coroutine_async(coro())
        assert fut.result(5) == 123
    finally:
        loop.shutdown()
