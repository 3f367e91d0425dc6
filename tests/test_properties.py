"""Property-based tests (hypothesis) for the serialization, DHT-metric and
compression invariants everything else is built on."""

import math

import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from hivemind_amd.dht.routing import DHTID
from hivemind_amd.utils.serializer import MSGPackSerializer
from hivemind_amd.utils.timed_storage import TimedStorage

# msgpack-representable values: scalars, bytes, strings, and nested lists /
# tuples / string-keyed dicts thereof
scalars = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(min_value=-(2**63), max_value=2**63 - 1),
    st.floats(allow_nan=False, allow_infinity=False, width=64),
    st.text(max_size=40),
    st.binary(max_size=40),
)
values = st.recursive(
    scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.lists(children, max_size=4).map(tuple),
        st.dictionaries(st.text(max_size=8), children, max_size=4),
    ),
    max_leaves=20,
)


@settings(max_examples=200, deadline=None)
@given(values)
def test_msgpack_roundtrip_property(obj):
    assert MSGPackSerializer.loads(MSGPackSerializer.dumps(obj)) == obj


@settings(max_examples=100, deadline=None)
@given(st.binary(min_size=1, max_size=64), st.binary(min_size=1, max_size=64), st.binary(min_size=1, max_size=64))
def test_dhtid_xor_metric_properties(a, b, c):
    """XOR distance is a metric: identity, symmetry, triangle inequality (the
    Kademlia routing invariants)."""
    ida, idb, idc = (DHTID.generate(source=x) for x in (a, b, c))
    assert ida.xor_distance(ida) == 0
    assert ida.xor_distance(idb) == idb.xor_distance(ida)
    if a != b:
        assert ida.xor_distance(idb) > 0
    # XOR triangle inequality: d(a,c) <= d(a,b) ^ d(b,c) <= d(a,b) + d(b,c)
    assert ida.xor_distance(idc) <= ida.xor_distance(idb) + idb.xor_distance(idc)


@settings(max_examples=50, deadline=None)
@given(
    st.lists(
        st.tuples(st.integers(0, 9), st.integers(0, 100), st.floats(1.0, 100.0)),
        min_size=1,
        max_size=30,
    )
)
def test_timed_storage_latest_expiration_wins(ops):
    """For any store sequence, get(key) returns the value with the LATEST
    expiration seen for that key (ties: first stored wins; earlier stores
    never overwrite later expirations)."""
    from hivemind_amd.utils.timed_storage import get_dht_time

    base = get_dht_time() + 1000.0
    storage = TimedStorage()
    best = {}
    for key, value, expiration in ops:
        exp = base + expiration
        stored = storage.store(key, value, exp)
        if key not in best or exp > best[key][1]:
            assert stored
            best[key] = (value, exp)
        for key, (value, exp) in best.items():
            entry = storage.get(key)
            assert entry is not None and entry.value == value and entry.expiration_time == exp


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 3), st.integers(1, 5000), st.floats(0.1, 1000.0))
def test_blockwise_quantization_error_bound(seed, numel, scale):
    """int8 blockwise absmax quantization error is bounded by absmax/127 per
    4096-block (the wire codec the averager uses for gradients)."""
    from hivemind_amd.ops import dequantize_blockwise, quantize_blockwise

    torch.manual_seed(seed)
    t = (torch.randn(numel) * scale).float()
    q, absmax = quantize_blockwise(t)
    restored = dequantize_blockwise(q, absmax).reshape(t.shape)
    blocks = t.reshape(-1)
    for b in range(0, numel, 4096):
        chunk = blocks[b : b + 4096]
        err = (restored.reshape(-1)[b : b + 4096] - chunk).abs().max()
        bound = chunk.abs().max() / 127.0 * 1.01 + 1e-8
        assert err <= bound, (err, bound)
