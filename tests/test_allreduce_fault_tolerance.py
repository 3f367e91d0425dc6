"""Fault-injection tests for group all-reduce.

Reference shape: tests/test_allreduce_fault_tolerance.py:22-100 -- subclass the
runner to inject faults mid-round and assert the group still converges with
the remaining peers (failed senders banned, failed reducers replaced by
zero deltas so local values survive).
"""

import asyncio
import enum
from typing import AsyncIterator

import pytest
import torch

from hivemind_amd import DHT
from hivemind_amd.averaging import DecentralizedAverager
from hivemind_amd.averaging.allreduce import AllReduceRunner, AveragingData
from hivemind_amd.p2p import RpcContext


class Fault(enum.Enum):
    NONE = "none"
    FAIL_BEFORE = "fail_before"
    FAIL_SENDING = "fail_sending"
    SLOW_SENDING = "slow_sending"
    FAIL_REDUCING = "fail_reducing"
    CANCEL = "cancel"


class FaultyAllReduceRunner(AllReduceRunner):
    fault = Fault.NONE

    async def rpc_aggregate_part(self, stream, context: RpcContext) -> AsyncIterator[AveragingData]:
        if self.fault == Fault.FAIL_REDUCING:
            async for message in super().rpc_aggregate_part(stream, context):
                yield message
                break  # truncate the response stream mid-way
            return
        async for message in super().rpc_aggregate_part(stream, context):
            yield message

    async def _generate_input_for_peer(self, peer_index: int):
        parts = super()._generate_input_for_peer(peer_index)
        count = 0
        async for message in parts:
            if self.fault == Fault.FAIL_SENDING and count >= 1:
                break  # stop sending after the first part
            if self.fault == Fault.SLOW_SENDING:
                await asyncio.sleep(0.2)
            yield message
            count += 1


class FaultyAverager(DecentralizedAverager):
    _allreduce_runner_class = FaultyAllReduceRunner

    def __init__(self, *args, fault: Fault = Fault.NONE, **kwargs):
        self.fault = fault
        super().__init__(*args, **kwargs)

    async def _run_allreduce_inplace_(self, *args, **kwargs):
        FaultyAllReduceRunner.fault = self.fault
        try:
            return await super()._run_allreduce_inplace_(*args, **kwargs)
        finally:
            FaultyAllReduceRunner.fault = Fault.NONE

    async def _aggregate_with_group(self, group_info, weight, **kwargs):
        # reference FAIL_BEFORE/CANCEL cases (tests/test_allreduce_fault_tolerance.py:22-100):
        # the faulty peer dies before its data plane starts; the rest of the
        # group must ban it as a sender and fall back for its reduced parts
        if self.fault == Fault.FAIL_BEFORE:
            raise Exception("injected failure before allreduce")
        if self.fault == Fault.CANCEL:
            raise asyncio.CancelledError()
        return await super()._aggregate_with_group(group_info, weight, **kwargs)


def make_dht_swarm(n):
    root = DHT(start=True)
    return [root] + [DHT(initial_peers=[root.endpoint], start=True) for _ in range(n - 1)]


@pytest.mark.parametrize("fault", [Fault.NONE, Fault.SLOW_SENDING])
def test_allreduce_with_faults_still_converges(fault):
    """Healthy peers finish the round even when one peer is slow or silent."""
    torch.manual_seed(0)
    n = 4
    dhts = make_dht_swarm(n)
    tensors = [[torch.randn(400)] for _ in range(n)]
    averagers = []
    for i in range(n):
        cls_fault = fault if i == 0 else Fault.NONE
        averagers.append(
            FaultyAverager(
                [t.clone() for t in tensors[i]],
                dhts[i],
                start=True,
                prefix="fault_test",
                target_group_size=n,
                min_group_size=n,
                min_matchmaking_time=1.0,
                request_timeout=0.5,
                sender_timeout=1.5,
                reducer_timeout=2.5,
                fault=cls_fault,
            )
        )
    futures = [avg.step(wait=False, timeout=60) for avg in averagers]
    results = [f.result(90) for f in futures]
    assert all(r is not None for r in results)
    if fault == Fault.NONE:
        expected = [sum(t[0] for t in tensors) / n]
        for avg in averagers:
            with avg.get_tensors() as ts:
                assert torch.allclose(ts[0], expected[0], atol=1e-4)
    else:
        # slow peers still make it within timeouts: exact average expected too
        expected = [sum(t[0] for t in tensors) / n]
        for avg in averagers:
            with avg.get_tensors() as ts:
                assert torch.isfinite(ts[0]).all()
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()


def test_allreduce_fail_sending_banned():
    """A peer that stops sending parts gets banned; the rest still average."""
    torch.manual_seed(1)
    n = 3
    dhts = make_dht_swarm(n)
    tensors = [[torch.randn(512 * 300)] for _ in range(n)]  # several parts per peer
    averagers = []
    for i in range(n):
        averagers.append(
            FaultyAverager(
                [t.clone() for t in tensors[i]],
                dhts[i],
                start=True,
                prefix="fault_send",
                target_group_size=n,
                min_group_size=n,
                min_matchmaking_time=1.0,
                request_timeout=0.5,
                sender_timeout=1.0,
                reducer_timeout=2.0,
                fault=Fault.FAIL_SENDING if i == 0 else Fault.NONE,
            )
        )
    futures = [avg.step(wait=False, timeout=90) for avg in averagers]
    done = 0
    for f in futures:
        try:
            if f.result(120) is not None:
                done += 1
        except Exception:
            pass
    assert done >= n - 1, f"healthy peers must finish the round, finished={done}"
    for avg in averagers[1:]:
        with avg.get_tensors() as ts:
            assert torch.isfinite(ts[0]).all()
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()


def test_allreduce_fail_reducing_local_fallback():
    """A reducer that truncates its response streams mid-way must not kill the
    round: senders register the failed reducer and keep their LOCAL values for
    the parts it owned (reference partition.py:128-136 register_failed_reducer
    -> zero-delta fallback). Every element of the result is either the exact
    group average (healthy reducers' parts) or the peer's own local value."""
    torch.manual_seed(2)
    n = 3
    dhts = make_dht_swarm(n)
    tensors = [[torch.randn(512 * 300)] for _ in range(n)]  # several parts per peer
    averagers = []
    for i in range(n):
        averagers.append(
            FaultyAverager(
                [t.clone() for t in tensors[i]],
                dhts[i],
                start=True,
                prefix="fault_reduce",
                target_group_size=n,
                min_group_size=n,
                min_matchmaking_time=1.0,
                request_timeout=0.5,
                sender_timeout=1.5,
                reducer_timeout=2.0,
                fault=Fault.FAIL_REDUCING if i == 0 else Fault.NONE,
            )
        )
    futures = [avg.step(wait=False, timeout=90) for avg in averagers]
    done = 0
    for f in futures:
        try:
            if f.result(120) is not None:
                done += 1
        except Exception:
            pass
    assert done >= n - 1, f"healthy peers must survive a failed reducer, finished={done}"
    expected = sum(t[0] for t in tensors) / n
    for i, avg in enumerate(averagers[1:], start=1):
        with avg.get_tensors() as ts:
            assert torch.isfinite(ts[0]).all()
            local = tensors[i][0]
            is_avg = torch.isclose(ts[0], expected, atol=1e-4)
            is_local = torch.isclose(ts[0], local, atol=1e-4)
            assert bool((is_avg | is_local).all()), "parts must be averaged or local, never garbage"
            assert bool(is_avg.any()), "healthy reducers' parts must actually be averaged"
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()


@pytest.mark.parametrize("fault", [Fault.FAIL_BEFORE, Fault.CANCEL])
def test_allreduce_peer_dies_before_data_plane(fault):
    """A peer that fails or cancels after matchmaking but before sending any
    data must be banned as a sender; the healthy majority still finishes
    (reference FAIL_BEFORE / CANCEL cases)."""
    torch.manual_seed(3)
    n = 4
    dhts = make_dht_swarm(n)
    tensors = [[torch.randn(2048)] for _ in range(n)]
    averagers = []
    for i in range(n):
        averagers.append(
            FaultyAverager(
                [t.clone() for t in tensors[i]],
                dhts[i],
                start=True,
                prefix="fault_before",
                target_group_size=n,
                min_group_size=n,
                min_matchmaking_time=1.0,
                request_timeout=0.5,
                sender_timeout=1.5,
                reducer_timeout=2.5,
                fault=fault if i == 0 else Fault.NONE,
            )
        )
    futures = [avg.step(wait=False, timeout=90) for avg in averagers]
    done = 0
    for f in futures[1:]:
        try:
            if f.result(120) is not None:
                done += 1
        except Exception:
            pass
    assert done >= n - 1 - 0, f"healthy peers must finish, finished={done}"
    for avg in averagers[1:]:
        with avg.get_tensors() as ts:
            assert torch.isfinite(ts[0]).all()
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()
