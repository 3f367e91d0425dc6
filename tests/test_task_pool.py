"""Unit tests for the MoE server's batching queue and runtime scheduler
(reference moe/server/task_pool.py and runtime.py semantics)."""

import concurrent.futures
import threading
import time

import pytest
import torch

from hivemind_amd.moe.server.module_backend import ModuleBackend
from hivemind_amd.moe.server.runtime import Runtime
from hivemind_amd.moe.server.task_pool import TaskPool
from hivemind_amd.utils.tensor_descr import BatchTensorDescriptor


def test_task_pool_batching_respects_max_batch_size():
    pool = TaskPool(lambda x: (x * 2,), name="t", max_batch_size=8)
    futures = [pool.submit_task(torch.full((3, 2), float(i))) for i in range(5)]
    # 5 tasks x 3 rows, max 8 rows -> batches of 2 tasks (6 rows), 2, 1
    sizes = []
    while len(pool):
        batch, inputs = pool.load_batch()
        assert inputs[0].shape[0] <= 8
        sizes.append(len(batch))
        pool.send_outputs(batch, (inputs[0] * 2,))
    assert sizes == [2, 2, 1]
    for i, f in enumerate(futures):
        (out,) = f.result(1)
        assert torch.equal(out, torch.full((3, 2), 2.0 * i))
    assert pool.total_processed == 5


def test_task_pool_priority_is_oldest_task():
    pool_a, pool_b = TaskPool(None, "a", 4), TaskPool(None, "b", 4)
    assert pool_a.priority == float("inf")
    pool_a.submit_task(torch.zeros(1))
    time.sleep(0.01)
    pool_b.submit_task(torch.zeros(1))
    assert pool_a.priority < pool_b.priority  # older task -> more urgent


def test_task_pool_exception_propagates():
    pool = TaskPool(None, "e", 4)
    f = pool.submit_task(torch.zeros(2))
    batch, _ = pool.load_batch()
    pool.send_exception(batch, RuntimeError("expert died"))
    with pytest.raises(RuntimeError, match="expert died"):
        f.result(1)


def test_runtime_serves_concurrent_submitters():
    """The runtime thread batches concurrent callers and returns per-task slices."""
    backend = ModuleBackend(
        name="rt.0",
        module=torch.nn.Linear(4, 4),
        optimizer=None,
        args_schema=(BatchTensorDescriptor(4),),
        max_batch_size=64,
    )
    runtime = Runtime({"rt.0": backend})
    runtime.start()
    runtime.ready.wait(10)
    try:
        def call(i):
            x = torch.randn(2, 4)
            (out,) = backend.forward_pool.submit_task(x).result(10)
            ref = backend.module(x)
            assert torch.allclose(out, ref.detach(), atol=1e-5)

        with concurrent.futures.ThreadPoolExecutor(8) as ex:
            list(ex.map(call, range(32)))
        assert backend.forward_pool.total_processed == 32
    finally:
        runtime.shutdown()
