"""Multi-process averaging over torch.distributed (gloo here, RCCL on the GPU node).

This covers the MI355X data-plane selection logic on CPU: two real processes,
each one "peer" with its own DHT, matched into one group whose membership is
exactly the torch.distributed world -> the bucketed dist all-reduce runs
instead of the RPC butterfly.
"""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp


def _dist_peer_main(rank: int, world_size: int, port: int, result_queue):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from hivemind_amd.averaging import DecentralizedAverager
        from hivemind_amd.dht import DHT

        if rank == 0:
            dht = DHT(start=True)
            endpoint = [dht.endpoint]
        else:
            dht = None
            endpoint = [None]
        dist.broadcast_object_list(endpoint, src=0)
        if rank != 0:
            dht = DHT(initial_peers=[endpoint[0]], start=True)

        tensors = [torch.full((100,), float(rank + 1)), torch.full((17,), float(10 * (rank + 1)))]
        averager = DecentralizedAverager(
            [t.clone() for t in tensors],
            dht,
            start=True,
            prefix="disttest",
            target_group_size=world_size,
            min_group_size=world_size,
            min_matchmaking_time=1.0,
            request_timeout=0.5,
        )
        result = averager.step(timeout=60)
        assert result is not None and len(result) == world_size
        expected0 = torch.full((100,), sum(range(1, world_size + 1)) / world_size)
        expected1 = torch.full((17,), 10 * sum(range(1, world_size + 1)) / world_size)
        with averager.get_tensors() as ts:
            ok = torch.allclose(ts[0], expected0, atol=1e-5) and torch.allclose(ts[1], expected1, atol=1e-5)
        result_queue.put((rank, ok, averager.last_data_plane))
        averager.shutdown()
        dht.shutdown()
    finally:
        dist.barrier()
        dist.destroy_process_group()


def test_dist_data_plane_two_processes():
    from hivemind_amd.utils.networking import get_free_port

    world_size = 2
    port = get_free_port()
    ctx = mp.get_context("spawn")
    result_queue = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_dist_peer_main, args=(rank, world_size, port, result_queue))
        for rank in range(world_size)
    ]
    for p in procs:
        p.start()
    results = []
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, f"peer process failed with exit code {p.exitcode}"
    while not result_queue.empty():
        results.append(result_queue.get())
    assert len(results) == world_size
    for rank, ok, data_plane in results:
        assert ok, f"rank {rank} got wrong average"
        assert data_plane == "rccl", f"rank {rank} used {data_plane}, expected the dist data plane"
