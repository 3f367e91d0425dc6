#!/usr/bin/env python3
"""Diagnostic: split declare_experts latency into traverse vs store phases.

Run on an idle machine; 16 in-process DHT peers make this GIL-sensitive.
"""
import random
import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from hivemind_amd import DHT
from hivemind_amd.moe.server.dht_handler import declare_experts
from hivemind_amd.utils.timed_storage import get_dht_time
import hivemind_amd.dht.node as node_mod
from hivemind_amd.dht.protocol import DHTProtocol


def main():
    num_peers = int(sys.argv[1]) if len(sys.argv) > 1 else 16

    orig_find = node_mod.DHTNode.find_nearest_nodes
    orig_store = DHTProtocol.call_store
    orig_call_find = DHTProtocol.call_find
    stats = {"find_ms": 0.0, "rpc_store": 0, "rpc_find": 0}

    async def timed_find(self, queries, **kw):
        t0 = time.perf_counter()
        r = await orig_find(self, queries, **kw)
        dt = 1000 * (time.perf_counter() - t0)
        stats["find_ms"] += dt
        print(f"  find_nearest: {len(list(queries))} queries in {dt:.1f} ms", file=sys.stderr)
        return r

    async def counted_store(self, *a, **kw):
        stats["rpc_store"] += 1
        return await orig_store(self, *a, **kw)

    async def counted_find(self, *a, **kw):
        stats["rpc_find"] += 1
        return await orig_call_find(self, *a, **kw)

    node_mod.DHTNode.find_nearest_nodes = timed_find
    DHTProtocol.call_store = counted_store
    DHTProtocol.call_find = counted_find

    root = DHT(start=True)
    peers = [DHT(initial_peers=[root.endpoint], start=True) for _ in range(num_peers - 1)]
    time.sleep(1)
    for trial in range(3):
        uids = [f"expert.{random.randint(0, 9999)}.{random.randint(0, 255)}" for _ in range(32)]
        stats.update(find_ms=0.0, rpc_store=0, rpc_find=0)
        t0 = time.perf_counter()
        declare_experts(random.choice(peers), uids, get_dht_time() + 300)
        total = 1000 * (time.perf_counter() - t0)
        print(f"declare 32 uids: total {total:.1f} ms, traverse {stats['find_ms']:.1f} ms, "
              f"rpc_find {stats['rpc_find']}, rpc_store {stats['rpc_store']}")
    for p in peers:
        p.shutdown()
    root.shutdown()


if __name__ == "__main__":
    main()
