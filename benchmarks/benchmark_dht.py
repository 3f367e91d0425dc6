#!/usr/bin/env python3
"""DHT store/get latency benchmark (reference benchmarks/benchmark_dht.py).

Reference headline: mean store 14.87 ms / mean get 6.64 ms at
1024 peers, 16384 keys, batch 64 on one machine.

  python benchmarks/benchmark_dht.py --num_peers 32 --num_experts 256 --expert_batch_size 32
"""

import argparse
import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_peers", type=int, default=32)
    parser.add_argument("--initial_peers", type=int, default=2)
    parser.add_argument("--num_experts", type=int, default=256)
    parser.add_argument("--expert_batch_size", type=int, default=32)
    parser.add_argument("--expiration", type=float, default=300.0)
    parser.add_argument("--wait_timeout", type=float, default=5.0)
    parser.add_argument("--increase_file_limit", action="store_true")
    args = parser.parse_args()

    if args.increase_file_limit:
        from hivemind_amd.utils.networking import increase_file_limit

        increase_file_limit()

    from hivemind_amd import DHT
    from hivemind_amd.moe.server.dht_handler import declare_experts, get_expert_infos
    from hivemind_amd.utils.timed_storage import get_dht_time

    print(f"spawning {args.num_peers} DHT peers...", file=sys.stderr)
    peers = [DHT(start=True, wait_timeout=args.wait_timeout)]
    for _ in range(args.num_peers - 1):
        initial = random.sample([p.endpoint for p in peers], min(args.initial_peers, len(peers)))
        peers.append(DHT(initial_peers=initial, start=True, wait_timeout=args.wait_timeout))

    expert_uids = [f"expert.{i}.{random.randint(0, 255)}" for i in range(args.num_experts)]
    random.shuffle(expert_uids)

    # stores
    store_times, successes = [], 0
    total_stores = 0
    for start in range(0, args.num_experts, args.expert_batch_size):
        batch = expert_uids[start : start + args.expert_batch_size]
        store_peer = random.choice(peers)
        t0 = time.perf_counter()
        result = declare_experts(store_peer, batch, expiration_time=get_dht_time() + args.expiration)
        dt = time.perf_counter() - t0
        store_times.append(dt / max(len(batch), 1))
        successes += sum(bool(v) for v in result.values())
        total_stores += len(result)

    # gets
    get_times, get_successes, total_gets = [], 0, 0
    for start in range(0, args.num_experts, args.expert_batch_size):
        batch = expert_uids[start : start + args.expert_batch_size]
        get_peer = random.choice(peers)
        t0 = time.perf_counter()
        infos = get_expert_infos(get_peer, batch)
        dt = time.perf_counter() - t0
        get_times.append(dt / max(len(batch), 1))
        get_successes += sum(info is not None for info in infos)
        total_gets += len(batch)

    alive = sum(p.is_alive for p in peers)
    result = {
        "metric": "DHT store/get latency",
        "num_peers": args.num_peers,
        "num_experts": args.num_experts,
        "mean_store_ms": round(1000 * sum(store_times) / len(store_times), 2),
        "mean_get_ms": round(1000 * sum(get_times) / len(get_times), 2),
        "store_success_rate": round(successes / max(total_stores, 1), 4),
        "get_success_rate": round(get_successes / max(total_gets, 1), 4),
        "node_survival_rate": round(alive / args.num_peers, 4),
        "vs_baseline_store_14.87ms": round(14.87 / (1000 * sum(store_times) / len(store_times)), 2),
        "vs_baseline_get_6.64ms": round(6.64 / (1000 * sum(get_times) / len(get_times)), 2),
    }
    print(json.dumps(result), flush=True)
    for p in peers:
        p.shutdown()


if __name__ == "__main__":
    try:
        main()
    except BaseException:
        import traceback

        traceback.print_exc()
        os._exit(1)
    os._exit(0)
