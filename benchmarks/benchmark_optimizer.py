#!/usr/bin/env python3
"""Decentralized optimizer benchmark (reference benchmarks/benchmark_optimizer.py:28-60):
N peers collaboratively train a small classifier on a synthetic dataset;
reports wall time to reach the target epoch and steps/s per peer.

  python benchmarks/benchmark_optimizer.py --num_peers 4 --target_epoch 6
"""

import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_peers", type=int, default=4)
    parser.add_argument("--num_clients", type=int, default=1)
    parser.add_argument("--target_batch_size", type=int, default=256)
    parser.add_argument("--batch_per_step", type=int, default=16)
    parser.add_argument("--target_epoch", type=int, default=6)
    parser.add_argument("--reuse_grad_buffers", action="store_true")
    args = parser.parse_args()

    from hivemind_amd import DHT, Optimizer

    torch.manual_seed(0)
    X = torch.randn(1024, 16)
    w = torch.randn(16, 4)
    y = (X @ w).argmax(-1)

    peers = [DHT(start=True)]
    for _ in range(args.num_peers - 1):
        peers.append(DHT(initial_peers=[peers[0].endpoint], start=True))

    results = [None] * args.num_peers

    def run_peer(idx):
        torch.manual_seed(idx)
        model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))
        client_mode = idx >= args.num_peers - args.num_clients and args.num_clients < args.num_peers
        opt = Optimizer(
            dht=peers[idx],
            run_id="bench_opt",
            target_batch_size=args.target_batch_size,
            batch_size_per_step=args.batch_per_step,
            optimizer=lambda pg: torch.optim.SGD(pg, lr=0.2),
            params=[{"params": list(model.parameters())}],
            matchmaking_time=1.0,
            averaging_timeout=60.0,
            reuse_grad_buffers=args.reuse_grad_buffers,
            client_mode=client_mode and False,  # client mode benched separately
            averager_opts=dict(request_timeout=0.5, min_group_size=2),
            tracker_opts=dict(min_refresh_period=0.2, default_refresh_period=0.5),
        )
        rng = np.random.RandomState(idx)
        steps = 0
        t0 = time.perf_counter()
        while opt.local_epoch < args.target_epoch and steps < 5000:
            sel = rng.choice(len(X), args.batch_per_step)
            loss = F.cross_entropy(model(X[sel]), y[sel])
            loss.backward()
            opt.step()
            if not args.reuse_grad_buffers:
                opt.zero_grad()
            steps += 1
        elapsed = time.perf_counter() - t0
        with torch.no_grad():
            acc = (model(X).argmax(-1) == y).float().mean().item()
        results[idx] = dict(epoch=opt.local_epoch, steps=steps, elapsed=elapsed, accuracy=round(acc, 3))
        opt.shutdown()

    threads = [threading.Thread(target=run_peer, args=(i,)) for i in range(args.num_peers)]
    t0 = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join(600)
    total = time.perf_counter() - t0

    assert all(r is not None for r in results), results
    print(
        json.dumps(
            {
                "metric": "epochs to target / wall time",
                "num_peers": args.num_peers,
                "target_epoch": args.target_epoch,
                "total_wall_s": round(total, 2),
                "sec_per_epoch": round(total / args.target_epoch, 2),
                "per_peer": results,
            }
        ),
        flush=True,
    )
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
