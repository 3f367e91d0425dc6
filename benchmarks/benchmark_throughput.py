#!/usr/bin/env python3
"""MoE server throughput benchmark.

Parity target: reference ``benchmarks/benchmark_throughput.py`` presets
(BASELINE.md headline: 28,581 samples/s forward+backward with 16 ffn experts,
hidden 1024, max_batch_size 8192, on a GTX 1080 Ti). Clients are threads
calling RemoteExpert over the local RPC stack; the server batches into the
GPU Runtime.

  python benchmarks/benchmark_throughput.py --preset default
  python benchmarks/benchmark_throughput.py --preset ffn_forward   (no backprop)
"""

import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def _client_proc_main(proc_idx, dht_endpoint, uid_list, cfg, clients_per_proc, error_queue):
    try:
        from hivemind_amd import DHT
        from hivemind_amd.moe import get_experts

        client_dht = DHT(initial_peers=[dht_endpoint], start=True, client_mode=True)
        client_experts = get_experts(client_dht, uid_list)
        assert all(e is not None for e in client_experts), "client could not resolve all experts"
        threads = []
        thread_errors = []

        def one_client(client_idx):
            try:
                torch.manual_seed(client_idx)
                for b in range(cfg["batches_per_client"]):
                    expert = client_experts[(client_idx + b) % len(client_experts)]
                    x = torch.randn(cfg["batch_size"], cfg["hidden_dim"])
                    if cfg["backprop"]:
                        x.requires_grad_(True)
                        out = expert(x)
                        out.sum().backward()
                        assert x.grad is not None
                    else:
                        with torch.no_grad():
                            out = expert(x)
                    assert out.shape == x.shape
            except BaseException:
                import traceback

                thread_errors.append(traceback.format_exc())

        for i in range(clients_per_proc):
            t = threading.Thread(target=one_client, args=(proc_idx * clients_per_proc + i,))
            t.start()
            threads.append(t)
        for t in threads:
            t.join()
        if thread_errors:
            error_queue.put(thread_errors[0])
        client_dht.shutdown()
    except BaseException:
        import traceback

        error_queue.put(traceback.format_exc())
    finally:
        os._exit(0)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--preset", choices=["default", "ffn_forward", "minimal"], default="default")
    parser.add_argument("--num-experts", type=int, default=None)
    parser.add_argument("--hidden-dim", type=int, default=None)
    parser.add_argument("--num-clients", type=int, default=None)
    parser.add_argument("--batches-per-client", type=int, default=None)
    parser.add_argument("--batch-size", type=int, default=None)
    parser.add_argument("--max-batch-size", type=int, default=None)
    parser.add_argument("--expert-cls", type=str, default="ffn",
                        help="expert class; 'nop' isolates pure transport cost")
    parser.add_argument("--num-handlers", type=int, default=8,
                        help="balanced connection-handler loops on the server (8 measured best on a 256-core MI355X node)")
    args = parser.parse_args()

    presets = {
        # reference preset: 16 experts, hid 1024, 128 clients x 16 batches x 2048 samples
        "default": dict(num_experts=16, hidden_dim=1024, num_clients=32, batches_per_client=16,
                        batch_size=2048, max_batch_size=8192, backprop=True),
        "ffn_forward": dict(num_experts=16, hidden_dim=1024, num_clients=32, batches_per_client=16,
                            batch_size=2048, max_batch_size=8192, backprop=False),
        "minimal": dict(num_experts=2, hidden_dim=64, num_clients=2, batches_per_client=2,
                        batch_size=32, max_batch_size=1024, backprop=True),
    }
    cfg = presets[args.preset]
    for key in ("num_experts", "hidden_dim", "num_clients", "batches_per_client", "batch_size", "max_batch_size"):
        cli = getattr(args, key.replace("num_experts", "num_experts"), None) if False else getattr(args, key, None)
        if cli is not None:
            cfg[key] = cli

    from hivemind_amd import DHT
    from hivemind_amd.moe import Server, get_experts

    device = "cuda" if torch.cuda.is_available() else "cpu"
    t_start = time.perf_counter()
    dht = DHT(start=True)
    uids = [f"bench_ffn.{i}" for i in range(cfg["num_experts"])]
    server = Server.create(
        dht=dht, expert_uids=uids, expert_cls=args.expert_cls, hidden_dim=cfg["hidden_dim"],
        optim_cls=(torch.optim.Adam if cfg["backprop"] else None),
        max_batch_size=cfg["max_batch_size"], device=device, start=True,
        num_connection_handlers=args.num_handlers,
    )
    t_server_ready = time.perf_counter()

    experts = get_experts(dht, uids)
    assert all(e is not None for e in experts)

    total_samples = cfg["num_clients"] * cfg["batches_per_client"] * cfg["batch_size"]

    # clients run in separate processes (their serialization must not share the
    # server's GIL) -- reference benchmark_throughput.py uses client processes too
    num_client_procs = max(1, min(8, cfg["num_clients"] // 4)) if cfg["num_clients"] >= 4 else 1
    clients_per_proc = cfg["num_clients"] // num_client_procs

    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    error_queue = ctx.SimpleQueue()
    t0 = time.perf_counter()
    procs = [
        ctx.Process(
            target=_client_proc_main,
            args=(i, dht.endpoint, uids, cfg, clients_per_proc, error_queue),
        )
        for i in range(num_client_procs)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join()
    elapsed = time.perf_counter() - t0
    assert error_queue.empty(), f"client errors: {error_queue.get()}"
    # cross-check: the server must have actually processed every client call
    # (pools count tasks; each client batch is one forward task + one backward task)
    processed_tasks = sum(
        backend.forward_pool.total_processed + backend.backward_pool.total_processed
        for backend in server.module_backends.values()
    )
    expected_tasks = cfg["num_clients"] * cfg["batches_per_client"] * (2 if cfg["backprop"] else 1)
    assert processed_tasks >= expected_tasks, f"server processed {processed_tasks} of {expected_tasks} tasks"

    result = {
        "metric": "MoE server throughput (samples/s)",
        "value": round(total_samples / elapsed, 1),
        "preset": args.preset,
        "backprop": cfg["backprop"],
        "num_experts": cfg["num_experts"],
        "hidden_dim": cfg["hidden_dim"],
        "total_samples": total_samples,
        "elapsed_s": round(elapsed, 2),
        "startup_s": round(t_server_ready - t_start, 2),
        "device": device,
        "vs_baseline_28581": round(total_samples / elapsed / 28581, 2) if cfg["backprop"] else None,
        "vs_baseline_97604_fwd": round(total_samples / elapsed / 97604, 2) if not cfg["backprop"] else None,
    }
    print(json.dumps(result), flush=True)
    server.shutdown()
    dht.shutdown()


if __name__ == "__main__":
    try:
        main()
    except BaseException:
        import traceback

        traceback.print_exc()
        os._exit(1)
    os._exit(0)
