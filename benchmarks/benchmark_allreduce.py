#!/usr/bin/env python3
"""All-reduce bandwidth microbenchmark over RCCL/xGMI.

Measures the gradient-averaging data plane (hivemind_amd/averaging/rccl.py)
against the MI355X interconnect roofline: each GPU has 7 point-to-point xGMI
links at ~153 GB/s, so the per-GPU bus-bandwidth ceiling for ring collectives
is ~153 GB/s per link and a direct-send butterfly can in principle load all 7
links at once (BASELINE.md "all-reduce GB/s").

Modes benchmarked per tensor size:
  fp32     -- bucketed all_reduce(SUM) at full precision
  bf16     -- buckets cast to bf16 on the wire (DEFAULT for grad averaging)
  int8     -- blockwise-int8 quantized butterfly (quantize -> all-to-all ->
              dequant-accumulate -> requant -> all-gather)

Reported numbers (NCCL convention, nccl-tests/doc/PERFORMANCE.md):
  algbw = payload_bytes / time           (application view)
  busbw = algbw * 2 * (W-1) / W          (link-load view, ring all-reduce)
For the int8 butterfly the wire moves ~payload/2 bytes per direction
(1 B/element + absmax overhead, vs 4 for fp32), so its busbw is reported for
the BYTES ACTUALLY SENT as well (wire_busbw) to compare against the link
roofline.

Launch (driver-style, one process per GPU):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/benchmark_allreduce.py
Single-process runs measure loopback only (no xGMI traffic) and say so.
Writes one JSON line per (mode, size) on rank 0.
"""

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from hivemind_amd.averaging.rccl import DistributedAllReduceRunner  # noqa: E402

XGMI_LINK_GBPS = 153.0
XGMI_LINKS_PER_GPU = 7


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--sizes-mb", type=float, nargs="+", default=[8, 48, 256, 1024],
                        help="payload sizes in MiB of fp32 elements (48 ~= ALBERT-base grads)")
    parser.add_argument("--iters", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--modes", type=str, nargs="+", default=["fp32", "bf16", "int8"])
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()

    if world_size > 1:
        dist.init_process_group("nccl" if use_gpu else "gloo", rank=rank, world_size=world_size)
    device = torch.device("cuda", local_rank) if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(local_rank)

    def sync():
        if use_gpu:
            torch.cuda.synchronize(device)
        if world_size > 1:
            dist.barrier()

    for size_mb in args.sizes_mb:
        numel = int(size_mb * (1 << 20) / 4)
        payload = torch.randn(numel, device=device, dtype=torch.float32)
        for mode in args.modes:
            wire_dtype = {"fp32": None, "bf16": torch.bfloat16, "int8": None}[mode]
            codec = "blockwise_int8" if mode == "int8" else None
            if world_size == 1 and mode == "int8":
                continue  # butterfly needs peers

            def run_once():
                runner = DistributedAllReduceRunner(
                    [payload], weight=1.0, wire_dtype=wire_dtype, codec=codec,
                )
                if world_size > 1:
                    runner.run()
                else:
                    # loopback: just the local scale+copy cost
                    payload.mul_(1.0)

            for _ in range(args.warmup):
                run_once()
            sync()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                run_once()
            sync()
            elapsed = (time.perf_counter() - t0) / args.iters
            if world_size > 1:
                t = torch.tensor([elapsed], dtype=torch.float64)
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
                elapsed = float(t.item())

            payload_bytes = numel * 4
            algbw = payload_bytes / elapsed / 1e9
            busbw = algbw * 2 * (world_size - 1) / max(world_size, 1)
            # bytes this rank actually puts on the wire per round
            if mode == "int8":
                wire_bytes = 2 * numel * (world_size - 1) / world_size * (1 + 4 / 4096)
            elif mode == "bf16":
                wire_bytes = 2 * 2 * numel * (world_size - 1) / world_size
            else:
                wire_bytes = 2 * 4 * numel * (world_size - 1) / world_size
            wire_busbw = wire_bytes / elapsed / 1e9
            if rank == 0:
                print(json.dumps({
                    "bench": "allreduce_gbps",
                    "mode": mode,
                    "size_mb": size_mb,
                    "n_gpus": world_size,
                    "ms": round(elapsed * 1e3, 3),
                    "algbw_GBps": round(algbw, 2),
                    "busbw_GBps": round(busbw, 2),
                    "wire_busbw_GBps": round(wire_busbw, 2),
                    "roofline_GBps_per_link": XGMI_LINK_GBPS,
                    "roofline_GBps_all_links": XGMI_LINK_GBPS * XGMI_LINKS_PER_GPU,
                    "loopback_only": world_size == 1,
                }), flush=True)

    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
