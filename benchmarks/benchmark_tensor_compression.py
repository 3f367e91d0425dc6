#!/usr/bin/env python3
"""Tensor codec throughput (reference benchmarks/benchmark_tensor_compression.py).

  python benchmarks/benchmark_tensor_compression.py --size 10000000
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--size", type=int, default=10_000_000)
    parser.add_argument("--repeats", type=int, default=3)
    args = parser.parse_args()

    from hivemind_amd.compression import CompressionType, deserialize_torch_tensor, serialize_torch_tensor

    torch.manual_seed(0)
    X = torch.randn(args.size)
    results = {}
    for ct in CompressionType:
        best_compress = best_extract = float("inf")
        for _ in range(args.repeats):
            t0 = time.perf_counter()
            serialized = serialize_torch_tensor(X, ct)
            best_compress = min(best_compress, time.perf_counter() - t0)
            t0 = time.perf_counter()
            restored = deserialize_torch_tensor(serialized)
            best_extract = min(best_extract, time.perf_counter() - t0)
        err = (restored - X).abs().mean().item()
        results[ct.name] = {
            "compress_GBps": round(args.size * 4 / best_compress / 1e9, 2),
            "extract_GBps": round(args.size * 4 / best_extract / 1e9, 2),
            "wire_bytes_per_el": round(len(serialized.buffer) / args.size, 3),
            "mean_abs_err": round(err, 5),
        }
    print(json.dumps({"metric": "codec throughput (CPU)", "size": args.size, "codecs": results}, indent=2))

    if torch.cuda.is_available():
        from hivemind_amd.ops import compress_fp16, decompress_fp16, dequantize_blockwise, quantize_blockwise

        Xg = X.cuda()
        torch.cuda.synchronize()
        gpu_results = {}
        for name, compress, extract in [
            ("FLOAT16_HIP", lambda t: compress_fp16(t), lambda c: decompress_fp16(c)),
            ("BLOCKWISE_8BIT_HIP", lambda t: quantize_blockwise(t), lambda c: dequantize_blockwise(*c)),
        ]:
            for _ in range(2):
                compressed = compress(Xg)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.repeats):
                compressed = compress(Xg)
            torch.cuda.synchronize()
            ct = (time.perf_counter() - t0) / args.repeats
            t0 = time.perf_counter()
            for _ in range(args.repeats):
                extract(compressed)
            torch.cuda.synchronize()
            et = (time.perf_counter() - t0) / args.repeats
            gpu_results[name] = {
                "compress_GBps": round(args.size * 4 / ct / 1e9, 2),
                "extract_GBps": round(args.size * 4 / et / 1e9, 2),
            }
        print(json.dumps({"metric": "codec throughput (MI355X HIP kernels)", "codecs": gpu_results}, indent=2))


if __name__ == "__main__":
    main()
