#!/usr/bin/env python3
"""Model-only throughput: ALBERT fwd/bwd without the distributed optimizer.

Separates model compute from optimizer/averaging overhead in bench.py.
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
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--batch", type=int, default=32)
    parser.add_argument("--seq-len", type=int, default=512)
    parser.add_argument("--model", type=str, default="albert-base")
    parser.add_argument("--no-backward", action="store_true")
    parser.add_argument("--adam", action="store_true", help="include fused AdamW step")
    args = parser.parse_args()

    from hivemind_amd.models import AlbertConfig, AlbertForMaskedLM
    from hivemind_amd.ops import FusedAdamW, bind_grad

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    config = {"albert-base": AlbertConfig.base, "albert-large": AlbertConfig.large, "tiny": AlbertConfig.tiny}[args.model]()
    if device.type == "cpu":
        config.dtype = torch.float32
    if args.seq_len > config.max_position_embeddings:
        args.seq_len = config.max_position_embeddings
    model = AlbertForMaskedLM(config).to(device)

    opt = None
    masters = {}
    if args.adam:
        params = []
        for p in model.parameters():
            master = torch.nn.Parameter(p.detach().float().clone())
            masters[master] = p
            params.append(master)
        opt = FusedAdamW(params, lr=1e-4)
        for master, live in masters.items():
            opt.set_mirror(master, live.data)

    ids = torch.randint(0, config.vocab_size, (args.batch, args.seq_len), device=device)
    labels = ids.clone()

    def step():
        loss, _ = model(ids, labels=labels)
        if not args.no_backward:
            loss.backward()
            if opt is not None:
                for master, live in masters.items():
                    bind_grad(master, live.grad)
                opt.step()
            for p in model.parameters():
                p.grad = None
        return loss

    for _ in range(args.warmup):
        step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    samples_per_sec = args.steps * args.batch / elapsed
    print(json.dumps({
        "metric": "model-only samples/s",
        "value": round(samples_per_sec, 2),
        "ms_per_step": round(elapsed / args.steps * 1000, 2),
        "batch": args.batch,
        "seq_len": args.seq_len,
        "backward": not args.no_backward,
        "adam": args.adam,
        "model": args.model,
    }))


if __name__ == "__main__":
    main()
