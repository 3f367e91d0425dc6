#!/usr/bin/env python3
"""Generate a TunableOp algorithm table for the flagship bench shapes.

Runs ALBERT fwd+bwd+optimizer at the given per-GPU batch with hipBLASLt
algorithm search enabled and writes the resulting CSV (merged with any
existing table) for bench.py to ship under profiles/.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", type=str, default="albert-base", choices=["albert-base", "llama-8b", "llama-1b"])
    ap.add_argument("--batch", type=int, default=410)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--out", type=str, default="gpurun_out/tunableop_albert.csv")
    ap.add_argument("--base", type=str, default="profiles/tunableop_albert_b128.csv")
    args = ap.parse_args()

    import torch.cuda.tunable as tunable

    tunable.set_filename(args.out)  # flushed at interpreter exit
    tunable.enable(True)
    if os.path.exists(args.base):
        try:
            tunable.read_file(args.base)
        except Exception as e:
            print(f"base table not loaded: {e}", file=sys.stderr)
    tunable.tuning_enable(True)

    from hivemind_amd.models import AlbertConfig, AlbertForMaskedLM, LlamaConfig, LlamaForCausalLM
    from hivemind_amd.ops import FusedAdamW

    torch.manual_seed(0)
    if args.model == "albert-base":
        config, cls = AlbertConfig.base(), AlbertForMaskedLM
    elif args.model == "llama-8b":
        config, cls = LlamaConfig.llama_3_8b(), LlamaForCausalLM
    else:
        config, cls = LlamaConfig.llama_1b(), LlamaForCausalLM
    model = cls(config).to("cuda")
    opt = FusedAdamW([{"params": [p for p in model.parameters() if p.dtype == torch.float32
                                  or p.dtype == torch.bfloat16]}], lr=1e-4)
    for i in range(args.steps):
        ids = torch.randint(0, config.vocab_size, (args.batch, args.seq_len), device="cuda")
        labels = ids.clone()
        if args.model == "albert-base":
            labels[torch.rand(labels.shape, device="cuda") > 0.15] = -100
        loss, _ = model(ids, labels=labels)
        loss.backward()
        opt.step()
        opt.zero_grad()
        torch.cuda.synchronize()
        print(f"step {i} loss={loss.item():.3f}", file=sys.stderr, flush=True)
    print(f"tuning done; table flushes to {args.out} at exit")


if __name__ == "__main__":
    main()
