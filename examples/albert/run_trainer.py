#!/usr/bin/env python3
"""Collaborative ALBERT pretraining recipe (reference examples/albert/run_trainer.py).

Each invocation is one peer. Peers find each other through the DHT and
jointly accumulate gradients toward ``--target_batch_size``; on one 8-GPU
MI355X node, launch one process per GPU:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/albert/run_trainer.py

or run standalone peers joining an existing swarm:

    python examples/albert/run_trainer.py --initial_peers host:port

There is no dataset download in this environment, so the recipe trains on
synthetic MLM batches; swap ``make_batch`` for a real tokenized corpus to
reproduce the reference's WikiText-103 run.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch

from hivemind_amd import DHT, Optimizer
from hivemind_amd.compression import Float16Compression
from hivemind_amd.models import AlbertConfig, AlbertForMaskedLM
from hivemind_amd.ops import FusedAdamW, hip_available
from hivemind_amd.utils.logging import get_logger

logger = get_logger("albert_trainer")


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--initial_peers", nargs="*", default=[])
    parser.add_argument("--run_id", default="albert_collab")
    parser.add_argument("--target_batch_size", type=int, default=4096)
    parser.add_argument("--batch_size", type=int, default=128)
    parser.add_argument("--seq_len", type=int, default=512)
    parser.add_argument("--lr", type=float, default=0.00176)
    parser.add_argument("--optimizer", choices=["lamb", "adamw"], default="lamb",
                        help="lamb = the reference recipe (LAMB + gradient clipping, "
                             "examples/albert/run_trainer.py:266); adamw = FusedAdamW")
    parser.add_argument("--clip_grad_norm", type=float, default=1.0)
    parser.add_argument("--max_epochs", type=int, default=10**9)
    parser.add_argument("--statistics_every", type=int, default=10)
    parser.add_argument("--backup_every_epochs", type=int, default=10)
    parser.add_argument("--state_path", default="albert_state.pt")
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if use_gpu else torch.device("cpu")

    if world_size > 1:
        import torch.distributed as dist

        dist.init_process_group("nccl" if use_gpu else "gloo", rank=rank, world_size=world_size)
        if rank == 0:
            dht = DHT(initial_peers=args.initial_peers, start=True)
            endpoint = [dht.endpoint]
        else:
            dht, endpoint = None, [None]
        dist.broadcast_object_list(endpoint, src=0)
        if rank != 0:
            dht = DHT(initial_peers=[endpoint[0]], start=True)
    else:
        dht = DHT(initial_peers=args.initial_peers, start=True)
    logger.info(f"DHT at {dht.endpoint}; share --initial_peers {dht.endpoint} with other peers")

    config = AlbertConfig.base()
    if not use_gpu:
        config.dtype = torch.float32
    model = AlbertForMaskedLM(config).to(device)

    # the reference collaborative-ALBERT recipe trains with LAMB + gradient
    # clipping (reference run_trainer.py:266 Lamb + ClippingWrapper)
    if args.optimizer == "lamb":
        from hivemind_amd.moe.server.layers.optim import ClippingWrapper
        from hivemind_amd.ops import Lamb

        opt_factory = ClippingWrapper.create(
            Lamb, lr=args.lr, weight_decay=0.01, clip_grad_norm=args.clip_grad_norm
        )
    else:
        opt_factory = lambda pg: FusedAdamW(pg, lr=args.lr, weight_decay=0.01)

    opt = Optimizer(
        dht=dht,
        run_id=args.run_id,
        target_batch_size=args.target_batch_size,
        batch_size_per_step=args.batch_size,
        optimizer=opt_factory,
        params=[{"params": list(model.parameters())}],
        offload_optimizer=True,
        delay_optimizer_step=True,
        delay_grad_averaging=True,
        grad_compression=Float16Compression(),
        matchmaking_time=3.0,
        averaging_timeout=120.0,
        verbose=rank == 0,
    )

    def make_batch():
        ids = torch.randint(0, config.vocab_size, (args.batch_size, args.seq_len), device=device)
        labels = ids.clone()
        labels[torch.rand(labels.shape, device=device) > 0.15] = -100
        return ids, labels

    samples_done, t0, last_epoch = 0, time.perf_counter(), -1
    while opt.local_epoch < args.max_epochs:
        ids, labels = make_batch()
        loss, _ = model(ids, labels=labels)
        if not torch.isfinite(loss):
            # reference run_trainer: restore the last backup on NaN/inf loss
            logger.warning(f"non-finite loss {loss.item()}; restoring from backup")
            opt.zero_grad()
            if os.path.exists(args.state_path):
                backup = torch.load(args.state_path, weights_only=False)
                model.load_state_dict(backup["model"])
                opt.load_state_dict(backup["optimizer"])
            continue
        loss.backward()
        opt.step()
        opt.zero_grad()
        samples_done += args.batch_size
        if opt.local_epoch != last_epoch:
            last_epoch = opt.local_epoch
            sps = samples_done / (time.perf_counter() - t0)
            logger.info(f"epoch {last_epoch}: loss={loss.item():.4f}, {sps:.1f} samples/s local")
            if args.backup_every_epochs and last_epoch % args.backup_every_epochs == 0 and rank == 0:
                torch.save({"model": model.state_dict(), "optimizer": opt.state_dict()}, args.state_path)
                logger.info(f"state backed up to {args.state_path}")


if __name__ == "__main__":
    main()
