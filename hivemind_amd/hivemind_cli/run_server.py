#!/usr/bin/env python3
"""Host a MoE expert server (reference hivemind_cli/run_server.py).

    python -m hivemind_amd.hivemind_cli.run_server --num_experts 4 \
        --expert_pattern "ffn.[0:256]" --expert_cls ffn --hidden_dim 1024
"""

from __future__ import annotations

import argparse
from pathlib import Path

import torch

from ..moe.server import Server
from ..utils.logging import get_logger

logger = get_logger(__name__)


def main():
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--num_experts", type=int, default=None)
    parser.add_argument("--expert_pattern", type=str, default=None, help='e.g. "ffn.[0:256]"')
    parser.add_argument("--expert_uids", nargs="*", default=None)
    parser.add_argument("--expert_cls", type=str, default="ffn")
    parser.add_argument("--hidden_dim", type=int, default=1024)
    parser.add_argument("--max_batch_size", type=int, default=16384)
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--optimizer", type=str, default="adam", choices=["adam", "sgd", "none"])
    parser.add_argument("--scheduler", type=str, default=None, choices=[None, "linear"])
    parser.add_argument("--num_warmup_steps", type=int, default=None)
    parser.add_argument("--num_total_steps", type=int, default=None)
    parser.add_argument("--clip_grad_norm", type=float, default=None)
    parser.add_argument("--initial_peers", nargs="*", default=[])
    parser.add_argument("--checkpoint_dir", type=Path, default=None)
    parser.add_argument("--load_experts", action="store_true")
    parser.add_argument("--stats_report_interval", type=float, default=60.0)
    parser.add_argument("--relay_endpoint", type=str, default=None,
                        help="host:port of a public relay peer; serve through it when behind NAT (no inbound sockets)")
    from ..utils.config import parse_args_with_config

    args = parse_args_with_config(parser)

    optim_cls = {"adam": torch.optim.Adam, "sgd": torch.optim.SGD, "none": None}[args.optimizer]
    server = Server.create(
        initial_peers=args.initial_peers,
        expert_uids=args.expert_uids,
        expert_pattern=args.expert_pattern,
        num_experts=args.num_experts,
        expert_cls=args.expert_cls,
        hidden_dim=args.hidden_dim,
        max_batch_size=args.max_batch_size,
        device=args.device,
        optim_cls=optim_cls,
        scheduler=args.scheduler,
        num_warmup_steps=args.num_warmup_steps,
        num_total_steps=args.num_total_steps,
        clip_grad_norm=args.clip_grad_norm,
        checkpoint_dir=args.checkpoint_dir,
        load_experts_from_dir=args.load_experts,
        stats_report_interval=args.stats_report_interval,
        relay_endpoint=args.relay_endpoint,
        start=True,
    )
    logger.info(
        f"Server started with {len(server.module_backends)} experts "
        f"({list(server.module_backends.keys())[:4]}...); DHT at {server.dht.endpoint}"
    )
    try:
        server.join()
    except KeyboardInterrupt:
        logger.info("shutting down server")
        server.shutdown()


if __name__ == "__main__":
    main()
