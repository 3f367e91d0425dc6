#!/usr/bin/env python3
"""Swarm monitor (reference examples/albert/run_training_monitor.py): an
auxiliary CPU peer that aggregates the swarm's training progress from the DHT
and periodically downloads + saves the latest model state from peer donors.

    python examples/albert/run_training_monitor.py --initial_peers host:port
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch

from hivemind_amd import DHT
from hivemind_amd.utils.logging import get_logger
from hivemind_amd.utils.timed_storage import get_dht_time

logger = get_logger("albert_monitor")


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--initial_peers", nargs="+", required=True)
    parser.add_argument("--run_id", default="albert_collab")
    parser.add_argument("--refresh_period", type=float, default=10.0)
    parser.add_argument("--save_checkpoint_to", default=None)
    args = parser.parse_args()

    dht = DHT(initial_peers=args.initial_peers, start=True)
    logger.info(f"monitor connected; watching run '{args.run_id}'")
    progress_key = f"{args.run_id}_progress"

    state_client = None
    if args.save_checkpoint_to:
        # a non-sharing client averager that can download the swarm's training
        # state from the best donor (reference monitor checkpointing)
        import torch

        from hivemind_amd.averaging import DecentralizedAverager

        state_client = DecentralizedAverager(
            [torch.zeros(1)], dht, start=True, prefix=f"{args.run_id}_state_averager",
            client_mode=True, allow_state_sharing=False, target_group_size=2,
        )

    while True:
        record = dht.get(progress_key, latest=True)
        if record is not None and isinstance(record.value, dict):
            num_peers = len(record.value)
            total_sps = 0.0
            max_epoch = 0
            total_samples = 0
            for _, entry in record.value.items():
                if not isinstance(entry.value, dict):
                    continue
                total_sps += entry.value.get("samples_per_second", 0.0)
                max_epoch = max(max_epoch, entry.value.get("epoch", 0))
                total_samples += entry.value.get("samples_accumulated", 0)
            logger.info(
                f"swarm: {num_peers} peers, epoch {max_epoch}, "
                f"{total_sps:.1f} samples/s aggregate, {total_samples} samples this epoch"
            )
        else:
            logger.info("no training progress published yet")
        if state_client is not None:
            try:
                import torch

                state = state_client.load_state_from_peers(timeout=30)
                if state is not None:
                    metadata, tensors = state
                    torch.save({"metadata": metadata, "tensors": tensors}, args.save_checkpoint_to)
                    logger.info(f"saved swarm checkpoint ({len(tensors)} tensors) to {args.save_checkpoint_to}")
            except Exception as e:
                logger.warning(f"checkpoint download failed: {e!r}")
        time.sleep(args.refresh_period)


if __name__ == "__main__":
    main()
