#!/usr/bin/env python3
"""Standalone bootstrap/monitor DHT peer (reference hivemind_cli/run_dht.py).

    python -m hivemind_amd.hivemind_cli.run_dht [--host 0.0.0.0] [--port P]
        [--initial_peers host:port ...] [--refresh_period 30]
"""

from __future__ import annotations

import argparse
import time

from ..dht import DHT
from ..utils.logging import get_logger
from ..utils.networking import LOCALHOST
from ..utils.timed_storage import get_dht_time

logger = get_logger(__name__)


def report_status(dht: DHT) -> None:
    async def _status(dht_obj, node):
        return (
            len(node.protocol.routing_table.peer_id_to_uid),
            len(node.protocol.storage),
        )

    num_peers, num_keys = dht.run_coroutine(_status)
    logger.info(f"DHT status: {num_peers} known peers, {num_keys} local keys, peer_id={dht.peer_id}")


def main():
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--host", default=LOCALHOST)
    parser.add_argument("--port", type=int, default=0)
    parser.add_argument("--initial_peers", nargs="*", default=[], help="host:port of existing peers")
    parser.add_argument("--refresh_period", type=float, default=30.0)
    from ..utils.config import parse_args_with_config

    args = parse_args_with_config(parser)

    dht = DHT(
        initial_peers=args.initial_peers,
        start=True,
        listen_host=args.host,
        port=args.port,
    )
    logger.info(f"Running a DHT instance at {dht.endpoint} (peer id {dht.peer_id})")
    logger.info(f"To connect other peers to this one, use --initial_peers {dht.endpoint}")
    try:
        while True:
            # periodic self-heartbeat: a get of a random key exercises the swarm
            dht.get(f"heartbeat_{get_dht_time():.0f}")
            report_status(dht)
            time.sleep(args.refresh_period)
    except KeyboardInterrupt:
        logger.info("shutting down DHT")
        dht.shutdown()


if __name__ == "__main__":
    main()
