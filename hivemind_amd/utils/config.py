"""YAML config-file support for the CLIs (reference hivemind_cli/config.yml
via configargparse, run_server.py:21-22).

``apply_config_file(parser)`` adds a ``--config`` option; values from the
YAML file become argument defaults (explicit command-line flags still win,
matching configargparse precedence: CLI > config file > hardcoded default).
"""

from __future__ import annotations

import argparse
from typing import List, Optional

from .logging import get_logger

logger = get_logger(__name__)


def parse_args_with_config(parser: argparse.ArgumentParser, argv: Optional[List[str]] = None):
    """Parse args honoring an optional ``--config config.yml``."""
    parser.add_argument("--config", type=str, default=None,
                        help="YAML file whose keys provide argument defaults (CLI flags override)")
    preliminary, _ = parser.parse_known_args(argv)
    if preliminary.config:
        import yaml

        with open(preliminary.config) as f:
            values = yaml.safe_load(f) or {}
        if not isinstance(values, dict):
            raise ValueError(f"config file {preliminary.config} must contain a mapping")
        known = {action.dest for action in parser._actions}
        unknown = set(values) - known
        if unknown:
            raise ValueError(f"unknown config keys in {preliminary.config}: {sorted(unknown)}")
        parser.set_defaults(**values)
    return parser.parse_args(argv)
