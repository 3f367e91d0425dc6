"""Address helpers for the TCP control plane.

Parity target: reference ``hivemind/utils/networking.py`` (choose_ip_address,
free-port probing). This framework uses plain ``host:port`` endpoint strings
instead of multiaddrs -- an intentional simplification: the data plane is RCCL
over xGMI, and the control plane only needs TCP reachability.
"""

from __future__ import annotations

import socket
from typing import Optional, Sequence

Endpoint = str  # "host:port"
LOCALHOST = "127.0.0.1"


def get_free_port(host: str = LOCALHOST) -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as sock:
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind((host, 0))
        return sock.getsockname()[1]


def split_endpoint(endpoint: Endpoint) -> tuple[str, int]:
    host, _, port = endpoint.rpartition(":")
    return host, int(port)


def make_endpoint(host: str, port: int) -> Endpoint:
    return f"{host}:{port}"


def choose_ip_address(endpoints: Sequence[Endpoint], prefer_global: bool = True) -> Optional[str]:
    """Pick the best address to announce from a list of endpoints."""

    def _is_global(ip: str) -> bool:
        return not (ip.startswith("127.") or ip.startswith("10.") or ip.startswith("192.168.") or ip == "0.0.0.0")

    hosts = [split_endpoint(ep)[0] for ep in endpoints]
    if prefer_global:
        for h in hosts:
            if _is_global(h):
                return h
    return hosts[0] if hosts else None


def increase_file_limit(new_soft: int = 2**15, new_hard: int = 2**15):
    """Raise RLIMIT_NOFILE (reference utils/limits.py:18)."""
    import resource

    soft, hard = resource.getrlimit(resource.RLIMIT_NOFILE)
    new_soft = max(soft, new_soft)
    new_hard = max(hard, min(new_hard, resource.getrlimit(resource.RLIMIT_NOFILE)[1]))
    try:
        resource.setrlimit(resource.RLIMIT_NOFILE, (new_soft, new_hard))
    except (ValueError, OSError):
        pass
