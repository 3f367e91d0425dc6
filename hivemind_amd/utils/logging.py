"""Colored logging with env-var control.

Mirrors the reference's logging surface (hivemind/utils/logging.py): a
``get_logger`` factory, an env-controlled log level (``HIVEMIND_LOGLEVEL``),
optional ANSI colors (``HIVEMIND_COLORS``), and a ``use_hivemind_log_handler``
switch -- re-implemented from scratch for this framework.
"""

from __future__ import annotations

import logging
import os
import sys
import threading

_LOCK = threading.Lock()
_INITIALIZED = False
_HANDLER: logging.Handler | None = None

_COLORS = {
    logging.DEBUG: "\033[36m",
    logging.INFO: "\033[32m",
    logging.WARNING: "\033[33m",
    logging.ERROR: "\033[31m",
    logging.CRITICAL: "\033[1;31m",
}
_RESET = "\033[0m"


def _use_colors() -> bool:
    env = os.environ.get("HIVEMIND_COLORS", "").lower()
    if env in ("0", "false", "no"):
        return False
    if env in ("1", "true", "yes"):
        return True
    return sys.stderr.isatty()


class _Formatter(logging.Formatter):
    def __init__(self, colors: bool):
        super().__init__(
            fmt="%(asctime)s.%(msecs)03d [%(levelname)s] [%(name)s.%(funcName)s:%(lineno)d] %(message)s",
            datefmt="%b %d %H:%M:%S",
        )
        self.colors = colors

    def format(self, record: logging.LogRecord) -> str:
        if self.colors:
            color = _COLORS.get(record.levelno, "")
            record.levelname = f"{color}{record.levelname}{_RESET}"
        return super().format(record)


def _init_once() -> None:
    global _INITIALIZED, _HANDLER
    with _LOCK:
        if _INITIALIZED:
            return
        level = os.environ.get("HIVEMIND_LOGLEVEL", "INFO").upper()
        root = logging.getLogger("hivemind_amd")
        root.setLevel(getattr(logging, level, logging.INFO))
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(_Formatter(_use_colors()))
        root.addHandler(handler)
        root.propagate = False
        _HANDLER = handler
        _INITIALIZED = True


def get_logger(name: str | None = None) -> logging.Logger:
    """Return a logger under the ``hivemind_amd`` namespace."""
    _init_once()
    if name is None:
        name = "hivemind_amd"
    elif not name.startswith("hivemind_amd"):
        name = f"hivemind_amd.{name}"
    return logging.getLogger(name)


def use_hivemind_log_handler(mode: str) -> None:
    """'in_root_logger' | 'in_hivemind' (default) | 'nowhere'."""
    _init_once()
    root_pkg = logging.getLogger("hivemind_amd")
    top = logging.getLogger()
    assert _HANDLER is not None
    if mode == "in_root_logger":
        if _HANDLER not in top.handlers:
            top.addHandler(_HANDLER)
        root_pkg.propagate = True
        if _HANDLER in root_pkg.handlers:
            root_pkg.removeHandler(_HANDLER)
    elif mode == "in_hivemind":
        if _HANDLER not in root_pkg.handlers:
            root_pkg.addHandler(_HANDLER)
        root_pkg.propagate = False
        if _HANDLER in top.handlers:
            top.removeHandler(_HANDLER)
    elif mode == "nowhere":
        for lg in (root_pkg, top):
            if _HANDLER in lg.handlers:
                lg.removeHandler(_HANDLER)
    else:
        raise ValueError(f"unknown log handler mode: {mode}")
