"""Expert UID grammar: ``prefix.i.j.k`` (reference hivemind/moe/expert_uid.py)."""

from __future__ import annotations

import re
from typing import NamedTuple, Optional, Tuple, Union

from ..p2p import PeerID

ExpertUID = str
ExpertPrefix = str
Coordinate = int

UID_DELIMITER = "."
FLAT_EXPERT = -1
UID_PATTERN = re.compile(r"^(([^.])+)([.](?:[0]|([1-9]([0-9]*))))+$")
PREFIX_PATTERN = re.compile(r"^(([^.])+)([.](?:[0]|([1-9]([0-9]*))))*[.]$")


class ExpertInfo(NamedTuple):
    uid: ExpertUID
    peer_id: PeerID
    endpoint: str = ""


def is_valid_uid(maybe_uid: str) -> bool:
    return bool(UID_PATTERN.fullmatch(maybe_uid))


def is_valid_prefix(maybe_prefix: str) -> bool:
    return bool(PREFIX_PATTERN.fullmatch(maybe_prefix))


def split_uid(uid_or_prefix: Union[ExpertUID, ExpertPrefix]) -> Tuple[ExpertPrefix, Coordinate]:
    """'foo.1.2' -> ('foo.1.', 2) (reference expert_uid.py:33)."""
    uid_or_prefix = uid_or_prefix.rstrip(UID_DELIMITER)
    pivot = uid_or_prefix.rindex(UID_DELIMITER) + 1
    return uid_or_prefix[:pivot], int(uid_or_prefix[pivot:])
