"""GroupInfo: the result of matchmaking (reference averaging/group_info.py)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

from ..p2p import PeerID


@dataclass(frozen=True)
class GroupInfo:
    """A group of peers assembled for one all-reduce round."""

    group_id: bytes  # unique identifier of this group, assigned by the leader
    peer_ids: Tuple[PeerID, ...]  # ordered peer ids (order fixed by the leader)
    gathered: Tuple[bytes, ...]  # user-provided metadata from each peer, same order

    @property
    def group_size(self) -> int:
        return len(self.peer_ids)

    def __contains__(self, peer_id: PeerID) -> bool:
        return peer_id in self.peer_ids
