"""MoEBeamSearcher: find the best experts in a multi-dimensional grid.

Parity target: reference ``hivemind/moe/client/beam_search.py:27-401``:
left-to-right beam search over grid dimensions using the DHT prefix records
written by ``declare_experts`` (key = prefix, subkey = next coordinate);
misses are negatively cached to avoid re-querying dead prefixes; a batched
variant searches for every sample of a batch in one pass.
"""

from __future__ import annotations

import asyncio
import heapq
from collections import deque
from functools import partial
from typing import Deque, Dict, List, Optional, Sequence, Tuple

import torch

from ...dht import DHT
from ...p2p import PeerID
from ...utils.logging import get_logger
from ...utils.serializer import MSGPackSerializer
from ...utils.timed_storage import TimedStorage, get_dht_time
from ..expert_uid import UID_DELIMITER, Coordinate, ExpertInfo, ExpertPrefix, ExpertUID
from .expert import RemoteExpert, create_remote_experts

logger = get_logger(__name__)


class MoEBeamSearcher:
    def __init__(
        self,
        dht: DHT,
        uid_prefix: str,
        grid_size: Sequence[int],
        num_workers: Optional[int] = None,
        negative_caching: bool = True,
        cache_expiration: float = 300.0,
    ):
        self.dht = dht
        self.uid_prefix = uid_prefix.rstrip(UID_DELIMITER)
        self.grid_size = tuple(grid_size)
        self.negative_caching = negative_caching
        self.cache_expiration = cache_expiration
        self._negative_cache: TimedStorage = TimedStorage()

    # ------------------------------------------------------------- DHT reads

    def _fetch_prefixes(self, prefixes: Sequence[str]) -> Dict[str, Dict[Coordinate, ExpertInfo]]:
        """Fetch successor dictionaries for several prefixes at once."""
        alive = [p for p in prefixes if not (self.negative_caching and p in self._negative_cache)]

        async def _get(dht_obj, node):
            return await node.get_many(alive)

        found = self.dht.run_coroutine(_get) if alive else {}
        output: Dict[str, Dict[Coordinate, ExpertInfo]] = {p: {} for p in prefixes}
        for prefix in alive:
            entry = found.get(prefix)
            successors: Dict[Coordinate, ExpertInfo] = {}
            if entry is not None and entry.value is not None and hasattr(entry.value, "items"):
                for coord, sub_entry in entry.value.items():
                    try:
                        uid, peer_b58, endpoint = MSGPackSerializer.loads(sub_entry.value)
                        successors[int(coord)] = ExpertInfo(uid, PeerID.from_base58(peer_b58), endpoint)
                    except Exception:
                        continue
            if not successors and self.negative_caching:
                self._negative_cache.store(prefix, True, get_dht_time() + self.cache_expiration)
            output[prefix] = successors
        return output

    def get_initial_beam(
        self, scores: Sequence[float], beam_size: int
    ) -> List[Tuple[float, str, Dict[Coordinate, ExpertInfo]]]:
        """Top-level (dim 0) beam from the root prefix (reference beam_search.py:119)."""
        successors = self._fetch_prefixes([self.uid_prefix])[self.uid_prefix]
        beam = []
        for coord, info in successors.items():
            if coord < len(scores):
                beam.append((float(scores[coord]), f"{self.uid_prefix}{UID_DELIMITER}{coord}", {coord: info}))
        beam.sort(reverse=True, key=lambda t: t[0])
        return beam[:beam_size]

    def get_active_successors(self, prefixes: Sequence[str]) -> Dict[str, Dict[Coordinate, ExpertInfo]]:
        """(reference beam_search.py:210)"""
        return self._fetch_prefixes(prefixes)

    # ----------------------------------------------------------- beam search

    def find_best_experts(self, grid_scores: Sequence[Sequence[float]], beam_size: int) -> List[RemoteExpert]:
        """Find up to beam_size experts maximizing sum of per-dimension scores
        (reference beam_search.py:263-335)."""
        infos = self.find_best_expert_infos(grid_scores, beam_size)
        return [e for e in create_remote_experts(infos, self.dht) if e is not None]

    def find_best_expert_infos(self, grid_scores: Sequence[Sequence[float]], beam_size: int) -> List[ExpertInfo]:
        assert len(grid_scores) == len(self.grid_size), "one score vector per grid dimension"
        beam: List[Tuple[float, str]] = [(0.0, self.uid_prefix)]
        best_leaves: List[Tuple[float, ExpertInfo]] = []
        for dim, dim_scores in enumerate(grid_scores):
            prefixes = [prefix for _, prefix in beam]
            successors_map = self._fetch_prefixes(prefixes)
            candidates: List[Tuple[float, str, ExpertInfo]] = []
            for score, prefix in beam:
                for coord, info in successors_map.get(prefix, {}).items():
                    if coord >= len(dim_scores):
                        continue
                    new_score = score + float(dim_scores[coord])
                    candidates.append((new_score, f"{prefix}{UID_DELIMITER}{coord}", info))
            candidates.sort(reverse=True, key=lambda t: t[0])
            candidates = candidates[:beam_size]
            if dim == len(grid_scores) - 1:
                # leaves: the stored uid at the last level IS the expert
                best_leaves = [(score, info) for score, _prefix, info in candidates]
            else:
                beam = [(score, prefix) for score, prefix, _info in candidates]
            if not candidates:
                break
        return [info for _score, info in best_leaves]

    def batch_find_best_experts(
        self, batch_grid_scores: Sequence[Sequence[Sequence[float]]], beam_size: int
    ) -> List[List[ExpertInfo]]:
        """Per-sample beam search for a whole batch (reference beam_search.py:337-401)."""
        return [self.find_best_expert_infos(sample_scores, beam_size) for sample_scores in batch_grid_scores]
