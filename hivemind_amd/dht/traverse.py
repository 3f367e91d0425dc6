"""Concurrent multi-query beam search over the DHT graph.

Parity target: reference ``hivemind/dht/traverse.py:13-258`` (``traverse_dht``
with a shared worker pool balancing exploration across queries and packing
multiple queries per RPC, and the single-query ``simple_traverse_dht``).
Re-implemented around the same contract:

``get_neighbors(peer, queries) -> {query: ([(neighbor_id, ...)], should_stop)}``
"""

from __future__ import annotations

import asyncio
import heapq
from collections import defaultdict
from typing import Any, Awaitable, Callable, Collection, Dict, List, Optional, Set, Tuple

from .routing import DHTID

ROOT = 0


async def simple_traverse_dht(
    query_id: DHTID,
    initial_nodes: Collection[DHTID],
    beam_size: int,
    get_neighbors: Callable[[DHTID], Awaitable[Tuple[Collection[DHTID], bool]]],
    visited_nodes: Collection[DHTID] = (),
) -> Tuple[Tuple[DHTID, ...], Set[DHTID]]:
    """Single-query beam search (reference traverse.py:13-69)."""
    visited_nodes = set(visited_nodes)
    initial_nodes = [node_id for node_id in initial_nodes if node_id not in visited_nodes]
    if not initial_nodes:
        return (), visited_nodes

    unvisited_nodes = [(query_id.xor_distance(uid), uid) for uid in initial_nodes]
    heapq.heapify(unvisited_nodes)
    nearest_nodes = [(-query_id.xor_distance(uid), uid) for uid in initial_nodes[:beam_size]]
    heapq.heapify(nearest_nodes)
    while len(nearest_nodes) > beam_size:
        heapq.heappop(nearest_nodes)
    visited_nodes |= set(initial_nodes)
    upper_bound = -nearest_nodes[ROOT][0]
    was_interrupted = False

    while (not was_interrupted) and unvisited_nodes and unvisited_nodes[ROOT][0] <= upper_bound:
        _, node_id = heapq.heappop(unvisited_nodes)
        neighbors, was_interrupted = await get_neighbors(node_id)
        neighbors = [uid for uid in neighbors if uid not in visited_nodes]
        visited_nodes.update(neighbors)
        for uid in neighbors:
            distance = query_id.xor_distance(uid)
            if distance <= upper_bound or len(nearest_nodes) < beam_size:
                heapq.heappush(unvisited_nodes, (distance, uid))
                heapq.heappushpop(nearest_nodes, (-distance, uid))
                upper_bound = -nearest_nodes[ROOT][0]
    return tuple(uid for _, uid in sorted(nearest_nodes, key=lambda t: -t[0])), visited_nodes


async def traverse_dht(
    queries: Collection[DHTID],
    initial_nodes: List[DHTID],
    beam_size: int,
    num_workers: int,
    queries_per_call: int,
    get_neighbors: Callable[[DHTID, Collection[DHTID]], Awaitable[Dict[DHTID, Tuple[Collection[DHTID], bool]]]],
    found_callback: Optional[Callable[[DHTID, List[DHTID], Set[DHTID]], Awaitable[Any]]] = None,
    await_all_tasks: bool = True,
    visited_nodes: Optional[Dict[DHTID, Set[DHTID]]] = None,
) -> Tuple[Dict[DHTID, List[DHTID]], Dict[DHTID, Set[DHTID]]]:
    """Multi-query beam search with a shared worker pool (reference traverse.py:72-258).

    Returns ({query: [nearest nodes]}, {query: visited set}).
    """
    queries = list(dict.fromkeys(queries))
    if not queries:
        return {}, {}
    visited: Dict[DHTID, Set[DHTID]] = defaultdict(set)
    if visited_nodes:
        for q, nodes in visited_nodes.items():
            visited[q] |= set(nodes)

    # per-query search state
    candidates: Dict[DHTID, List[Tuple[int, DHTID]]] = {}
    nearest: Dict[DHTID, List[Tuple[int, DHTID]]] = {}  # max-heap via negation
    finished: Set[DHTID] = set()
    active_tasks: List[asyncio.Task] = []

    for q in queries:
        candidates[q] = [(q.xor_distance(uid), uid) for uid in initial_nodes if uid not in visited[q]]
        heapq.heapify(candidates[q])
        nearest[q] = [(-d, uid) for d, uid in sorted(candidates[q])[:beam_size]]
        heapq.heapify(nearest[q])
        for uid in initial_nodes:
            visited[q].add(uid)

    in_flight: Dict[DHTID, int] = {q: 0 for q in queries}  # RPCs that may still add candidates
    queried: Dict[DHTID, Set[DHTID]] = {q: set() for q in queries}  # peers this query was sent to
    # (visited = "known, don't re-add as a candidate"; queried = "an RPC already
    # asked this peer about this query" -- initial_nodes are visited but NOT
    # queried, so packing can still batch them)

    def upper_bound(q: DHTID) -> int:
        if len(nearest[q]) < beam_size:
            return DHTID.MAX + 1
        return -nearest[q][ROOT][0]

    def maybe_finish(q: DHTID):
        if q in finished:
            return
        if in_flight.get(q, 0) > 0:
            return  # an outstanding request may still bring candidates or the value
        if not candidates[q] or candidates[q][ROOT][0] > upper_bound(q):
            finished.add(q)
            if found_callback is not None:
                result = [uid for _, uid in sorted(nearest[q], key=lambda t: -t[0])]
                task = asyncio.create_task(found_callback(q, result, set(visited[q])))
                active_tasks.append(task)

    search_lock = asyncio.Lock()
    progress_event = asyncio.Event()

    async def worker():
        while True:
            async with search_lock:
                # pick the (query, candidate) pair with the smallest distance among active queries
                best: Optional[Tuple[int, DHTID, DHTID]] = None  # (distance, query, peer)
                idle_wait = False
                _heappop = heapq.heappop
                for q in queries:
                    if q in finished:
                        continue
                    cand_q = candidates[q]
                    ub = upper_bound(q)  # constant within this pass (nearest[q] unchanged)
                    queried_q = queried[q]
                    while cand_q and (cand_q[ROOT][0] > ub or cand_q[ROOT][1] in queried_q):
                        _heappop(cand_q)
                    maybe_finish(q)
                    if q in finished:
                        continue
                    if not cand_q:
                        idle_wait = True  # in-flight elsewhere may refill this query
                        continue
                    d, uid = cand_q[ROOT]
                    if best is None or d < best[0]:
                        best = (d, q, uid)
                if best is None:
                    if not idle_wait:
                        return
                    progress_event.clear()
                else:
                    _, main_query, peer = best
                    heapq.heappop(candidates[main_query])
                    # pack additional active queries for which this peer is
                    # RELEVANT -- within the query's current search radius --
                    # nearest-first (reference traverse.py heuristic_priority).
                    # Packing arbitrary unqueried queries bloated every
                    # response ~16x with far neighbors that get dropped on
                    # arrival (measured 125 -> ~8 ms/key on a batched
                    # 128-peer declare).
                    packed = [main_query]
                    if queries_per_call > 1:
                        relevant = []
                        for q in queries:
                            if q is main_query or q in finished or peer in queried[q]:
                                continue
                            d_pq = q.xor_distance(peer)
                            if d_pq <= upper_bound(q):
                                relevant.append((d_pq, q))
                        relevant.sort(key=lambda t: t[0])
                        packed.extend(q for _, q in relevant[: queries_per_call - 1])
                    for q in packed:
                        visited[q].add(peer)
                        queried[q].add(peer)
                        in_flight[q] += 1
                        # the peer stays in q's candidate heap; the pick loop
                        # skips entries already in queried[q] lazily
            if best is None:
                # no candidates right now but other workers are mid-RPC: wait for progress
                try:
                    await asyncio.wait_for(progress_event.wait(), timeout=1.0)
                except asyncio.TimeoutError:
                    pass
                if all(q in finished for q in queries):
                    return
                continue
            try:
                responses = await get_neighbors(peer, packed)
            except Exception:
                responses = {}
            async with search_lock:
                for q in packed:
                    in_flight[q] -= 1
                for q, (neighbors, should_stop) in responses.items():
                    if q not in candidates:
                        continue
                    if should_stop:
                        finished.add(q)
                        if found_callback is not None:
                            result = [uid for _, uid in sorted(nearest[q], key=lambda t: -t[0])]
                            task = asyncio.create_task(found_callback(q, result, set(visited[q])))
                            active_tasks.append(task)
                        continue
                    for uid in neighbors:
                        if uid in visited[q]:
                            continue
                        visited[q].add(uid)
                        distance = q.xor_distance(uid)
                        if distance <= upper_bound(q) or len(nearest[q]) < beam_size:
                            heapq.heappush(candidates[q], (distance, uid))
                            if len(nearest[q]) < beam_size:
                                heapq.heappush(nearest[q], (-distance, uid))
                            else:
                                heapq.heappushpop(nearest[q], (-distance, uid))
                for q in packed:
                    maybe_finish(q)
                progress_event.set()

    workers = [asyncio.create_task(worker()) for _ in range(max(1, num_workers))]
    try:
        await asyncio.gather(*workers)
        for q in queries:
            maybe_finish(q)
        if await_all_tasks and active_tasks:
            await asyncio.gather(*active_tasks, return_exceptions=True)
    finally:
        for w in workers:
            if not w.done():
                w.cancel()

    results = {q: [uid for _, uid in sorted(nearest[q], key=lambda t: -t[0])] for q in queries}
    return results, {q: set(v) for q, v in visited.items()}
