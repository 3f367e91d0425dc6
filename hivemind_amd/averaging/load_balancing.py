"""Optimal per-peer vector fractions from peer bandwidths.

Parity target: reference ``hivemind/averaging/load_balancing.py:13-105``:
minimax linear program over butterfly transfer times followed by
Hagenbach-Bischoff integer apportionment.

Model (same as reference): peer i sends ``(1 - w_i)`` of its vector out as a
client and receives ``(group_size - 1) * w_i`` as an aggregator, so its
transfer time is ``(1 + (group_size - 2) * w_i) / bandwidth_i``; minimize the
maximum over peers. A vanishing bandwidth-proportional tie-break is added on
top because the program is degenerate at group_size == 2 (transfer time is
independent of w there, and we still want faster peers to own more).
"""

from __future__ import annotations

from typing import Optional, Sequence, Tuple

import numpy as np
import scipy.optimize

from ..utils.logging import get_logger

logger = get_logger(__name__)

LOAD_BALANCING_LP_DECIMALS = 9


def load_balance_peers(vector_size: int, bandwidths: Sequence[Optional[float]], min_size: int = 0) -> Tuple[int, ...]:
    """Split `vector_size` elements between peers to minimize the slowest transfer.

    ``bandwidths[i] == 0`` means client mode (no fraction); ``None`` means
    unspecified (resolved as the mean of the specified ones).
    """
    specified = [item for item in bandwidths if item is not None and item > 0]
    if specified:
        default_bandwidth = float(np.mean(specified))
        resolved = [item if item is not None else default_bandwidth for item in bandwidths]
        scores = optimize_parts_lp(vector_size, np.asarray(resolved, dtype=np.float64), min_size)
    else:
        assert not all(item == 0 for item in bandwidths), "need at least one non-client peer"
        scores = np.asarray([1.0 if item is None else 0.0 for item in bandwidths])
    return tuple(hagenbach_bishoff(vector_size, scores))


def optimize_parts_lp(vector_size: int, bandwidths: np.ndarray, min_size: int = 0) -> np.ndarray:
    assert np.all(bandwidths >= 0) and np.any(bandwidths > 0)
    bandwidths = np.asarray(bandwidths, dtype=np.float64)
    permutation = np.argsort(-bandwidths)
    bandwidths = bandwidths[permutation]
    is_nonzero = bandwidths != 0

    group_size = len(bandwidths)
    num_variables = group_size + 1  # [w_1, ..., w_n, xi]
    safe_bandwidths = np.maximum(bandwidths, 10**-LOAD_BALANCING_LP_DECIMALS)

    # objective: minimize xi, with a vanishing preference for giving weight to
    # high-bandwidth peers (breaks the group_size==2 degeneracy)
    c = np.zeros(num_variables)
    c[-1] = 1.0
    c[:group_size] = -1e-6 * (bandwidths / bandwidths.max())

    # (group_size - 2)/b_i * w_i - xi <= -1/b_i   for peers with bandwidth > 0
    A_time = np.hstack([np.diag((group_size - 2.0) / safe_bandwidths), -np.ones((group_size, 1))])[is_nonzero]
    b_time = (-1.0 / safe_bandwidths)[is_nonzero]

    A_eq = np.r_[np.ones(group_size), [0.0]][None, :]
    b_eq = np.array([1.0])
    bounds = [(0.0, 1.0 if nz else 0.0) for nz in is_nonzero] + [(0.0, None)]

    solution = scipy.optimize.linprog(c, A_ub=A_time, b_ub=b_time, A_eq=A_eq, b_eq=b_eq, bounds=bounds, method="highs")
    if solution.success:
        peer_scores = solution.x[:group_size]
        if vector_size and np.max(peer_scores) >= min_size / float(vector_size):
            peer_scores[peer_scores < min_size / float(vector_size)] = 0.0
        peer_scores = np.round(peer_scores, LOAD_BALANCING_LP_DECIMALS)
    else:
        logger.error(f"load balancing LP failed ({solution.message}); assigning equal weights")
        peer_scores = np.ones(group_size, dtype=np.float64)

    return peer_scores[np.argsort(permutation)]


def hagenbach_bishoff(vector_size: int, scores: Sequence[float]) -> Sequence[int]:
    """Largest-remainder integer apportionment (reference load_balancing.py:89-105)."""
    total_score = sum(scores)
    allocated = [int(vector_size * score_i / total_score) for score_i in scores]
    while sum(allocated) < vector_size:
        quotients = [score / (allocated[idx] + 1) for idx, score in enumerate(scores)]
        idx_max = quotients.index(max(quotients))
        allocated[idx_max] += 1
    return allocated
