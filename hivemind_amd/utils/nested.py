"""Flatten / repack nested python structures (dicts, lists, tuples).

Parity target: reference ``hivemind/utils/nested.py`` (nested_flatten,
nested_pack, nested_map, nested_compare) -- used by MoE schemas and the state
averager's optimizer state dump.
"""

from __future__ import annotations

from typing import Any, Iterator


def nested_flatten(t: Any) -> Iterator[Any]:
    """Yield leaves of a nested dict/list/tuple structure in deterministic order."""
    if isinstance(t, (list, tuple)):
        for x in t:
            yield from nested_flatten(x)
    elif isinstance(t, dict):
        for k in sorted(t):
            yield from nested_flatten(t[k])
    else:
        yield t


def nested_pack(flat: Any, structure: Any) -> Any:
    """Inverse of nested_flatten: fill `structure`'s leaves from iterator `flat`."""
    return _nested_pack(iter(flat), structure)


def _nested_pack(flat_iter: Iterator[Any], structure: Any) -> Any:
    if isinstance(structure, (list, tuple)):
        return type(structure)(_nested_pack(flat_iter, x) for x in structure)
    if isinstance(structure, dict):
        return {k: _nested_pack(flat_iter, structure[k]) for k in sorted(structure)}
    return next(flat_iter)


def is_namedtuple(x: Any) -> bool:
    return isinstance(x, tuple) and hasattr(x, "_fields")


def nested_compare(t: Any, u: Any) -> bool:
    """True if t and u have the same nested structure (leaves may differ)."""
    if isinstance(t, (list, tuple)) and isinstance(u, (list, tuple)):
        return len(t) == len(u) and all(map(nested_compare, t, u))
    if isinstance(t, dict) and isinstance(u, dict):
        return set(t.keys()) == set(u.keys()) and all(nested_compare(t[k], u[k]) for k in t)
    if isinstance(t, (list, tuple, dict)) or isinstance(u, (list, tuple, dict)):
        return False
    return True


def nested_map(fn, *t: Any) -> Any:
    """Apply fn to corresponding leaves of one or more identically-shaped structures."""
    assert t, "expected at least one structure"
    first = t[0]
    if isinstance(first, (list, tuple)):
        return type(first)(nested_map(fn, *parts) for parts in zip(*t))
    if isinstance(first, dict):
        return {k: nested_map(fn, *(d[k] for d in t)) for k in first}
    return fn(*t)
