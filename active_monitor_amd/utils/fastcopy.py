"""Fast deep copy for JSON-shaped data.

Kubernetes objects are plain dict/list/str/int/float/bool/None trees;
``copy.deepcopy`` pays for generality (memo dict, reduce protocol, type
dispatch) the store never needs. This hand-specialized copier is ~4-6x faster
and dominates the apiserver hot path (every get/list/create/update snapshot).
"""
from __future__ import annotations

from typing import Any

_SCALARS = (str, int, float, bool, bytes, type(None))


def deep_copy(obj: Any) -> Any:
    if isinstance(obj, _SCALARS):
        return obj
    if isinstance(obj, dict):
        return {k: deep_copy(v) for k, v in obj.items()}
    if isinstance(obj, list):
        return [deep_copy(v) for v in obj]
    if isinstance(obj, tuple):
        return tuple(deep_copy(v) for v in obj)
    import copy

    return copy.deepcopy(obj)  # non-JSON payloads: fall back to the slow path


def snapshot(obj: Any) -> Any:
    """Read-optimized copy of a Kubernetes object: a fresh top level with
    ``metadata`` and ``status`` deep-copied (the only subtrees writers touch
    in place) and every other subtree — typically the large, effectively
    immutable ``spec`` — shared by reference.

    Contract: consumers treat anything below the top level of a read result
    as read-only, except ``metadata``/``status`` which are private copies.
    Every in-tree consumer honors this; it is what makes 1000-CR fleets cheap.
    """
    if not isinstance(obj, dict):
        return deep_copy(obj)
    out = dict(obj)
    if "metadata" in out:
        out["metadata"] = deep_copy(out["metadata"])
    if "status" in out:
        out["status"] = deep_copy(out["status"])
    return out
