"""Fast deep copy for JSON-shaped data.

Kubernetes objects are plain dict/list/str/int/float/bool/None trees;
``copy.deepcopy`` pays for generality (memo dict, reduce protocol, type
dispatch) the store never needs. This hand-specialized copier is ~4-6x faster
and dominates the apiserver hot path (every get/list/create/update snapshot).

When the native extension is built in-tree (``make native`` /
``python setup.py build_ext --inplace`` compiles ``native/_amcore.c``), its C
implementations replace both functions — another ~3x on the same profile. On
a GPU box the native path must load (no silent fallback): set
``AM_REQUIRE_NATIVE=1`` to make a missing extension fatal.
"""
from __future__ import annotations

import os
from typing import Any

_SCALARS = (str, int, float, bool, bytes, type(None))


def _py_deep_copy(obj: Any) -> Any:
    if isinstance(obj, _SCALARS):
        return obj
    if isinstance(obj, dict):
        return {k: _py_deep_copy(v) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_py_deep_copy(v) for v in obj]
    if isinstance(obj, tuple):
        return tuple(_py_deep_copy(v) for v in obj)
    import copy

    return copy.deepcopy(obj)  # non-JSON payloads: fall back to the slow path


def _py_snapshot(obj: Any) -> Any:
    """Read-optimized copy of a Kubernetes object: a fresh top level with
    ``metadata`` and ``status`` deep-copied (the only subtrees writers touch
    in place) and every other subtree — typically the large, effectively
    immutable ``spec`` — shared by reference.

    Contract: consumers treat anything below the top level of a read result
    as read-only, except ``metadata``/``status`` which are private copies.
    Every in-tree consumer honors this; it is what makes 1000-CR fleets cheap.
    """
    if not isinstance(obj, dict):
        return _py_deep_copy(obj)
    out = dict(obj)
    if "metadata" in out:
        out["metadata"] = _py_deep_copy(out["metadata"])
    if "status" in out:
        out["status"] = _py_deep_copy(out["status"])
    return out


try:
    from active_monitor_amd import _amcore  # built from native/_amcore.c

    deep_copy = _amcore.deep_copy
    snapshot = _amcore.snapshot
    NATIVE = True
except ImportError:  # pragma: no cover - toolchain-less installs
    if os.environ.get("AM_REQUIRE_NATIVE") == "1":
        raise ImportError(
            "active_monitor_amd._amcore is required (AM_REQUIRE_NATIVE=1) but "
            "not built; run `python setup.py build_ext --inplace`"
        )
    deep_copy = _py_deep_copy
    snapshot = _py_snapshot
    NATIVE = False
