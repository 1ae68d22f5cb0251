"""In-memory Kubernetes apiserver — the framework's envtest equivalent.

The reference's integration tier runs against envtest (a real kube-apiserver +
etcd, suite_test.go:67-134). This module provides the same role as a
dependency-free in-process store with the apiserver semantics the controller
exercises:

- optimistic concurrency via ``metadata.resourceVersion`` (conflict errors on
  stale writes, so the reconciler's retry-on-conflict paths are real),
- ``metadata.generateName`` server-side name generation,
- the HealthCheck status subresource (plain updates cannot touch status and
  vice versa),
- ownerReference cascade GC (Workflows vanish when their HealthCheck is
  deleted — the behavior relied on at healthcheck_controller.go:512-522),
- finalizer-driven deletionTimestamp handling,
- label-selector list filtering and watch streams (ADDED/MODIFIED/DELETED).

The core is synchronous and thread-safe; watches are asyncio queues fed via
``call_soon_threadsafe`` so any thread may mutate the store.
"""
from __future__ import annotations

import asyncio
import random
import threading
from collections import deque
from typing import Any, AsyncIterator, Dict, List, Optional, Tuple

from ..api.types import k8s_now
from ..utils.fastcopy import deep_copy, snapshot
from .errors import (
    AlreadyExistsError,
    ConflictError,
    ExpiredError,
    InvalidError,
    NotFoundError,
)
from .registry import DEFAULT_REGISTRY, STATUS_SUBRESOURCE_KINDS, Registry

Obj = Dict[str, Any]
Key = Tuple[str, str, str, str]  # (apiVersion, kind, namespace, name)

_SUFFIX_ALPHABET = "bcdfghjklmnpqrstvwxz2456789"  # k8s-style name suffix chars
_suffix_counter = [random.randrange(27 ** 4)]


def _rand_suffix(n: int = 5) -> str:
    """Unique k8s-style suffix: a randomly-seeded counter encoded in the
    apiserver's suffix alphabet — collision-free and far cheaper than
    per-character randomness on the create hot path."""
    _suffix_counter[0] += 1
    v = _suffix_counter[0]
    out = []
    for _ in range(n):
        v, r = divmod(v, len(_SUFFIX_ALPHABET))
        out.append(_SUFFIX_ALPHABET[r])
    return "".join(out)


def parse_label_selector(selector: Optional[str]) -> Dict[str, str]:
    """Parse an equality-based selector string ``k=v,k2=v2``."""
    if not selector:
        return {}
    out: Dict[str, str] = {}
    for part in selector.split(","):
        part = part.strip()
        if not part:
            continue
        if "=" not in part:
            raise InvalidError(f"unsupported selector term: {part!r}")
        k, _, v = part.partition("=")
        out[k.strip()] = v.lstrip("=").strip()  # tolerate '=='
    return out


def _labels_match(obj: Obj, selector: Dict[str, str]) -> bool:
    labels = (obj.get("metadata") or {}).get("labels") or {}
    return all(labels.get(k) == v for k, v in selector.items())


def parse_field_selector(selector: Optional[str]) -> List[Tuple[str, str, bool]]:
    """Parse ``a.b=v,c!=w`` into (dotted-path, value, negated) terms —
    the equality-based fieldSelector subset kubectl uses (e.g. `kubectl
    describe` filters Events by involvedObject.name/namespace)."""
    if not selector:
        return []
    out: List[Tuple[str, str, bool]] = []
    for part in selector.split(","):
        part = part.strip()
        if not part:
            continue
        if "!=" in part:
            k, _, v = part.partition("!=")
            out.append((k.strip(), v.strip(), True))
        elif "=" in part:
            k, _, v = part.partition("=")
            out.append((k.strip(), v.lstrip("=").strip(), False))
        else:
            raise InvalidError(f"unsupported field selector term: {part!r}")
    return out


def _fields_match(obj: Obj, terms: List[Tuple[str, str, bool]]) -> bool:
    for path, want, negated in terms:
        cur: Any = obj
        for seg in path.split("."):
            if not isinstance(cur, dict):
                cur = None
                break
            cur = cur.get(seg)
        have = "" if cur is None else str(cur)
        if (have == want) == negated:
            return False
    return True


class Subscription:
    """One watch stream. Async-iterate to receive ``{"type": ..., "object": ...}``."""

    def __init__(self, server: "MemoryApiServer", api_version: str, kind: str,
                 namespace: Optional[str], loop: asyncio.AbstractEventLoop):
        self._server = server
        self.api_version = api_version
        self.kind = kind
        self.namespace = namespace
        self._loop = loop
        self._queue: "asyncio.Queue[Optional[Dict[str, Any]]]" = asyncio.Queue()
        self._closed = False

    def _offer(self, event: Dict[str, Any]) -> None:
        if self._closed:
            return
        obj = event["object"]
        meta = obj.get("metadata") or {}
        if obj.get("apiVersion") != self.api_version or obj.get("kind") != self.kind:
            return
        if self.namespace is not None and meta.get("namespace", "") != self.namespace:
            return
        try:
            # events carry a shared read-only snapshot (copied once at
            # publish, not per subscriber) — consumers must not mutate it
            self._loop.call_soon_threadsafe(self._queue.put_nowait, event)
        except RuntimeError:
            self._closed = True  # loop gone

    def close(self) -> None:
        if not self._closed:
            self._closed = True
            self._server._unsubscribe(self)
            try:
                self._loop.call_soon_threadsafe(self._queue.put_nowait, None)
            except RuntimeError:
                pass

    def __aiter__(self) -> AsyncIterator[Dict[str, Any]]:
        return self

    async def __anext__(self) -> Dict[str, Any]:
        ev = await self._queue.get()
        if ev is None:
            raise StopAsyncIteration
        return ev


class MemoryApiServer:
    def __init__(self, registry: Registry = DEFAULT_REGISTRY):
        self.registry = registry
        self._objects: Dict[Key, Obj] = {}
        self._rv = 0
        self._lock = threading.RLock()
        self._subs: List[Subscription] = []
        # ownerReference uid → dependent keys (O(1) cascade GC)
        self._by_owner: Dict[str, set] = {}
        # Event objects are TTL'd by the real apiserver; cap them here so
        # long-running fleets don't grow the store unboundedly
        self._event_keys: deque = deque()
        self.max_events = 10000
        # counters for observability/benchmarks
        self.op_counts: Dict[str, int] = {"get": 0, "list": 0, "create": 0,
                                          "update": 0, "update_status": 0, "delete": 0}
        # bounded watch-event history: lets a resumed watch replay exactly the
        # events after its resourceVersion, and makes 410 Gone REAL — a resume
        # point older than the window raises ExpiredError, like etcd's
        # compacted-revision error surfaced by the apiserver. Tests shrink
        # history_window to force expiry deterministically.
        self._history: deque = deque()  # (rv:int, ev_type, obj snapshot)
        self.history_window = 4096
        # highest rv evicted from history — the compaction horizon; resuming
        # below it is 410 Gone
        self._compacted_rv = 0

    # -- internals ---------------------------------------------------------

    def _next_rv(self) -> str:
        self._rv += 1
        return str(self._rv)

    def _key(self, obj: Obj) -> Key:
        meta = obj.get("metadata") or {}
        info = self.registry.by_kind(obj.get("apiVersion", ""), obj.get("kind", ""))
        ns = meta.get("namespace", "") if info.namespaced else ""
        return (obj.get("apiVersion", ""), obj.get("kind", ""), ns, meta.get("name", ""))

    def _not_found(self, api_version: str, kind: str, name: str) -> NotFoundError:
        info = self.registry.by_kind(api_version, kind)
        group = api_version.split("/")[0] if "/" in api_version else ""
        full = f"{info.plural}.{group}" if group else info.plural
        return NotFoundError(f'{full} "{name}" not found')

    def _index_owners(self, key: Key, obj: Obj) -> None:
        for ref in (obj.get("metadata") or {}).get("ownerReferences") or []:
            uid = ref.get("uid")
            if uid:
                self._by_owner.setdefault(uid, set()).add(key)

    def _unindex_owners(self, key: Key, obj: Obj) -> None:
        for ref in (obj.get("metadata") or {}).get("ownerReferences") or []:
            uid = ref.get("uid")
            if uid:
                deps = self._by_owner.get(uid)
                if deps is not None:
                    deps.discard(key)
                    if not deps:
                        self._by_owner.pop(uid, None)

    def _publish(self, ev_type: str, obj: Obj) -> None:
        event = {"type": ev_type, "object": obj}
        try:
            rv = int((obj.get("metadata") or {}).get("resourceVersion", 0))
        except (TypeError, ValueError):
            rv = self._rv
        self._history.append((rv, ev_type, obj))
        while len(self._history) > self.history_window:
            self._compacted_rv = self._history.popleft()[0]
        for sub in list(self._subs):
            sub._offer(event)

    def events_since(self, api_version: str, kind: str, namespace: Optional[str],
                     resource_version: str) -> List[Dict[str, Any]]:
        """Replay retained events newer than ``resource_version`` for one
        (apiVersion, kind[, namespace]); raises :class:`ExpiredError` when the
        resume point predates the retained window (apiserver 410 semantics)."""
        try:
            since = int(resource_version)
        except (TypeError, ValueError):
            raise ExpiredError(f"resourceVersion {resource_version!r} is invalid")
        with self._lock:
            if since < self._compacted_rv:
                # events in (since, compacted_rv] are gone — the client can't
                # reconstruct the stream and must re-list
                raise ExpiredError(
                    f"too old resource version: {since} ({self._compacted_rv})"
                )
            out = []
            for rv, ev_type, obj in self._history:
                if rv <= since:
                    continue
                if obj.get("apiVersion") != api_version or obj.get("kind") != kind:
                    continue
                if namespace is not None and (obj.get("metadata") or {}).get(
                    "namespace", ""
                ) != namespace:
                    continue
                out.append({"type": ev_type, "object": obj})
            return out

    def _unsubscribe(self, sub: Subscription) -> None:
        with self._lock:
            if sub in self._subs:
                self._subs.remove(sub)

    # -- public API --------------------------------------------------------

    def create(self, obj: Obj, transfer: bool = False) -> Obj:
        """``transfer=True`` lets the store take ownership of ``obj`` (no
        copy-in) and return a read-optimized snapshot (no copy-out). Only for
        callers that built the dict themselves and never touch it again —
        the controller's submit/event/RBAC paths."""
        if not transfer:
            obj = deep_copy(obj)
        meta = obj.setdefault("metadata", {})
        with self._lock:
            self.op_counts["create"] += 1
            if not meta.get("name"):
                gen = meta.get("generateName")
                if not gen:
                    raise InvalidError("name or generateName is required")
                # retry suffixes on collision, like the apiserver
                for _ in range(16):
                    candidate = gen + _rand_suffix()
                    meta["name"] = candidate
                    if self._key(obj) not in self._objects:
                        break
                else:
                    raise AlreadyExistsError(f"could not generate unique name for {gen}")
            key = self._key(obj)
            if key in self._objects:
                raise AlreadyExistsError(
                    f'{obj.get("kind", "object")} "{meta["name"]}" already exists'
                )
            meta["uid"] = meta.get("uid") or ("uid-" + _rand_suffix(12))
            meta["resourceVersion"] = self._next_rv()
            meta["creationTimestamp"] = meta.get("creationTimestamp") or k8s_now()
            meta["generation"] = 1
            self._objects[key] = obj
            self._index_owners(key, obj)
            if obj.get("kind") == "Event":
                self._event_keys.append(key)
                while len(self._event_keys) > self.max_events:
                    old = self._event_keys.popleft()
                    dropped = self._objects.pop(old, None)
                    if dropped is not None:
                        self._unindex_owners(old, dropped)
            out = snapshot(obj)
            self._publish("ADDED", out)
            if not transfer:
                out = deep_copy(obj)
        return out

    def get(self, api_version: str, kind: str, namespace: str, name: str,
            snapshot_read: bool = False) -> Obj:
        """``snapshot_read=True`` returns a read-optimized copy whose subtrees
        other than metadata/status are shared with the store — callers must
        treat those as read-only (the controller hot path opts in; the default
        is a fully private deep copy, safe to mutate)."""
        info = self.registry.by_kind(api_version, kind)
        ns = namespace if info.namespaced else ""
        with self._lock:
            self.op_counts["get"] += 1
            obj = self._objects.get((api_version, kind, ns, name))
            if obj is None:
                raise self._not_found(api_version, kind, name)
            return snapshot(obj) if snapshot_read else deep_copy(obj)

    def list(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: Optional[str] = None,
        snapshot_read: bool = False,
        field_selector: Optional[str] = None,
    ) -> List[Obj]:
        selector = parse_label_selector(label_selector)
        fields = parse_field_selector(field_selector)
        copier = snapshot if snapshot_read else deep_copy
        with self._lock:
            self.op_counts["list"] += 1
            out = []
            for (av, k, ns, _), obj in self._objects.items():
                if av != api_version or k != kind:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                if selector and not _labels_match(obj, selector):
                    continue
                if fields and not _fields_match(obj, fields):
                    continue
                out.append(copier(obj))
            return out

    def update(self, obj: Obj) -> Obj:
        obj = deep_copy(obj)
        key = self._key(obj)
        meta = obj.setdefault("metadata", {})
        with self._lock:
            self.op_counts["update"] += 1
            existing = self._objects.get(key)
            if existing is None:
                raise self._not_found(key[0], key[1], key[3])
            ex_meta = existing["metadata"]
            rv = meta.get("resourceVersion")
            if rv and str(rv) != str(ex_meta.get("resourceVersion")):
                raise ConflictError(
                    f'Operation cannot be fulfilled on {obj.get("kind")} '
                    f'"{meta.get("name")}": the object has been modified; please apply '
                    f"your changes to the latest version and try again"
                )
            # immutable metadata
            meta["uid"] = ex_meta.get("uid")
            meta["creationTimestamp"] = ex_meta.get("creationTimestamp")
            if ex_meta.get("deletionTimestamp"):
                meta["deletionTimestamp"] = ex_meta["deletionTimestamp"]
            # status subresource: plain update cannot change status
            if (key[0], key[1]) in STATUS_SUBRESOURCE_KINDS:
                if "status" in existing:
                    obj["status"] = deep_copy(existing["status"])
                else:
                    obj.pop("status", None)
            # no-op updates don't bump the resourceVersion or emit watch
            # events (apiserver semantics — prevents self-triggering loops)
            meta["resourceVersion"] = ex_meta.get("resourceVersion")
            meta["generation"] = ex_meta.get("generation", 1)
            if obj == existing:
                return deep_copy(existing)
            if obj.get("spec") != existing.get("spec"):
                meta["generation"] = int(ex_meta.get("generation", 1)) + 1
            else:
                meta["generation"] = ex_meta.get("generation", 1)
            meta["resourceVersion"] = self._next_rv()
            self._unindex_owners(key, existing)
            self._objects[key] = obj
            self._index_owners(key, obj)
            # finalizer removal completes a pending delete
            if meta.get("deletionTimestamp") and not meta.get("finalizers"):
                del self._objects[key]
                self._unindex_owners(key, obj)
                self._publish("DELETED", snapshot(obj))
                self._cascade_delete(meta.get("uid"))
                return deep_copy(obj)
            self._publish("MODIFIED", snapshot(obj))
            out = deep_copy(obj)
        return out

    def update_status(self, obj: Obj) -> Obj:
        obj = deep_copy(obj)
        key = self._key(obj)
        meta = obj.get("metadata") or {}
        with self._lock:
            self.op_counts["update_status"] += 1
            existing = self._objects.get(key)
            if existing is None:
                raise self._not_found(key[0], key[1], key[3])
            ex_meta = existing["metadata"]
            rv = meta.get("resourceVersion")
            if rv and str(rv) != str(ex_meta.get("resourceVersion")):
                raise ConflictError(
                    f'Operation cannot be fulfilled on {obj.get("kind")} '
                    f'"{meta.get("name")}": the object has been modified; please apply '
                    f"your changes to the latest version and try again"
                )
            updated = snapshot(existing)
            if "status" in obj:
                updated["status"] = deep_copy(obj["status"])
            else:
                updated.pop("status", None)
            if updated == existing:  # no-op status write (apiserver semantics)
                return deep_copy(existing)
            updated["metadata"]["resourceVersion"] = self._next_rv()
            self._objects[key] = updated
            self._publish("MODIFIED", snapshot(updated))
            out = deep_copy(updated)
        return out

    @staticmethod
    def _merge(base: Obj, patch: Obj) -> None:
        """RFC 7386 JSON merge patch: dicts merge recursively, null deletes,
        everything else replaces (strategic-merge degenerates to this for
        our types — the CRD's list fields are x-kubernetes-list-type:
        atomic)."""
        for k, v in patch.items():
            if v is None:
                base.pop(k, None)
            elif isinstance(v, dict) and isinstance(base.get(k), dict):
                MemoryApiServer._merge(base[k], v)
            else:
                base[k] = deep_copy(v) if isinstance(v, (dict, list)) else v

    def patch(self, api_version: str, kind: str, namespace: str, name: str,
              patch_obj: Obj, subresource: str = "", upsert: bool = False) -> Obj:
        """Merge-patch an object (kubectl patch / the merge half of kubectl
        apply). ``subresource='status'`` touches only status; a plain patch
        cannot touch status for kinds with the subresource (enforced by the
        update path it reuses). ``upsert=True`` creates the object when
        absent (server-side-apply shape)."""
        info = self.registry.by_kind(api_version, kind)
        ns = namespace if info.namespaced else ""
        with self._lock:
            existing = self._objects.get((api_version, kind, ns, name))
            if existing is None:
                if not upsert:
                    raise self._not_found(api_version, kind, name)
                obj = deep_copy(patch_obj)
                obj.setdefault("apiVersion", api_version)
                obj.setdefault("kind", kind)
                meta = obj.setdefault("metadata", {})
                meta.setdefault("name", name)
                if info.namespaced:
                    meta.setdefault("namespace", namespace)
                return self.create(obj, transfer=True)
            merged = deep_copy(existing)
            if subresource == "status":
                status = merged.setdefault("status", {})
                patch_status = patch_obj.get("status")
                if isinstance(patch_status, dict):
                    self._merge(status, patch_status)
                return self.update_status(merged)
            self._merge(merged, deep_copy(patch_obj))
            # identity is immutable under patch
            merged["apiVersion"], merged["kind"] = api_version, kind
            m = merged.setdefault("metadata", {})
            m["name"], m["uid"] = name, existing["metadata"].get("uid")
            if info.namespaced:
                m["namespace"] = ns
            # honor an explicit rv in the patch (optimistic concurrency);
            # otherwise patch applies to the current version
            if "resourceVersion" not in (patch_obj.get("metadata") or {}):
                m["resourceVersion"] = existing["metadata"].get("resourceVersion")
            return self.update(merged)

    def delete(self, api_version: str, kind: str, namespace: str, name: str) -> None:
        info = self.registry.by_kind(api_version, kind)
        ns = namespace if info.namespaced else ""
        with self._lock:
            self.op_counts["delete"] += 1
            key = (api_version, kind, ns, name)
            obj = self._objects.get(key)
            if obj is None:
                raise self._not_found(api_version, kind, name)
            meta = obj["metadata"]
            if meta.get("finalizers"):
                if not meta.get("deletionTimestamp"):
                    meta["deletionTimestamp"] = k8s_now()
                    meta["resourceVersion"] = self._next_rv()
                    self._publish("MODIFIED", snapshot(obj))
                return
            del self._objects[key]
            self._unindex_owners(key, obj)
            # deletion bumps the rv (apiserver semantics) so a watch resuming
            # from just before the delete replays the DELETED event
            meta["resourceVersion"] = self._next_rv()
            self._publish("DELETED", snapshot(obj))
            self._cascade_delete(meta.get("uid"))

    def _cascade_delete(self, owner_uid: Optional[str]) -> None:
        """Background-propagation GC: delete dependents whose ownerReferences
        name the deleted uid (the mechanism behind Workflow cleanup on
        HealthCheck delete, healthcheck_controller.go:512-522). O(dependents)
        via the owner-uid index, not a store scan."""
        if not owner_uid:
            return
        dependents = list(self._by_owner.pop(owner_uid, ()))
        for key in dependents:
            obj = self._objects.pop(key, None)
            if obj is not None:
                self._unindex_owners(key, obj)
                obj["metadata"]["resourceVersion"] = self._next_rv()
                self._publish("DELETED", snapshot(obj))
                self._cascade_delete(obj["metadata"].get("uid"))

    def resource_version(self) -> str:
        """The store's current global resourceVersion (list metadata rv)."""
        with self._lock:
            return str(self._rv)

    def watch(self, api_version: str, kind: str, namespace: Optional[str] = None) -> Subscription:
        loop = asyncio.get_running_loop()
        sub = Subscription(self, api_version, kind, namespace, loop)
        with self._lock:
            self._subs.append(sub)
        return sub

    def __len__(self) -> int:
        with self._lock:
            return len(self._objects)

    def __bool__(self) -> bool:
        return True  # an empty store is still a store (see MemoryClient note)
