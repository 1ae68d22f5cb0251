"""Async Kubernetes client facade.

The reconciler programs against this interface the way the reference programs
against controller-runtime's ``client.Client`` + the dynamic client + the typed
clientset (healthcheck_controller.go:133-137). Backends:

- :class:`MemoryClient` — wraps :class:`~active_monitor_amd.kube.memory.MemoryApiServer`
  (the envtest/bench backend), with optional simulated per-op latency so
  benchmarks can model apiserver RTT honestly.
- ``HttpClient`` (kube/http.py) — a real apiserver over HTTP(S).
"""
from __future__ import annotations

import asyncio
import time as _time
import zlib
from typing import Any, Dict, List, Optional, Protocol

from ..api.types import k8s_now
from .errors import AlreadyExistsError, NotFoundError
from .memory import MemoryApiServer, Subscription

Obj = Dict[str, Any]


class KubeClient(Protocol):
    async def get(self, api_version: str, kind: str, namespace: str, name: str) -> Obj: ...

    async def list(
        self, api_version: str, kind: str,
        namespace: Optional[str] = None, label_selector: Optional[str] = None,
    ) -> List[Obj]: ...

    async def create(self, obj: Obj) -> Obj: ...

    async def update(self, obj: Obj) -> Obj: ...

    async def update_status(self, obj: Obj) -> Obj: ...

    async def delete(self, api_version: str, kind: str, namespace: str, name: str) -> None: ...

    def watch(self, api_version: str, kind: str, namespace: Optional[str] = None) -> Subscription: ...


class MemoryClient:
    """KubeClient over the in-memory apiserver.

    ``latency`` (seconds) is awaited before every request to emulate a real
    apiserver round-trip; 0 disables it.
    """

    def __init__(self, server: Optional[MemoryApiServer] = None, latency: float = 0.0):
        # NB: not `server or ...` — an empty MemoryApiServer is falsy (__len__)
        self.server = server if server is not None else MemoryApiServer()
        self.latency = latency

    _yield_counter = 0

    async def _lat(self) -> None:
        if self.latency > 0:
            await asyncio.sleep(self.latency)
        else:
            # amortized cooperative yield: enough to keep tight loops from
            # starving the event loop without paying a call_soon per request
            MemoryClient._yield_counter += 1
            if MemoryClient._yield_counter % 16 == 0:
                await asyncio.sleep(0)

    async def get(self, api_version: str, kind: str, namespace: str, name: str,
                  snapshot_read: bool = False) -> Obj:
        await self._lat()
        return self.server.get(api_version, kind, namespace, name, snapshot_read)

    async def list(
        self, api_version: str, kind: str,
        namespace: Optional[str] = None, label_selector: Optional[str] = None,
        snapshot_read: bool = False, field_selector: Optional[str] = None,
    ) -> List[Obj]:
        await self._lat()
        return self.server.list(api_version, kind, namespace, label_selector,
                                snapshot_read, field_selector)

    async def create(self, obj: Obj, transfer: bool = False) -> Obj:
        await self._lat()
        return self.server.create(obj, transfer)

    async def update(self, obj: Obj) -> Obj:
        await self._lat()
        return self.server.update(obj)

    async def update_status(self, obj: Obj) -> Obj:
        await self._lat()
        return self.server.update_status(obj)

    async def delete(self, api_version: str, kind: str, namespace: str, name: str) -> None:
        await self._lat()
        self.server.delete(api_version, kind, namespace, name)

    def watch(self, api_version: str, kind: str, namespace: Optional[str] = None) -> Subscription:
        return self.server.watch(api_version, kind, namespace)


class EventRecorder:
    """Kubernetes Event emission (record.EventRecorder equivalent — the
    reference emits events on nearly every transition, e.g.
    healthcheck_controller.go:243,280,532,636,663).

    Like client-go's event broadcaster, emission is non-blocking: ``event()``
    enqueues and returns; a background task writes to the apiserver. Events
    are best-effort and never slow the reconcile hot path.

    Write amplification is bounded the way client-go's EventCorrelator bounds
    it: repeats of the same (object, type, reason,
    message) within ``AGG_TTL`` dedup into ONE Event object whose ``count``
    increments via update instead of a fresh create, and a per-key token
    bucket (burst ``SPAM_BURST``, refill 1/``SPAM_REFILL_SECS``) caps the
    write rate — further repeats only bump the local count, which the next
    allowed write carries to the server. A steady-state fleet therefore
    emits near-zero Event traffic instead of ~6-8 creates per cycle."""

    NORMAL = "Normal"
    WARNING = "Warning"

    AGG_TTL = 600.0  # client-go aggregation interval (10 min)
    SPAM_BURST = 25  # client-go EventSourceObjectSpamFilter defaults
    SPAM_REFILL_SECS = 300.0
    _AGG_MAX = 8192  # cache bound; prune lazily

    def __init__(self, client: KubeClient, component: str = "active-monitor",
                 buffer: int = 4096):
        self.client = client
        self.component = component
        self._queue: Optional["asyncio.Queue"] = None
        self._task: Optional[asyncio.Task] = None
        self._buffer = buffer
        self.dropped = 0
        #: dedup state per (ns, kind, name, type, reason, message)
        self._agg: Dict[tuple, Dict[str, Any]] = {}
        #: spam-filter token buckets per INVOLVED OBJECT (ns, kind, name) —
        #: client-go's EventSourceObjectSpamFilter granularity; surviving
        #: aggregation-window rollovers keeps long-running fleets suppressed
        self._spam: Dict[tuple, Dict[str, float]] = {}
        #: observability: wire writes suppressed by the spam filter
        self.suppressed = 0

    def _ensure_pump(self) -> None:
        if self._task is None or self._task.done():
            self._queue = asyncio.Queue(maxsize=self._buffer)
            self._task = asyncio.get_running_loop().create_task(self._pump())

    async def _write(self, op: str, key: tuple, ev: Obj) -> None:
        if op == "create":
            try:
                try:
                    await self.client.create(ev, transfer=True)
                except TypeError:  # backend without transfer support
                    await self.client.create(ev)
            except AlreadyExistsError:
                # survived a restart (deterministic name): fall through to
                # an update carrying the new count
                await self.client.update(ev)
        else:
            try:
                await self.client.update(ev)
            except NotFoundError:
                # server TTL'd the Event object out; recreate it
                ev["metadata"].pop("resourceVersion", None)
                await self.client.create(ev)

    async def _pump(self) -> None:
        while True:
            op, key, ev = await self._queue.get()
            try:
                await self._write(op, key, ev)
            except asyncio.CancelledError:
                raise
            except Exception:  # best-effort
                pass
            finally:
                self._queue.task_done()

    def _build_event(self, involved: Obj, ev_type: str, reason: str,
                     message: str, name: str, ns: str, count: int,
                     first: str, last: str) -> Obj:
        meta = involved.get("metadata") or {}
        return {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {"name": name, "namespace": ns},
            "involvedObject": {
                "apiVersion": involved.get("apiVersion"),
                "kind": involved.get("kind"),
                "name": meta.get("name"),
                "namespace": meta.get("namespace", ""),
                "uid": meta.get("uid", ""),
            },
            "reason": reason,
            "message": message,
            "type": ev_type,
            "source": {"component": self.component},
            "count": count,
            "firstTimestamp": first,
            "lastTimestamp": last,
        }

    def _prune_agg(self, now: float) -> None:
        if len(self._agg) <= self._AGG_MAX:
            return
        stale = [k for k, e in self._agg.items() if now - e["t0"] > self.AGG_TTL]
        for k in stale:
            del self._agg[k]
        if len(self._agg) > self._AGG_MAX:  # still hot: drop oldest half
            for k in sorted(self._agg, key=lambda k: self._agg[k]["t0"])[
                : len(self._agg) // 2
            ]:
                del self._agg[k]
        if len(self._spam) > self._AGG_MAX:
            stale = [k for k, b in self._spam.items()
                     if now - b["refill_at"] > self.SPAM_REFILL_SECS]
            for k in stale:
                del self._spam[k]

    def _spend_token(self, obj_key: tuple, now: float) -> bool:
        """Per-object write-rate token bucket (burst SPAM_BURST, refill
        1/SPAM_REFILL_SECS — client-go EventSourceObjectSpamFilter)."""
        bucket = self._spam.get(obj_key)
        if bucket is None:
            bucket = {"tokens": float(self.SPAM_BURST), "refill_at": now}
            self._spam[obj_key] = bucket
        elapsed = now - bucket["refill_at"]
        bucket["refill_at"] = now
        bucket["tokens"] = min(
            float(self.SPAM_BURST),
            bucket["tokens"] + elapsed / self.SPAM_REFILL_SECS,
        )
        if bucket["tokens"] < 1.0:
            return False
        bucket["tokens"] -= 1.0
        return True

    async def event(self, involved: Obj, ev_type: str, reason: str, message: str) -> None:
        meta = involved.get("metadata") or {}
        ns = meta.get("namespace", "") or "default"
        obj_name = meta.get("name", "object")
        key = (ns, involved.get("kind"), obj_name, ev_type, reason, message)
        now = _time.monotonic()
        stamp = k8s_now()

        obj_key = (ns, involved.get("kind"), obj_name)
        entry = self._agg.get(key)
        if entry is not None and now - entry["t0"] > self.AGG_TTL:
            entry = None  # aggregation window rolled over: start fresh
        self._ensure_pump()
        if entry is None:
            self._prune_agg(now)
            # deterministic per-key name so repeats update in place
            ev_name = f"{obj_name}.{zlib.crc32(repr(key).encode()):08x}"
            entry = {"t0": now, "count": 1, "first": stamp, "name": ev_name}
            self._agg[key] = entry
            if not self._spend_token(obj_key, now):
                self.suppressed += 1  # object over its write budget
                return
            item = ("create", key,
                    self._build_event(involved, ev_type, reason, message,
                                      ev_name, ns, 1, stamp, stamp))
        else:
            entry["count"] += 1
            if not self._spend_token(obj_key, now):
                self.suppressed += 1  # local count keeps accruing
                return
            item = ("update", key,
                    self._build_event(involved, ev_type, reason, message,
                                      entry["name"], ns, entry["count"],
                                      entry["first"], stamp))
        try:
            self._queue.put_nowait(item)
        except asyncio.QueueFull:
            self.dropped += 1  # drop rather than block (broadcaster behavior)

    async def flush(self, timeout: float = 5.0) -> None:
        if self._queue is not None:
            try:
                await asyncio.wait_for(self._queue.join(), timeout)
            except asyncio.TimeoutError:
                pass

    async def stop(self) -> None:
        await self.flush(1.0)
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass


class FakeRecorder(EventRecorder):
    """Test recorder capturing events in-memory
    (record.NewFakeRecorder equivalent, used across the reference's unit
    tests, healthcheck_controller_unit_test.go:40-46)."""

    def __init__(self, capacity: int = 100):
        self.events: List[str] = []
        self.capacity = capacity
        self._queue = None
        self._task = None

    async def event(self, involved: Obj, ev_type: str, reason: str, message: str) -> None:
        if len(self.events) < self.capacity:
            self.events.append(f"{ev_type} {reason} {message}")
