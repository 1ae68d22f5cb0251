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
from typing import Any, Dict, List, Optional, Protocol

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
        snapshot_read: bool = False,
    ) -> List[Obj]:
        await self._lat()
        return self.server.list(api_version, kind, namespace, label_selector, snapshot_read)

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
    are best-effort and never slow the reconcile hot path."""

    NORMAL = "Normal"
    WARNING = "Warning"

    def __init__(self, client: KubeClient, component: str = "active-monitor",
                 buffer: int = 4096):
        self.client = client
        self.component = component
        self._queue: Optional["asyncio.Queue[Obj]"] = None
        self._task: Optional[asyncio.Task] = None
        self._buffer = buffer
        self.dropped = 0

    def _ensure_pump(self) -> None:
        if self._task is None or self._task.done():
            self._queue = asyncio.Queue(maxsize=self._buffer)
            self._task = asyncio.get_running_loop().create_task(self._pump())

    async def _pump(self) -> None:
        while True:
            ev = await self._queue.get()
            try:
                create = self.client.create
                try:
                    await create(ev, transfer=True)
                except TypeError:  # backend without transfer support
                    await create(ev)
            except asyncio.CancelledError:
                raise
            except Exception:  # best-effort
                pass
            finally:
                self._queue.task_done()

    async def event(self, involved: Obj, ev_type: str, reason: str, message: str) -> None:
        meta = involved.get("metadata") or {}
        ns = meta.get("namespace", "") or "default"
        ev = {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {"generateName": (meta.get("name", "object") + "."), "namespace": ns},
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
            "firstTimestamp": None,
        }
        self._ensure_pump()
        try:
            self._queue.put_nowait(ev)
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
