"""Workflow watch hub: event-driven completion discovery.

The reference discovers workflow completion purely by polling the apiserver on
an inverse-exponential cadence (healthcheck_controller.go:613-624), so the
completion-detection latency is O(poll interval) — up to Timeout/2 seconds.
This hub subscribes once to Workflow watch events and wakes any interested
watcher the moment its workflow changes, collapsing that latency to
milliseconds. The IEB poll remains as the fallback cadence and still governs
the synthesized-failure deadline, so CR-visible semantics are unchanged — the
hub is purely a wake accelerator ("make the watch non-blocking
(watch/informer on Workflow phase or short resumable requeues)").
"""
from __future__ import annotations

import asyncio
import time
from collections import deque
from typing import Dict, List, Optional, Tuple

from ..kube.client import KubeClient
from ..kube.registry import WF_API_VERSION, WF_KIND

Key = Tuple[str, str]  # (namespace, name)


class WorkflowWatchHub:
    def __init__(self, client: KubeClient, namespace: Optional[str] = None):
        self.client = client
        self.namespace = namespace
        self._waiters: Dict[Key, List[asyncio.Future]] = {}
        # per-key monotonic change counter: lets a watcher detect events that
        # fired between its poll and its wait registration (no lost wakeups)
        self._seq: Dict[Key, int] = {}
        # keys whose object was DELETED, kept (seq bumped, not popped) so a
        # late wait_change(since=N) returns immediately and discovers the
        # NotFound at once instead of sleeping a full IEB interval; pruned
        # lazily after a grace period — workflow names are generateName-unique
        # so these entries would otherwise accumulate forever
        self._deleted: "deque[Tuple[float, Key]]" = deque()
        # last event per key: the watch payload already carries the full
        # Workflow object, so completion polls can read status from here
        # instead of a GET round-trip (controller-runtime informer-cache
        # shape). Entries are read-only shared snapshots.
        self._last: Dict[Key, Tuple[str, dict]] = {}
        self._sub = None
        self._task: Optional[asyncio.Task] = None

    def seq(self, namespace: str, name: str) -> int:
        return self._seq.get((namespace, name), 0)

    def cached(self, namespace: str, name: str) -> Optional[Tuple[str, dict]]:
        """Last (event_type, object) seen for the workflow, or None if no
        event reached the hub yet. The object is a shared read-only snapshot;
        an entry with event_type DELETED means the object is gone."""
        return self._last.get((namespace, name))

    def forget(self, namespace: str, name: str) -> None:
        """Drop a workflow's cache/seq state. Called by the watch loop when
        it finishes with a workflow: names are generateName-unique, so
        nothing will ever wait on the key again — without this, one cache
        entry per completed workflow accrues until the server-side TTL
        delete (which can be 30 minutes away), a real leak at fleet rates."""
        key = (namespace, name)
        if key not in self._waiters:
            self._last.pop(key, None)
            self._seq.pop(key, None)

    async def start(self) -> None:
        self._sub = self.client.watch(WF_API_VERSION, WF_KIND, self.namespace)
        self._task = asyncio.ensure_future(self._consume())

    async def stop(self) -> None:
        if self._sub is not None:
            self._sub.close()
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
        for waiters in self._waiters.values():
            for fut in waiters:
                if not fut.done():
                    fut.set_result(None)
        self._waiters.clear()

    async def _consume(self) -> None:
        async for ev in self._sub:
            meta = ev["object"].get("metadata") or {}
            key = (meta.get("namespace", ""), meta.get("name", ""))
            waiters = self._waiters.pop(key, ())
            self._seq[key] = self._seq.get(key, 0) + 1
            if ev["type"] == "DELETED":
                # tombstone only — holding the full object here costs ~6 KB ×
                # (deletion rate × prune TTL) of resident memory at fleet
                # rates, and cached() readers only look at the event type
                self._last[key] = ("DELETED", None)
                self._deleted.append((time.monotonic(), key))
            else:
                self._last[key] = (ev["type"], ev["object"])
            for fut in waiters:  # wake everyone watching it
                if not fut.done():
                    fut.set_result(ev["type"])
            self._prune_deleted()

    _DELETED_TTL = 60.0

    def _prune_deleted(self) -> None:
        cutoff = time.monotonic() - self._DELETED_TTL
        while self._deleted and self._deleted[0][0] < cutoff:
            _, key = self._deleted.popleft()
            if key not in self._waiters:
                self._seq.pop(key, None)
                self._last.pop(key, None)

    async def wait_change(
        self, namespace: str, name: str, timeout: float, since: Optional[int] = None
    ) -> Optional[str]:
        """Block until the workflow changes (returns the event type) or the
        timeout elapses (returns None). Callers re-poll the object either way,
        so a missed event only costs one poll interval. Pass ``since`` (a value
        from :meth:`seq` captured before the caller's poll) to return
        immediately when a change already landed in between."""
        key = (namespace, name)
        if since is not None and self._seq.get(key, 0) > since:
            return "CHANGED"
        fut = asyncio.get_running_loop().create_future()
        self._waiters.setdefault(key, []).append(fut)
        try:
            return await asyncio.wait_for(fut, timeout)
        except asyncio.TimeoutError:
            return None
        finally:
            waiters = self._waiters.get(key)
            if waiters is not None:
                try:
                    waiters.remove(fut)
                except ValueError:
                    pass
                if not waiters:
                    self._waiters.pop(key, None)
