"""Controller manager — the controller-runtime equivalent.

Wires what the reference's ``cmd/main.go`` + controller-runtime provide:

- an informer: initial list + watch of HealthCheck CRs feeding the workqueue
  (the apiserver watch streams controller-runtime maintains),
- ``max_workers`` reconcile workers (``MaxConcurrentReconciles``,
  healthcheck_controller.go:298, flag ``max-workers`` default 10,
  cmd/main.go:144) — THE 1/2/4/8 scaling knob of BASELINE.md,
- error policy: reconcile errors are re-queued with per-item exponential
  backoff; ``requeue_after`` honors the reference's 1s error requeue (:204),
- /healthz and /readyz probes plus the Prometheus /metrics endpoint
  (cmd/main.go:74-85,121-126),
- optional Lease-based leader election (cmd/main.go:87-88).

Per-reconcile latencies are recorded (monotonic clock) for the benchmark
harness — this is the instrumentation BASELINE.md's p50 metric reads.
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import List, Optional, Tuple

from .. import API_VERSION
from ..kube.client import EventRecorder, KubeClient
from .leader import LeaderElector
from .reconciler import HC_KIND, HealthCheckReconciler
from .workqueue import WorkQueue

log = logging.getLogger("active_monitor_amd.manager")


class Manager:
    def __init__(
        self,
        client: KubeClient,
        max_workers: int = 10,
        recorder: Optional[EventRecorder] = None,
        namespace: Optional[str] = None,
        metrics_addr: Optional[Tuple[str, int]] = None,
        health_addr: Optional[Tuple[str, int]] = None,
        leader_elect: bool = False,
        leader_identity: str = "",
        metrics_security=None,
        record_latencies: bool = True,
        shard_index: int = 0,
        shard_count: int = 1,
        enable_wf_hub: bool = True,
        shard_ha: bool = False,
        shard_lease_duration: float = 15.0,
        shard_renew_interval: float = 5.0,
        resync_period: float = 10 * 3600.0,
    ):
        self.client = client
        self.max_workers = max_workers
        self.namespace = namespace
        self.queue = WorkQueue()
        self.recorder = recorder or EventRecorder(client)
        self.reconciler = HealthCheckReconciler(
            client, self.recorder, max_parallel=max_workers, queue=self.queue
        )
        self.metrics_addr = metrics_addr
        self.health_addr = health_addr
        # optional endpoints.MetricsSecurity: TLS + bearer authn on /metrics
        # (reference default is the secure path, cmd/main.go:74-85,139)
        self.metrics_security = metrics_security
        self.leader_elect = leader_elect
        self.leader_identity = leader_identity
        self._tasks: List[asyncio.Task] = []
        self._sub = None
        self._started = asyncio.Event()
        self._stopped = False
        self._servers: List[object] = []
        # set when the manager must die (e.g. lost leadership lease); the
        # entrypoint awaits it alongside the stop signal — in the reference
        # losing the lease is fatal for the replica (cmd/main.go:87-88)
        self.fatal = asyncio.Event()
        self.fatal_reason: Optional[str] = None
        self.record_latencies = record_latencies
        from collections import deque

        # bounded: an unbounded list leaks ~8B/reconcile on week-long fleets
        self.latencies: "deque" = deque(maxlen=200_000)
        # horizontal scale-out: shard the CR keyspace by stable name hash so
        # N cooperating controller processes each own a disjoint subset (one
        # event loop saturates one core — see docs/DESIGN.md scaling model)
        if not (0 <= shard_index < shard_count):
            raise ValueError("shard_index must be in [0, shard_count)")
        self.shard_index = shard_index
        self.shard_count = shard_count
        # shard failure takeover (engine/shards.py): lease-per-shard with
        # adoption of dead shards and preferredHolder reclaim. Off by default
        # (static sharding, the round-1 behavior) — enable with --shard-ha.
        self.shard_ha = shard_ha and shard_count > 1
        self.shard_lease_duration = shard_lease_duration
        self.shard_renew_interval = shard_renew_interval
        self.coordinator = None
        # event-driven completion wakeups; False falls back to the reference's
        # pure inverse-exponential polling (same semantics, higher latency)
        self.enable_wf_hub = enable_wf_hub
        self.wf_hub = None
        # informer cache: the watch stream already delivers every HealthCheck
        # object; serving reconciler reads from it (controller-runtime cached
        # client, healthcheck_controller.go:133) saves ~2.5 apiserver GETs per
        # cycle on the wire. Events update the cache BEFORE enqueueing, so a
        # dequeued key's cache entry is at least as new as the event that
        # queued it. Only this shard's keys are cached.
        self.hc_cache: dict = {}
        self._cache_synced = False
        self.enable_hc_cache = True
        # informer resync (controller-runtime SyncPeriod, default 10h): a
        # periodic full re-list that re-enqueues every owned key, healing any
        # state a lost watch event could have left stale; 0 disables
        self.resync_period = resync_period

    # -- lifecycle ----------------------------------------------------------

    async def start(self) -> None:
        """Start informer + workers (+ endpoints); returns once running."""
        if self.leader_elect:
            elector = LeaderElector(
                self.client,
                # LeaderElectionID from the reference (cmd/main.go:88)
                name="689451f8.keikoproj.io",
                namespace=self.namespace or "default",
                identity=self.leader_identity or f"manager-{id(self):x}",
            )
            await elector.acquire()
            renew = asyncio.ensure_future(elector.renew_loop())
            renew.add_done_callback(self._renew_done)
            self._tasks.append(renew)
            self.elector = elector

        if self.shard_ha:
            from .shards import ShardCoordinator

            self.coordinator = ShardCoordinator(
                self.client,
                namespace=self.namespace or "default",
                shard_index=self.shard_index,
                shard_count=self.shard_count,
                identity=self.leader_identity or f"shard-{self.shard_index}-{id(self):x}",
                lease_duration=self.shard_lease_duration,
                renew_interval=self.shard_renew_interval,
                on_adopt=self._adopt_shard,
                on_drop=self._drop_shard,
                on_home_lost=self._home_shard_lost,
            )
            # blocks until the home shard's lease is held (a live previous
            # holder is asked to hand it over via preferredHolder)
            await self.coordinator.start()

        if self.health_addr is not None or self.metrics_addr is not None:
            from .endpoints import serve_endpoints

            self._servers = await serve_endpoints(
                self, health=self.health_addr, metrics=self.metrics_addr,
                metrics_security=self.metrics_security,
            )

        if self.enable_wf_hub:
            from .watchhub import WorkflowWatchHub

            self.wf_hub = WorkflowWatchHub(self.client, self.namespace)
            await self.wf_hub.start()
            self.reconciler.wf_hub = self.wf_hub

        self._tasks.append(asyncio.ensure_future(self._informer()))
        if self.resync_period > 0:
            self._tasks.append(asyncio.ensure_future(self._resync_loop()))
        for i in range(self.max_workers):
            self._tasks.append(asyncio.ensure_future(self._worker(i)))
        self._started.set()

    async def run_forever(self) -> None:
        await self.start()
        try:
            await asyncio.gather(*self._tasks)
        except asyncio.CancelledError:
            pass

    async def stop(self) -> None:
        self._stopped = True
        if self._sub is not None:
            self._sub.close()
        if getattr(self, "wf_hub", None) is not None:
            await self.wf_hub.stop()
        stop_rec = getattr(self.recorder, "stop", None)
        if stop_rec is not None:
            await stop_rec()
        await self.queue.shutdown()
        self.reconciler.stop_all()
        for t in self._tasks:
            t.cancel()
        await asyncio.gather(*self._tasks, return_exceptions=True)
        # release shard leases only after all reconcile work has stopped —
        # releasing first would let an adopter start driving keys this
        # process's timers/watches could still briefly touch
        if self.coordinator is not None:
            await self.coordinator.stop()
        for srv in self._servers:
            srv.close()

    # -- shard-HA reactions (engine/shards.py callbacks) --------------------

    async def _adopt_shard(self, shard: int) -> None:
        """An orphaned shard is ours now: surface its CRs (cache + queue) so
        normal reconciles re-arm their timers and resume their schedules —
        the restart-resume path, applied to a subset of the keyspace."""
        from .shards import shard_of

        try:
            objs = await self.client.list(API_VERSION, HC_KIND, self.namespace)
        except Exception as e:
            log.error("adopt shard %d: list failed: %s", shard, e)
            return
        n = 0
        for obj in objs:
            meta = obj.get("metadata") or {}
            name = meta.get("name", "")
            if shard_of(name, self.shard_count) != shard:
                continue
            key = (meta.get("namespace", ""), name)
            self.hc_cache[key] = obj
            await self.queue.add(key)
            n += 1
        log.warning("adopted shard %d: %d healthchecks enqueued", shard, n)

    async def _drop_shard(self, shard: int) -> None:
        """The rightful owner reclaimed the shard: stop its timers/watches and
        forget its CRs so exactly one process drives each key."""
        from .shards import shard_of

        dropped = 0
        for (ns, name) in list(self.hc_cache):
            if shard_of(name, self.shard_count) != shard:
                continue
            self.hc_cache.pop((ns, name), None)
            self.reconciler._stop_timer(name, ns)
            self.reconciler._cancel_watches((ns, name))
            dropped += 1
        log.info("dropped shard %d: released %d healthchecks", shard, dropped)

    def _home_shard_lost(self) -> None:
        self.fatal_reason = f"home shard {self.shard_index} lease lost"
        log.error("%s — shutting down", self.fatal_reason)
        self.fatal.set()

    def _renew_done(self, task: "asyncio.Task") -> None:
        """A finished renew loop means the lease is gone: a deposed replica
        must stop reconciling immediately or two actives double-submit
        workflows (split-brain). Surface it as a fatal manager condition."""
        if task.cancelled() or self._stopped:
            return
        exc = task.exception()
        self.fatal_reason = str(exc) if exc else "leader-election renew loop exited"
        log.error("leadership lost: %s — shutting down", self.fatal_reason)
        self.fatal.set()

    @property
    def ready(self) -> bool:
        return self._started.is_set() and not self._stopped and not self.fatal.is_set()

    # -- informer -----------------------------------------------------------

    def _owns(self, name: str) -> bool:
        if self.shard_count <= 1:
            return True
        from .shards import shard_of

        shard = shard_of(name, self.shard_count)
        if self.coordinator is not None:
            return shard in self.coordinator.owned
        return shard == self.shard_index

    def _cache_lookup(self, namespace: str, name: str):
        """Reconciler-facing cache read; see HealthCheckReconciler.hc_lookup."""
        from .reconciler import CACHE_MISS

        if not self._cache_synced:
            return CACHE_MISS
        return self.hc_cache.get((namespace, name))

    async def _informer(self) -> None:
        self._sub = self.client.watch(API_VERSION, HC_KIND, self.namespace)
        # initial list AFTER subscribing so no event can slip between the two
        for obj in await self.client.list(API_VERSION, HC_KIND, self.namespace):
            meta = obj.get("metadata") or {}
            name = meta.get("name", "")
            if self._owns(name):
                self.hc_cache[(meta.get("namespace", ""), name)] = obj
                await self.queue.add((meta.get("namespace", ""), name))
        if self.enable_hc_cache:
            self._cache_synced = True
            self.reconciler.hc_lookup = self._cache_lookup
        async for ev in self._sub:
            obj = ev["object"]
            meta = obj.get("metadata") or {}
            name = meta.get("name", "")
            if self._owns(name):
                key = (meta.get("namespace", ""), name)
                if ev["type"] == "DELETED":
                    self.hc_cache.pop(key, None)
                else:
                    self.hc_cache[key] = obj
                await self.queue.add(key)

    async def _resync_loop(self) -> None:
        """Periodic full re-list + re-enqueue of owned keys (the informer
        resync controller-runtime performs every SyncPeriod). Level-triggered
        reconciles make this a no-op when nothing diverged."""
        while True:
            await asyncio.sleep(self.resync_period)
            try:
                objs = await self.client.list(API_VERSION, HC_KIND, self.namespace)
            except asyncio.CancelledError:
                raise
            except Exception as e:
                log.warning("resync list failed: %s", e)
                continue
            live = set()
            for obj in objs:
                meta = obj.get("metadata") or {}
                name = meta.get("name", "")
                if not self._owns(name):
                    continue
                key = (meta.get("namespace", ""), name)
                live.add(key)
                self.hc_cache[key] = obj
                await self.queue.add(key)
            # heal cache entries whose DELETED event was lost. A key absent
            # from the list snapshot may simply have been created AFTER the
            # list (its watch event already consumed) — evicting it then
            # would orphan a live CR forever, so suspects are verified with
            # a direct read before reaping.
            from ..kube.errors import NotFoundError

            for key in [k for k in self.hc_cache if k not in live]:
                try:
                    obj = await self.client.get(
                        API_VERSION, HC_KIND, key[0], key[1], snapshot_read=True
                    )
                except NotFoundError:
                    self.hc_cache.pop(key, None)
                    await self.queue.add(key)  # reconcile observes the NotFound
                except Exception as e:
                    log.warning("resync verify %s failed: %s", key, e)
                else:
                    self.hc_cache[key] = obj  # alive: refresh instead

    # -- workers ------------------------------------------------------------

    async def _worker(self, idx: int) -> None:
        while True:
            item = await self.queue.get()
            if item is None:
                return
            key, flags = item
            ns, name = key
            t0 = time.monotonic()
            result = None
            try:
                result = await self.reconciler.reconcile(ns, name, flags)
            except asyncio.CancelledError:
                raise
            except Exception as e:  # reconciler already guards; belt & braces
                log.error("worker %d: reconcile %s/%s raised: %s", idx, ns, name, e)
            finally:
                await self.queue.done(key)
            if self.record_latencies:
                self.latencies.append(time.monotonic() - t0)
            if result is not None and result.error is not None:
                # controller-runtime: error ⇒ rate-limited requeue
                await self.queue.add_rate_limited(key)
            elif result is not None and result.requeue_after > 0:
                await self.queue.add_after(key, result.requeue_after)
            else:
                self.queue.forget(key)

    # -- bench/test helpers --------------------------------------------------

    def drain_latencies(self) -> List[float]:
        out = list(self.latencies)
        self.latencies.clear()
        return out
