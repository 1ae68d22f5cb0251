"""Shard ownership with failure takeover.

Round 1 split the CR keyspace across N controller processes by a static
``crc32(name) % N`` (manager._owns) — with no failure story: a dead shard
silently orphaned 1/N of the fleet.

This coordinator gives each shard a coordination.k8s.io/v1 Lease
(``active-monitor-shard-<i>-of-<N>``) and lets live processes adopt expired
ones:

- every process holds its HOME shard's lease and renews it; losing the home
  lease is fatal (same split-brain rule as leader election),
- each cycle it scans the other shard leases; an expired or unheld lease is
  acquired and that shard's keys are ADOPTED (the manager re-lists and
  enqueues them, arms their timers via normal reconciles),
- a restarted home process reclaims its shard gracefully via the Lease's
  ``preferredHolder`` field (the KEP-4355 coordinated-leader-election
  handshake): the home process writes ``preferredHolder=<me>`` on the lease
  it wants back; the adopter observes it at its next renewal, releases, and
  drops the shard's local state; the home process acquires on its next scan.

The coordinator only decides OWNERSHIP; the manager reacts through two
callbacks (adopt/drop) and keeps filtering keys with crc32 % N against the
owned set.
"""
from __future__ import annotations

import asyncio
import logging
import time
import zlib
from typing import Awaitable, Callable, Optional, Set

from ..api.types import k8s_now, parse_k8s_time
from ..kube.client import KubeClient
from ..kube.errors import AlreadyExistsError, ApiError, ConflictError, NotFoundError

log = logging.getLogger("active_monitor_amd.shards")

LEASE_API_VERSION = "coordination.k8s.io/v1"


def shard_of(name: str, shard_count: int) -> int:
    """Stable, process-independent key→shard map (builtin hash() is salted)."""
    return zlib.crc32(name.encode()) % shard_count


class ShardCoordinator:
    def __init__(
        self,
        client: KubeClient,
        namespace: str,
        shard_index: int,
        shard_count: int,
        identity: str,
        lease_duration: float = 15.0,
        renew_interval: float = 5.0,
        on_adopt: Optional[Callable[[int], Awaitable[None]]] = None,
        on_drop: Optional[Callable[[int], Awaitable[None]]] = None,
        on_home_lost: Optional[Callable[[], None]] = None,
    ):
        self.client = client
        self.namespace = namespace
        self.home = shard_index
        self.shard_count = shard_count
        self.identity = identity
        self.lease_duration = lease_duration
        self.renew_interval = renew_interval
        self.on_adopt = on_adopt
        self.on_drop = on_drop
        self.on_home_lost = on_home_lost
        self.owned: Set[int] = set()
        self._task: Optional[asyncio.Task] = None

    def _lease_name(self, shard: int) -> str:
        return f"active-monitor-shard-{shard}-of-{self.shard_count}"

    def _lease_obj(self, shard: int) -> dict:
        return {
            "apiVersion": LEASE_API_VERSION,
            "kind": "Lease",
            "metadata": {"name": self._lease_name(shard), "namespace": self.namespace},
            "spec": {
                "holderIdentity": self.identity,
                "leaseDurationSeconds": int(self.lease_duration),
                "renewTime": k8s_now(),
            },
        }

    def _expired(self, spec: dict) -> bool:
        holder = spec.get("holderIdentity")
        if not holder:
            return True
        renew = parse_k8s_time(spec.get("renewTime"))
        # k8s timestamps have 1s resolution; pad the window so a
        # freshly-renewed lease never reads as expired
        return renew is None or (time.time() - renew.timestamp()) > (
            self.lease_duration + 1.0
        )

    # -- lease ops ----------------------------------------------------------

    async def _try_acquire(self, shard: int) -> bool:
        """Acquire the shard's lease if unheld/expired/ours/preferred-to-us."""
        try:
            lease = await self.client.get(
                LEASE_API_VERSION, "Lease", self.namespace, self._lease_name(shard)
            )
        except NotFoundError:
            try:
                await self.client.create(self._lease_obj(shard))
                return True
            except AlreadyExistsError:
                return False
        spec = lease.get("spec") or {}
        holder = spec.get("holderIdentity")
        preferred = spec.get("preferredHolder")
        if holder != self.identity:
            if not self._expired(spec):
                return False
            if preferred and preferred != self.identity:
                # a graceful release aimed at the preferred owner: give it a
                # full lease window before anyone else may adopt (its own
                # death is then indistinguishable from expiry and the lease
                # becomes free-for-all)
                renew = parse_k8s_time(spec.get("renewTime"))
                if renew is not None and (
                    time.time() - renew.timestamp()
                ) <= self.lease_duration + 1.0:
                    return False
        fresh = self._lease_obj(shard)["spec"]
        if preferred and preferred != self.identity:
            fresh["preferredHolder"] = preferred  # keep a pending reclaim
        lease["spec"] = fresh
        try:
            await self.client.update(lease)
            return True
        except (ConflictError, NotFoundError):
            return False

    async def _request_reclaim(self, shard: int) -> None:
        """Home process asks a live adopter for its shard back
        (Lease.spec.preferredHolder, the KEP-4355 handshake)."""
        try:
            lease = await self.client.get(
                LEASE_API_VERSION, "Lease", self.namespace, self._lease_name(shard)
            )
            spec = lease.get("spec") or {}
            if spec.get("preferredHolder") != self.identity:
                spec["preferredHolder"] = self.identity
                await self.client.update(lease)
        except (ApiError, asyncio.CancelledError):
            pass

    async def _renew_owned(self, shard: int) -> bool:
        """Renew an owned shard; honors a pending reclaim by releasing.
        Returns False when the shard was lost/released."""
        try:
            lease = await self.client.get(
                LEASE_API_VERSION, "Lease", self.namespace, self._lease_name(shard)
            )
        except NotFoundError:
            return await self._try_acquire(shard)
        spec = lease.get("spec") or {}
        if spec.get("holderIdentity") != self.identity:
            return False  # someone took it (we must have expired)
        preferred = spec.get("preferredHolder")
        if shard != self.home and preferred and preferred != self.identity:
            # the rightful owner wants it back: release gracefully (the
            # renewTime stamp opens the preferred owner's priority window)
            spec["holderIdentity"] = ""
            spec["renewTime"] = k8s_now()
            try:
                await self.client.update(lease)
            except (ConflictError, NotFoundError):
                pass
            return False
        spec["renewTime"] = k8s_now()
        if preferred == self.identity:
            spec.pop("preferredHolder", None)  # reclaim satisfied
        try:
            await self.client.update(lease)
            return True
        except (ConflictError, NotFoundError):
            return False

    # -- lifecycle ----------------------------------------------------------

    async def start(self) -> None:
        """Block until the home shard's lease is held, then start the scan
        loop. A live adopter is asked to hand the shard over (preferredHolder)
        and releases within one renew interval."""
        while not await self._try_acquire(self.home):
            await self._request_reclaim(self.home)
            await asyncio.sleep(self.renew_interval)
        self.owned.add(self.home)
        log.info("shard %d/%d: home lease acquired as %s",
                 self.home, self.shard_count, self.identity)
        self._task = asyncio.ensure_future(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
        # release everything we hold so successors need not wait out the TTL
        for shard in list(self.owned):
            try:
                lease = await self.client.get(
                    LEASE_API_VERSION, "Lease", self.namespace, self._lease_name(shard)
                )
                spec = lease.get("spec") or {}
                if spec.get("holderIdentity") == self.identity:
                    spec["holderIdentity"] = ""
                    spec.pop("renewTime", None)  # immediate expiry: no window
                    await self.client.update(lease)
            except (ApiError, Exception):
                pass
        self.owned.clear()

    def crash(self) -> None:
        """Test hook: die without releasing leases (a crashed process)."""
        if self._task is not None:
            self._task.cancel()

    async def _loop(self) -> None:
        while True:
            await asyncio.sleep(self.renew_interval)
            if not await self._scan_once():
                return  # home lease lost: the process is going down

    async def _scan_once(self) -> bool:
        """One renew/adopt pass; returns False when the home lease was lost
        (the caller stops scanning — a deposed process must not keep
        acquiring shards while it shuts down)."""
        for shard in range(self.shard_count):
            if shard in self.owned:
                ok = await self._renew_owned(shard)
                if ok:
                    continue
                self.owned.discard(shard)
                if shard == self.home:
                    log.error("shard %d: HOME lease lost", shard)
                    if self.on_home_lost is not None:
                        self.on_home_lost()
                    return False
                log.info("shard %d: released/lost (reclaimed by owner)", shard)
                if self.on_drop is not None:
                    await self.on_drop(shard)
            else:
                if await self._try_acquire(shard):
                    self.owned.add(shard)
                    log.warning("shard %d: ADOPTED by %s (previous holder dead)",
                                shard, self.identity)
                    if self.on_adopt is not None:
                        await self.on_adopt(shard)
        return True
