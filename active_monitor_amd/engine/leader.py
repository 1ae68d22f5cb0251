"""Lease-based leader election (cmd/main.go:87-88 equivalent).

A single active replica holds a coordination.k8s.io/v1 Lease named with the
reference's LeaderElectionID ``689451f8.keikoproj.io``; others block in
``acquire()`` until the holder's lease expires.
"""
from __future__ import annotations

import asyncio
import logging
import time
from ..api.types import k8s_now, parse_k8s_time
from ..kube.client import KubeClient
from ..kube.errors import AlreadyExistsError, ConflictError, NotFoundError

log = logging.getLogger("active_monitor_amd.leader")

LEASE_API_VERSION = "coordination.k8s.io/v1"


class LeaderElector:
    def __init__(
        self,
        client: KubeClient,
        name: str,
        namespace: str,
        identity: str,
        lease_duration: float = 15.0,
        renew_interval: float = 5.0,
        retry_interval: float = 2.0,
    ):
        import os

        # ops/test override for failover latency (mirrors controller-runtime's
        # LeaseDuration/RenewDeadline tunables the reference inherits)
        env_lease = os.environ.get("AM_LEADER_LEASE_SECS")
        env_renew = os.environ.get("AM_LEADER_RENEW_SECS")
        if env_lease:
            lease_duration = float(env_lease)
        if env_renew:
            renew_interval = float(env_renew)
            retry_interval = min(retry_interval, renew_interval)
        self.client = client
        self.name = name
        self.namespace = namespace
        self.identity = identity
        self.lease_duration = lease_duration
        self.renew_interval = renew_interval
        self.retry_interval = retry_interval
        self.is_leader = False

    def _lease_obj(self) -> dict:
        return {
            "apiVersion": LEASE_API_VERSION,
            "kind": "Lease",
            "metadata": {"name": self.name, "namespace": self.namespace},
            "spec": {
                "holderIdentity": self.identity,
                "leaseDurationSeconds": int(self.lease_duration),
                "renewTime": k8s_now(),
            },
        }

    async def try_acquire(self) -> bool:
        try:
            lease = await self.client.get(LEASE_API_VERSION, "Lease", self.namespace, self.name)
        except NotFoundError:
            try:
                await self.client.create(self._lease_obj())
                self.is_leader = True
                return True
            except AlreadyExistsError:
                return False
        spec = lease.get("spec") or {}
        holder = spec.get("holderIdentity")
        renew = parse_k8s_time(spec.get("renewTime"))
        expired = renew is None or (time.time() - renew.timestamp()) > self.lease_duration
        if holder == self.identity or expired or not holder:
            lease["spec"] = self._lease_obj()["spec"]
            try:
                await self.client.update(lease)
                self.is_leader = True
                return True
            except (ConflictError, NotFoundError):
                return False
        return False

    async def acquire(self) -> None:
        while not await self.try_acquire():
            await asyncio.sleep(self.retry_interval)
        log.info("acquired leadership as %s", self.identity)

    async def renew_loop(self) -> None:
        while True:
            await asyncio.sleep(self.renew_interval)
            if not await self.try_acquire():
                # lost the lease — in the reference losing leadership is fatal
                # for the replica; raise so the manager's task group surfaces it
                self.is_leader = False
                raise RuntimeError(f"lost leadership lease {self.name}")

    async def release(self) -> None:
        if not self.is_leader:
            return
        try:
            lease = await self.client.get(LEASE_API_VERSION, "Lease", self.namespace, self.name)
            spec = lease.get("spec") or {}
            if spec.get("holderIdentity") == self.identity:
                spec["holderIdentity"] = ""
                await self.client.update(lease)
        except Exception:
            pass
        self.is_leader = False
