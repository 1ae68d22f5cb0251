"""Shard failure takeover.

Two cooperating controller shards over one apiserver; when one dies, the
survivor must adopt the orphaned half of the keyspace within a lease window,
and a restarted shard must reclaim its keys gracefully (preferredHolder
handshake) so exactly one process drives each CR at all times.
"""
import asyncio

import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.engine import Manager
from active_monitor_amd.engine.shards import shard_of
from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.workflow import ScriptedWorkflowEngine, always_succeed

from .conftest import make_hc

FAST = dict(shard_lease_duration=1.2, shard_renew_interval=0.15)


def _mgr(server, idx, count=2, **kw):
    return Manager(
        MemoryClient(server), max_workers=2, shard_index=idx, shard_count=count,
        shard_ha=True, leader_identity=f"mgr-{idx}",
        **{**FAST, **kw},
    )


def _names_by_shard(n, count):
    """Generate CR names until both shards have a few."""
    buckets = {i: [] for i in range(count)}
    i = 0
    while any(len(v) < 3 for v in buckets.values()):
        name = f"ha-{i:03d}"
        buckets[shard_of(name, count)].append(name)
        i += 1
    return buckets


async def _runs(client, name):
    obj = await client.get(API_VERSION, "HealthCheck", "health", name)
    return (obj.get("status") or {}).get("totalHealthCheckRuns", 0)


async def _wait(pred, timeout=15.0, msg="condition"):
    deadline = asyncio.get_running_loop().time() + timeout
    while asyncio.get_running_loop().time() < deadline:
        if await pred():
            return
        await asyncio.sleep(0.05)
    raise AssertionError(f"timed out waiting for {msg}")


def test_survivor_adopts_dead_shard(run):
    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        buckets = _names_by_shard(6, 2)
        for names in buckets.values():
            for n in names:
                await client.create(make_hc(name=n, repeat=1, timeout=2))

        m0 = _mgr(server, 0)
        m1 = _mgr(server, 1)
        await m0.start()
        await m1.start()
        try:
            # both shards drive their own keys
            a, b = buckets[0][0], buckets[1][0]
            await _wait(lambda: _runs(client, a), msg="shard0 first run")
            await _wait(lambda: _runs(client, b), msg="shard1 first run")
            assert m0.coordinator.owned == {0}
            assert m1.coordinator.owned == {1}

            # shard 1 CRASHES (no lease release)
            m1.coordinator.crash()
            m1.reconciler.stop_all()
            for t in m1._tasks:
                t.cancel()

            # survivor adopts within the lease window and keeps shard-1 CRs
            # cycling (runs keep advancing under m0)
            await _wait(lambda: _ownership(m0, {0, 1}), msg="adoption")
            before = await _runs(client, b)
            await _wait(
                lambda: _advanced(client, b, before),
                msg="adopted CR cycles under the survivor",
            )
        finally:
            await m0.stop()
            await m1.stop()
            await engine.stop()

    async def _ownership(m, want):
        return m.coordinator.owned == want

    async def _advanced(client, name, before):
        return await _runs(client, name) > before

    run(go(), timeout=60)


def test_restarted_shard_reclaims_its_keys(run):
    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        buckets = _names_by_shard(6, 2)
        for names in buckets.values():
            for n in names:
                await client.create(make_hc(name=n, repeat=1, timeout=2))

        m0 = _mgr(server, 0)
        await m0.start()
        m1 = None
        try:
            # alone, shard 0 adopts shard 1 (its lease never existed → free)
            await _wait(lambda: _owned(m0, {0, 1}), msg="initial adoption")
            b = buckets[1][0]
            await _wait(
                lambda: _runs_pos(client, b), msg="shard-1 CR runs under m0"
            )

            # the rightful shard-1 process arrives: preferredHolder handshake
            m1 = _mgr(server, 1)
            await asyncio.wait_for(m1.start(), 30)  # blocks until reclaimed
            await _wait(lambda: _owned(m0, {0}), msg="m0 dropped shard 1")
            assert m1.coordinator.owned == {1}
            # m0 released shard-1 state: no timers/watches for its keys
            for n in buckets[1]:
                assert m0.reconciler.get_timer_by_name(n, "health") is None

            # and the reclaimed CRs keep cycling under m1
            before = await _runs(client, b)
            await _wait(
                lambda: _advanced(client, b, before),
                msg="reclaimed CR cycles under m1",
            )
        finally:
            await m0.stop()
            if m1 is not None:
                await m1.stop()
            await engine.stop()

    async def _owned(m, want):
        return m.coordinator.owned == want

    async def _runs_pos(client, name):
        return await _runs(client, name) > 0

    async def _advanced(client, name, before):
        return await _runs(client, name) > before

    run(go(), timeout=90)


def test_home_shard_lease_loss_is_fatal(run):
    """A process that cannot renew its HOME lease must go fatal, not keep
    reconciling a shard someone else now owns."""

    async def go():
        server = MemoryApiServer()
        m0 = _mgr(server, 0)
        await m0.start()
        try:
            # forcibly steal the home lease (fresh renew, different holder)
            from active_monitor_amd.api.types import k8s_now

            name = m0.coordinator._lease_name(0)
            lease = server.get("coordination.k8s.io/v1", "Lease", "default", name)
            lease["spec"]["holderIdentity"] = "intruder"
            lease["spec"]["renewTime"] = k8s_now()
            server.update(lease)

            await asyncio.wait_for(m0.fatal.wait(), 15)
            assert "home shard" in (m0.fatal_reason or "")
        finally:
            await m0.stop()

    run(go(), timeout=60)


def test_double_crash_single_survivor_adopts_all(run):
    """3 shards, 2 crash: the lone survivor must end up owning the whole
    keyspace and cycling every CR."""

    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        buckets = _names_by_shard(9, 3)
        for names in buckets.values():
            for n in names:
                await client.create(make_hc(name=n, repeat=1, timeout=2))

        managers = [_mgr(server, i, count=3) for i in range(3)]
        for m in managers:
            await m.start()
        try:
            await _wait(lambda: _all_home(managers), 20, "all homes held")

            # shards 1 and 2 crash simultaneously
            for m in managers[1:]:
                m.coordinator.crash()
                m.reconciler.stop_all()
                for t in m._tasks:
                    t.cancel()

            await _wait(lambda: _owns_all(managers[0]), 30, "survivor owns 0,1,2")
            # a CR from each dead shard keeps cycling under the survivor
            for shard in (1, 2):
                name = buckets[shard][0]
                before = await _runs(client, name)
                await _wait(
                    lambda n=name, b=before: _advanced2(client, n, b), 30,
                    f"shard-{shard} CR cycles under the survivor",
                )
        finally:
            for m in managers:
                await m.stop()
            await engine.stop()

    async def _all_home(managers):
        return all(m.coordinator.owned == {i} for i, m in enumerate(managers))

    async def _owns_all(m):
        return m.coordinator.owned == {0, 1, 2}

    async def _advanced2(client, name, before):
        return await _runs(client, name) > before

    run(go(), timeout=90)
