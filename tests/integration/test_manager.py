"""Manager-level tests: worker bounding, restart resume, leader election
wiring, and a mini fleet soak."""
import asyncio

from active_monitor_amd import API_VERSION
from active_monitor_amd.engine import Manager
from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.workflow import ScriptedWorkflowEngine, always_succeed

from .conftest import Env, make_hc


def test_max_workers_bounds_concurrent_reconciles(run):
    """MaxConcurrentReconciles genuinely bounds reconcile concurrency (the
    SURVEY §7 design goal; the reference's timer goroutines escape its
    bound)."""

    async def go():
        client = MemoryClient(MemoryApiServer(), latency=0.01)
        manager = Manager(client, max_workers=2)
        peak = {"now": 0, "max": 0}
        orig = manager.reconciler.reconcile

        async def counting(ns, name, flags=None):
            peak["now"] += 1
            peak["max"] = max(peak["max"], peak["now"])
            try:
                return await orig(ns, name, flags)
            finally:
                peak["now"] -= 1

        manager.reconciler.reconcile = counting
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        await manager.start()
        for i in range(30):
            await client.create(make_hc(name=f"c-{i}", repeat=3600, timeout=2))
        deadline = asyncio.get_running_loop().time() + 20
        while asyncio.get_running_loop().time() < deadline:
            if manager.reconciler.completed_runs >= 30:
                break
            await asyncio.sleep(0.05)
        await manager.stop()
        await engine.stop()
        assert manager.reconciler.completed_runs >= 30
        assert peak["max"] <= 2

    run(go(), timeout=40)


def test_restart_resumes_from_status_checkpoint(run):
    """Timers are in-memory only; after a controller restart the informer
    resync re-runs due checks immediately (reference resume semantics,
    SURVEY §5 checkpoint/resume)."""

    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()

        m1 = Manager(client, max_workers=2)
        await m1.start()
        await client.create(make_hc(name="persist", repeat=3600, timeout=2))
        deadline = asyncio.get_running_loop().time() + 15
        while asyncio.get_running_loop().time() < deadline:
            obj = await client.get(API_VERSION, "HealthCheck", "health", "persist")
            if (obj.get("status") or {}).get("successCount", 0) >= 1:
                break
            await asyncio.sleep(0.02)
        await m1.stop()  # controller "crashes": all timers lost

        # counters survived in the status subresource
        obj = await client.get(API_VERSION, "HealthCheck", "health", "persist")
        count_before = obj["status"]["successCount"]
        assert count_before >= 1

        m2 = Manager(client, max_workers=2)
        await m2.start()
        # repeat interval (3600s) has NOT elapsed, but with no timer in the
        # new process the dedup conjunct fails → immediate re-run
        deadline = asyncio.get_running_loop().time() + 15
        ok = False
        while asyncio.get_running_loop().time() < deadline:
            obj = await client.get(API_VERSION, "HealthCheck", "health", "persist")
            if obj["status"]["successCount"] > count_before:
                ok = True
                break
            await asyncio.sleep(0.02)
        await m2.stop()
        await engine.stop()
        assert ok, "restart did not re-run the due check"

    run(go(), timeout=45)


def test_leader_election_single_active_manager(run):
    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        m1 = Manager(client, max_workers=1, leader_elect=True,
                     leader_identity="replica-1", namespace="health")
        await asyncio.wait_for(m1.start(), 10)  # acquires immediately
        assert m1.ready

        m2 = Manager(client, max_workers=1, leader_elect=True,
                     leader_identity="replica-2", namespace="health")
        start2 = asyncio.ensure_future(m2.start())
        await asyncio.sleep(0.3)
        assert not start2.done()  # blocked waiting for the lease
        await m1.stop()
        start2.cancel()
        await m2.stop()

    run(go(), timeout=30)


def test_mini_fleet_soak(run):
    """150 mixed CRs complete two full waves without loss (CPU-sized version
    of the 1000-CR bench fleet)."""

    async def go():
        async with Env(workers=8) as env:
            for i in range(150):
                cr = make_hc(name=f"fleet-{i:03d}", repeat=3600 if i % 3 else 0,
                             cron="@every 1h" if i % 3 == 0 else "", timeout=5)
                await env.create_hc(cr)
            rec = env.manager.reconciler

            async def settled():
                return rec.completed_runs >= 150

            await env.wait_for(settled, timeout=45, msg="first wave")
            for i in range(150):
                env.manager.queue.add_nowait(("health", f"fleet-{i:03d}"), {"timer"})

            async def all_advanced():
                objs = await env.client.list(API_VERSION, "HealthCheck", "health")
                return all(
                    (o.get("status") or {}).get("totalHealthCheckRuns", 0) >= 2
                    for o in objs
                )

            try:
                await env.wait_for(all_advanced, timeout=45, interval=0.2,
                                   msg="second wave on every CR")
            except AssertionError:
                objs = await env.client.list(API_VERSION, "HealthCheck", "health")
                lagging = {
                    o["metadata"]["name"]: (o.get("status") or {}).get(
                        "totalHealthCheckRuns", 0)
                    for o in objs
                    if (o.get("status") or {}).get("totalHealthCheckRuns", 0) < 2
                }
                raise AssertionError(
                    f"second wave incomplete; lagging CRs: {lagging}; "
                    f"queue={len(env.manager.queue)} "
                    f"watches={rec.active_watches()} "
                    f"completed={rec.completed_runs}"
                )

    run(go(), timeout=90)


def test_sharded_managers_partition_the_fleet(run):
    """N shard managers over one apiserver each own a disjoint CR subset and
    together cover the whole fleet exactly once."""

    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        managers = [
            Manager(client, max_workers=2, shard_index=i, shard_count=3)
            for i in range(3)
        ]
        for m in managers:
            await m.start()
        for i in range(60):
            await client.create(make_hc(name=f"sh-{i:03d}", repeat=3600, timeout=2))

        async def all_ran():
            objs = await client.list(API_VERSION, "HealthCheck", "health")
            return all(
                (o.get("status") or {}).get("successCount", 0) >= 1 for o in objs
            )

        deadline = asyncio.get_running_loop().time() + 25
        while asyncio.get_running_loop().time() < deadline:
            if await all_ran():
                break
            await asyncio.sleep(0.05)
        assert await all_ran(), "sharded fleet did not fully reconcile"
        # disjoint ownership: each CR reconciled by exactly one shard
        per_shard = [m.reconciler.completed_runs for m in managers]
        assert sum(per_shard) == 60
        assert all(c > 0 for c in per_shard)  # crc32 spreads 60 names over 3
        for m in managers:
            await m.stop()
        await engine.stop()

    run(go(), timeout=45)


def test_idle_fleet_is_quiescent(run):
    """Armed timers with far-future repeats must not burn reconciles — the
    controller is event-driven at idle."""

    async def go():
        async with Env(workers=4) as env:
            for i in range(40):
                await env.create_hc(make_hc(name=f"idle-{i}", repeat=3600, timeout=2))
            rec = env.manager.reconciler

            async def settled():
                return rec.completed_runs >= 40

            await env.wait_for(settled, timeout=20, msg="initial runs")
            await asyncio.sleep(1.0)  # let status-update reconciles drain
            count = rec.reconcile_count
            await asyncio.sleep(2.0)
            assert rec.reconcile_count == count, "reconciles while idle"
            assert len(env.manager.queue) == 0

    run(go(), timeout=45)
