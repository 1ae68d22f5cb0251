"""Informer-cache read discipline (round 2): steady-state reconcile cycles
must not re-read HealthChecks or Workflows over the wire — the watch streams
already deliver them (controller-runtime cached-client shape) — and the hub
must release per-workflow state when a watch completes (leak canary)."""
import asyncio

from active_monitor_amd import API_VERSION

from .conftest import Env, make_hc


def test_steady_state_cycles_do_minimal_reads(run):
    async def go():
        async with Env(workers=4) as env:
            await env.create_hc(make_hc(name="cached", repeat=1, timeout=2))
            rec = env.manager.reconciler

            # settle: first run complete, caches warm
            await env.wait_for(lambda: _runs_at_least(env, 2), msg="two runs")

            gets_before = env.server.op_counts["get"]
            runs_before = rec.completed_runs
            await env.wait_for(
                lambda: _runs_at_least(env, runs_before + 5), msg="five more runs"
            )
            cycles = rec.completed_runs - runs_before
            gets = env.server.op_counts["get"] - gets_before
            # per cycle the reconcile fetch, completion re-fetch and workflow
            # polls are all cache-served; only conflict retries and the RBAC
            # ensure-TTL refresh (30s) may read — at 1s repeat that is ≈0
            assert gets <= cycles, (
                f"{gets} apiserver GETs for {cycles} cycles — cache not used"
            )

    async def _runs_at_least(env, n):
        return env.manager.reconciler.completed_runs >= n

    run(go(), timeout=40)


def test_hub_releases_state_after_watch_completion(run):
    async def go():
        async with Env(workers=2) as env:
            await env.create_hc(make_hc(name="hubrel", repeat=0, timeout=2,
                                        extra_spec={"repeatAfterSec": 3600}))
            rec = env.manager.reconciler
            await env.wait_for(lambda: _done(rec), msg="one completed run")
            # allow the finally-block forget to run
            await asyncio.sleep(0.1)
            hub = env.manager.wf_hub
            stale = [k for k in hub._last if k[1].startswith("hubrel-")]
            assert stale == [], f"hub retained completed-workflow state: {stale}"

    async def _done(rec):
        return rec.completed_runs >= 1

    run(go(), timeout=40)


def test_cache_serves_deletion_as_notfound(run):
    """Deleting a CR must flow through the cache as NotFound so the timer
    stops — the reconcile path reads the cache, not the wire."""

    async def go():
        async with Env(workers=2) as env:
            await env.create_hc(make_hc(name="cachedel", repeat=1, timeout=2))
            rec = env.manager.reconciler
            await env.wait_for(lambda: _done(rec), msg="first run")
            assert rec.get_timer_by_name("cachedel", "health") is not None
            await env.client.delete(API_VERSION, "HealthCheck", "health", "cachedel")
            await env.wait_for(
                lambda: _timer_gone(rec), msg="timer stopped after delete"
            )

    async def _done(rec):
        return rec.completed_runs >= 1

    async def _timer_gone(rec):
        return rec.get_timer_by_name("cachedel", "health") is None

    run(go(), timeout=40)


def test_periodic_resync_heals_lost_events(run):
    """The informer resync (controller-runtime SyncPeriod) must repair state
    a lost watch event left stale: a cache entry for a CR deleted while the
    event was lost gets cleaned up (timer stopped) at the next resync."""
    import asyncio

    from active_monitor_amd import API_VERSION
    from active_monitor_amd.engine import Manager
    from active_monitor_amd.kube import MemoryApiServer, MemoryClient
    from active_monitor_amd.workflow import ScriptedWorkflowEngine, always_succeed

    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        manager = Manager(client, max_workers=2, resync_period=0.8)
        await manager.start()
        rec = manager.reconciler
        try:
            from .conftest import make_hc

            await client.create(make_hc(name="rsync-1", repeat=3600, timeout=2))
            deadline = asyncio.get_running_loop().time() + 20
            while rec.completed_runs < 1:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.05)
            assert rec.get_timer_by_name("rsync-1", "health") is not None

            # simulate a lost DELETED event: remove from the store without
            # publishing (direct dict surgery — the watch never hears it)
            key = (API_VERSION, "HealthCheck", "health", "rsync-1")
            with server._lock:
                server._objects.pop(key)

            # the resync must discover the orphaned cache entry, reconcile
            # the NotFound, and stop the timer
            deadline = asyncio.get_running_loop().time() + 20
            while rec.get_timer_by_name("rsync-1", "health") is not None:
                assert asyncio.get_running_loop().time() < deadline, (
                    "resync did not heal the lost delete"
                )
                await asyncio.sleep(0.1)
            assert ("health", "rsync-1") not in manager.hc_cache
        finally:
            await manager.stop()
            await engine.stop()

    run(go(), timeout=60)


def test_read_your_writes_floor_rejects_stale_cache(run):
    """Deterministic repro of the lost-update race: after a status write
    produced rv N, a cache entry at rv<N (our own event still in flight)
    must NOT be served — the read goes direct instead."""
    import asyncio

    from active_monitor_amd import API_VERSION
    from active_monitor_amd.engine.reconciler import HealthCheckReconciler
    from active_monitor_amd.kube import MemoryApiServer, MemoryClient
    from active_monitor_amd.kube.client import FakeRecorder

    from .conftest import make_hc

    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        rec = HealthCheckReconciler(client, FakeRecorder())
        created = server.create(make_hc(name="ryw"))
        stale = {k: v for k, v in created.items()}  # rv N

        # a later write bumps the floor past the stale snapshot
        fresh = server.get(API_VERSION, "HealthCheck", "health", "ryw")
        fresh["status"] = {"successCount": 1}
        written = server.update_status(fresh)  # rv N+1
        rec.note_written("health", "ryw", written)

        # lookup serves the STALE entry (the event hasn't landed yet)
        rec.hc_lookup = lambda ns, name: stale
        got = await rec._get_hc("health", "ryw")
        # ...but the reconciler read the CURRENT object from the server
        assert (got.get("status") or {}).get("successCount") == 1
        assert got["metadata"]["resourceVersion"] == written["metadata"]["resourceVersion"]

        # once the cache catches up (rv ≥ floor) it is served again
        rec.hc_lookup = lambda ns, name: written
        got = await rec._get_hc("health", "ryw")
        assert got["metadata"]["resourceVersion"] == written["metadata"]["resourceVersion"]

    run(go(), timeout=20)
