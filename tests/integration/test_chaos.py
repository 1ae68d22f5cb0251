"""Chaos/convergence test: random workflow latencies and failures, CR
deletions and spec edits mid-flight — the controller must converge with no
leaked watches, no stuck queue, and consistent status on every survivor."""
import asyncio
import random

from active_monitor_amd import API_VERSION

from .conftest import Env, make_hc


def test_chaos_convergence(run):
    rng = random.Random(1234)

    def chaotic_policy(wf):
        # ~30% failures; the engine applies its own per-workflow delay
        if rng.random() < 0.3:
            return ("Failed", "chaos failure")
        return ("Succeeded", "")

    async def go():
        async with Env(workers=6, policy=chaotic_policy, engine_delay=0.02) as env:
            names = [f"chaos-{i:03d}" for i in range(60)]
            for i, name in enumerate(names):
                await env.create_hc(make_hc(
                    name=name, repeat=3600 if i % 4 else 0,
                    cron="@every 1h" if i % 4 == 0 else "", timeout=3,
                    remedy=(i % 10 == 0),
                ))
            rec = env.manager.reconciler

            async def first_wave():
                return rec.completed_runs >= 60

            await env.wait_for(first_wave, timeout=40, msg="first chaotic wave")

            # chaos: delete a fifth, edit a fifth, re-trigger everyone
            doomed = names[::5]
            for name in doomed:
                await env.client.delete(API_VERSION, "HealthCheck", "health", name)
            for name in names[1::5]:
                obj = await env.client.get(API_VERSION, "HealthCheck", "health", name)
                obj["spec"]["description"] = "edited mid-flight"
                await env.client.update(obj)
            survivors = [n for n in names if n not in doomed]
            for name in survivors:
                env.manager.queue.add_nowait(("health", name), {"timer"})

            async def converged():
                objs = await env.client.list(API_VERSION, "HealthCheck", "health")
                if len(objs) != len(survivors):
                    return False
                return all(
                    (o.get("status") or {}).get("totalHealthCheckRuns", 0) >= 2
                    for o in objs
                )

            await env.wait_for(converged, timeout=40, interval=0.2, msg="convergence")

            # no workflows left for deleted CRs (ownerRef cascade), and their
            # timers are gone
            wfs = await env.workflows()
            for name in doomed:
                assert all(not w["metadata"]["name"].startswith(name + "-") for w in wfs)

            async def doomed_cleanup():
                return all(rec.get_timer_by_name(n) is None for n in doomed)

            await env.wait_for(doomed_cleanup, timeout=15, msg="doomed timers gone")

            # let in-flight work drain; watches for survivors finish
            await rec.drain(timeout=20)
            assert rec.active_watches() == 0

            # status sanity on every survivor
            for name in survivors[::7]:
                hc = await env.get_hc(name)
                st = hc.status
                assert st.total_healthcheck_runs == st.success_count + st.failed_count
                assert st.status in ("Succeeded", "Failed")
                assert st.started_at and st.finished_at

    run(go(), timeout=120)
