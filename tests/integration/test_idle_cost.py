"""An idle fleet must cost (almost) nothing: CRs with long repeat intervals
sit on armed timers — no polling loops, no busy wakeups, no apiserver
traffic. This is what lets one controller hold tens of thousands of CRs."""
import asyncio
import time

from .conftest import Env, make_hc


def test_idle_fleet_burns_no_cpu_and_no_requests(run):
    async def go():
        async with Env(workers=4) as env:
            for i in range(100):
                await env.create_hc(make_hc(name=f"idle-{i:03d}", repeat=3600,
                                            timeout=30))
            rec = env.manager.reconciler

            async def settled():
                return rec.completed_runs >= 100

            await env.wait_for(settled, timeout=30, msg="first wave")
            # drain any event-pump stragglers
            await asyncio.sleep(0.3)

            ops_before = sum(env.server.op_counts.values())
            cpu_before = time.process_time()
            await asyncio.sleep(2.0)
            cpu_used = time.process_time() - cpu_before
            ops = sum(env.server.op_counts.values()) - ops_before

            # 100 armed timers, zero due work: a handful of best-effort event
            # writes at most, and single-digit-% CPU (scheduler ticks only)
            assert ops <= 5, f"{ops} apiserver ops while idle"
            assert cpu_used < 0.5, f"{cpu_used:.2f}s CPU over 2s idle wall"
            assert len(env.manager.queue) == 0
            assert len(rec.repeat_timers_by_name) == 100

    run(go(), timeout=60)
