"""Chaos over the wire: chaotic workflow outcomes, mid-flight CR deletes and
spec edits, repeated watch-stream kicks AND a full apiserver restart — all
at once, through real HTTP. The controller must converge: every surviving
CR keeps cycling, no fatal, no stuck queue, no leaked hub state."""
import asyncio
import random

from active_monitor_amd import API_VERSION
from active_monitor_amd.engine import Manager
from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.kube.http import HttpClient
from active_monitor_amd.kube.server import ApiServerFrontend
from active_monitor_amd.workflow import ScriptedWorkflowEngine

from .conftest import make_hc


def test_wire_chaos_convergence(run):
    rng = random.Random(4321)

    def chaotic_policy(wf):
        if rng.random() < 0.3:
            return ("Failed", "chaos failure")
        return ("Succeeded", "")

    async def go():
        store = MemoryApiServer()
        fe = ApiServerFrontend(store)
        await fe.start()
        port = fe.port
        engine = ScriptedWorkflowEngine(MemoryClient(store), policy=chaotic_policy)
        await engine.start()
        client = HttpClient(fe.url, qps=0)
        await client.start()
        manager = Manager(client, max_workers=4)
        names = [f"chaos-{i:02d}" for i in range(20)]
        try:
            for n in names:
                await client.create(make_hc(name=n, repeat=1, timeout=2))
            await manager.start()
            rec = manager.reconciler

            deleted = set()
            # chaos loop: kicks, deletes, edits, one full apiserver restart
            for round_no in range(6):
                await asyncio.sleep(0.7)
                fe.kick_watches()
                victim = rng.choice(names)
                if victim not in deleted and len(deleted) < 5:
                    deleted.add(victim)
                    try:
                        await client.delete(API_VERSION, "HealthCheck", "health", victim)
                    except Exception:
                        deleted.discard(victim)
                editee = rng.choice([n for n in names if n not in deleted])
                try:
                    obj = await client.get(API_VERSION, "HealthCheck", "health", editee)
                    obj["spec"]["description"] = f"edit-{round_no}"
                    await client.update(obj)
                except Exception:
                    pass
                if round_no == 3:  # full apiserver restart mid-chaos
                    await fe.stop()
                    await asyncio.sleep(1.0)
                    fe = ApiServerFrontend(store, port=port)
                    await fe.start()

            # convergence: every survivor advances past its current count
            survivors = [n for n in names if n not in deleted]
            baseline = {}
            for n in survivors:
                obj = await client.get(API_VERSION, "HealthCheck", "health", n)
                baseline[n] = (obj.get("status") or {}).get("totalHealthCheckRuns", 0)
            deadline = asyncio.get_running_loop().time() + 30
            pending = set(survivors)
            while pending:
                assert asyncio.get_running_loop().time() < deadline, (
                    f"{len(pending)} CRs stopped cycling: {sorted(pending)[:5]}"
                )
                for n in list(pending):
                    obj = await client.get(API_VERSION, "HealthCheck", "health", n)
                    if (obj.get("status") or {}).get("totalHealthCheckRuns", 0) > baseline[n]:
                        pending.discard(n)
                await asyncio.sleep(0.2)

            assert not manager.fatal.is_set()
            # deleted CRs left no timers behind
            for n in deleted:
                assert rec.get_timer_by_name(n, "health") is None
        finally:
            await manager.stop()
            await engine.stop()
            await client.close()
            await fe.stop()

    run(go(), timeout=120)
