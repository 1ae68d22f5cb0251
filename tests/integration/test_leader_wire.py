"""Leader-election failover end-to-end over the wire: two CLI controller
processes with --leader-elect against the standalone apiserver; the standby
must take over reconciling after the active leader is SIGKILLed (lease
expiry), and the fleet's schedules continue. This is the reference's
single-active-replica HA mode (cmd/main.go:87-88) under a real crash."""
import asyncio
import json
import os
import subprocess
import sys

from active_monitor_amd import API_VERSION
from active_monitor_amd.kube.http import HttpClient

from .conftest import make_hc

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _controller(url):
    env = dict(os.environ)
    env["AM_LEADER_LEASE_SECS"] = "2.5"
    env["AM_LEADER_RENEW_SECS"] = "0.3"
    return subprocess.Popen(
        [sys.executable, "-m", "active_monitor_amd.cmd.main",
         "--backend", "http", "--server", url,
         "--leader-elect", "--max-workers", "2", "--namespace", "health",
         "--metrics-bind-address", "0", "--health-probe-bind-address", "0"],
        cwd=REPO, env=env,
        stdout=subprocess.DEVNULL, stderr=subprocess.PIPE,
    )


def test_standby_takes_over_after_leader_crash(run):
    async def go():
        apiserver = await asyncio.create_subprocess_exec(
            sys.executable, "-m", "active_monitor_amd.kube.standalone",
            "--engine", "scripted-bench", "--remedy-frac", "0",
            stdout=asyncio.subprocess.PIPE, stderr=asyncio.subprocess.DEVNULL,
            cwd=REPO,
        )
        procs = []
        client = None
        try:
            line = await asyncio.wait_for(apiserver.stdout.readline(), 60)
            url = json.loads(line[len(b"READY "):])["url"]
            client = HttpClient(url, qps=0)
            await client.start()
            for i in range(3):
                await client.create(make_hc(name=f"le-{i}", repeat=1, timeout=2))

            async def runs(name="le-0"):
                obj = await client.get(API_VERSION, "HealthCheck", "health", name)
                return (obj.get("status") or {}).get("totalHealthCheckRuns", 0)

            async def holder():
                try:
                    lease = await client.get("coordination.k8s.io/v1", "Lease",
                                             "health", "689451f8.keikoproj.io")
                    return (lease.get("spec") or {}).get("holderIdentity") or ""
                except Exception:
                    return ""

            async def wait(pred, timeout, msg):
                deadline = asyncio.get_running_loop().time() + timeout
                while asyncio.get_running_loop().time() < deadline:
                    if await pred():
                        return
                    await asyncio.sleep(0.15)
                raise AssertionError(f"timed out: {msg}")

            # first replica leads and reconciles
            procs.append(_controller(url))
            await wait(lambda: _pos(runs), 30, "leader reconciles")
            leader_id = await holder()
            assert leader_id

            # standby joins: blocks on the lease, must NOT reconcile
            procs.append(_controller(url))
            await asyncio.sleep(1.5)
            assert await holder() == leader_id, "standby stole a live lease"

            # leader crashes (no release)
            procs[0].kill()
            procs[0].wait(10)

            # standby acquires within the lease window and the fleet resumes
            await wait(lambda: _new_holder(holder, leader_id), 20,
                       "standby acquired the lease")
            before = await runs()
            await wait(lambda: _advanced(runs, before), 30,
                       "fleet cycles under the new leader")
        finally:
            for p in procs:
                if p.poll() is None:
                    p.terminate()
            for p in procs:
                try:
                    p.wait(15)
                except subprocess.TimeoutExpired:
                    p.kill()
            if client is not None:
                await client.close()
            apiserver.terminate()
            await apiserver.wait()

    async def _pos(runs):
        return await runs() > 0

    async def _new_holder(holder, old):
        h = await holder()
        return h and h != old

    async def _advanced(runs, before):
        return await runs() > before

    run(go(), timeout=180)
