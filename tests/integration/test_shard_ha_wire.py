"""Shard takeover end-to-end over the real wire: two controller PROCESSES
(the shipped CLI, --backend http --shard-ha) against the standalone
apiserver process; one is SIGKILLed and the survivor must adopt its keyspace
within the lease window. This is the production deployment shape — nothing
in-process, nothing shared but the apiserver."""
import asyncio
import json
import os
import signal
import subprocess
import sys

import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.engine.shards import shard_of
from active_monitor_amd.kube.http import HttpClient

from .conftest import make_hc

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
LEASE_API = "coordination.k8s.io/v1"


def _controller(url, idx):
    return subprocess.Popen(
        [sys.executable, "-m", "active_monitor_amd.cmd.main",
         "--backend", "http", "--server", url,
         "--shard-ha", "--shard-index", str(idx), "--shard-count", "2",
         "--shard-lease-duration", "2.0", "--shard-renew-interval", "0.3",
         "--max-workers", "2", "--namespace", "health",
         "--metrics-bind-address", "0", "--health-probe-bind-address", "0"],
        cwd=REPO, stdout=subprocess.DEVNULL, stderr=subprocess.PIPE,
    )


async def _wait(pred, timeout, msg):
    deadline = asyncio.get_running_loop().time() + timeout
    while asyncio.get_running_loop().time() < deadline:
        if await pred():
            return
        await asyncio.sleep(0.15)
    raise AssertionError(f"timed out: {msg}")


def test_process_kill_takeover_over_http(run):
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

            # a few CRs per shard
            names = {0: [], 1: []}
            i = 0
            while any(len(v) < 2 for v in names.values()):
                n = f"wirehc-{i:03d}"
                names[shard_of(n, 2)].append(n)
                i += 1
            for ns in names.values():
                for n in ns:
                    await client.create(make_hc(name=n, repeat=1, timeout=2))

            procs[:] = [_controller(url, 0), _controller(url, 1)]

            async def lease_holder(shard):
                try:
                    lease = await client.get(
                        LEASE_API, "Lease", "health",
                        f"active-monitor-shard-{shard}-of-2")
                    return (lease.get("spec") or {}).get("holderIdentity") or ""
                except Exception:
                    return ""

            async def runs(name):
                obj = await client.get(API_VERSION, "HealthCheck", "health", name)
                return (obj.get("status") or {}).get("totalHealthCheckRuns", 0)

            # both shards held by their home processes, both making progress
            await _wait(
                lambda: _both_held(lease_holder), 30, "both shard leases held")
            h0, h1 = await lease_holder(0), await lease_holder(1)
            assert h0 != h1
            a, b = names[0][0], names[1][0]
            await _wait(lambda: _pos(runs, a), 30, "shard-0 CR cycles")
            await _wait(lambda: _pos(runs, b), 30, "shard-1 CR cycles")

            # SIGKILL the shard-1 process: no lease release, no cleanup
            procs[1].kill()
            procs[1].wait(10)

            # the survivor adopts shard 1 within the lease window and its
            # CRs keep cycling
            await _wait(
                lambda: _adopted(lease_holder, h0), 20,
                "survivor holds the dead shard's lease")
            before = await runs(b)
            await _wait(
                lambda: _advanced(runs, b, before), 30,
                "orphaned CR cycles under the survivor")
        finally:
            for p in procs:
                if p.poll() is None:
                    p.send_signal(signal.SIGTERM)
            for p in procs:
                try:
                    p.wait(15)
                except subprocess.TimeoutExpired:
                    p.kill()
            if client is not None:
                await client.close()
            apiserver.terminate()
            await apiserver.wait()

    async def _both_held(lease_holder):
        return await lease_holder(0) and await lease_holder(1)

    async def _pos(runs, name):
        return await runs(name) > 0

    async def _adopted(lease_holder, survivor):
        return await lease_holder(1) == survivor

    async def _advanced(runs, name, before):
        return await runs(name) > before

    run(go(), timeout=180)
