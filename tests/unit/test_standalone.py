"""Standalone apiserver process contract (kube/standalone.py): the READY
line, the serving endpoint, and the bench policy the wire regime relies on."""
import asyncio
import json
import sys

from active_monitor_amd.kube.standalone import bench_policy


def test_bench_policy_name_decoding():
    p = bench_policy(0.2)
    wf = lambda name: {"metadata": {"name": name}}
    # remedy workflows always succeed
    assert p(wf("hc-00001-remedy-wf-abcde")) == ("Succeeded", "")
    # CRs with index%100 < 20 fail their checks
    assert p(wf("hc-00019-wf-abcde")) == ("Failed", "synthetic failure")
    assert p(wf("hc-00119-wf-abcde")) == ("Failed", "synthetic failure")
    # the rest succeed
    assert p(wf("hc-00020-wf-abcde")) == ("Succeeded", "")
    assert p(wf("hc-00099-wf-abcde")) == ("Succeeded", "")
    # non-bench names succeed (no accidental failures)
    assert p(wf("something-else-wf-x")) == ("Succeeded", "")
    # 0 fraction never fails
    assert bench_policy(0.0)(wf("hc-00000-wf-x")) == ("Succeeded", "")


def test_standalone_process_ready_contract(run):
    async def go():
        proc = await asyncio.create_subprocess_exec(
            sys.executable, "-m", "active_monitor_amd.kube.standalone",
            "--engine", "none",
            stdout=asyncio.subprocess.PIPE, stderr=asyncio.subprocess.DEVNULL,
        )
        try:
            line = await asyncio.wait_for(proc.stdout.readline(), 60)
            assert line.startswith(b"READY ")
            info = json.loads(line[len(b"READY "):])
            assert info["url"].startswith("http://127.0.0.1:")
            assert info["port"] > 0

            # it serves the k8s API surface
            from active_monitor_amd.kube.http import HttpClient

            client = HttpClient(info["url"], qps=0)
            await client.start()
            await client.ping()
            items = await client.list(
                "activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health")
            assert items == []
            await client.close()
        finally:
            proc.terminate()
            await proc.wait()

    run(go(), timeout=90)
