"""GPU-box validation tier.

The workload is pure Kubernetes control-plane (BASELINE.json: no GPU code
paths); these tests validate the framework end-to-end on the MI355X Linux
host — device visibility, the full controller cycle, the local subprocess
workflow executor, and a short bench run producing the contract JSON line.
"""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu


def test_device_visible_and_host_sane():
    import torch

    assert torch.cuda.is_available(), "expected an MI355X visible as cuda:0"
    x = torch.randn(1024, device="cuda:0")
    torch.cuda.synchronize()
    assert x.shape[0] == 1024


def test_smoke_cycle_on_gpu_host():
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(__file__))))
    import __graft_entry__ as entry

    entry.build()
    entry.smoke()


def test_short_bench_contract_line():
    repo = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"),
         "--crs", "200", "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=repo,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    data = json.loads(line)
    assert data["metric"] == "sustained_healthcheck_cycles_per_sec"
    assert data["value"] > 0
    assert data["n_gpus"] == 1
    assert data["config"]["p50_reconcile_latency_ms"] > 0


def test_local_workflow_engine_runs_subprocesses(run):
    """The local executor actually runs commands on the host."""
    import asyncio

    from active_monitor_amd.kube import MemoryApiServer, MemoryClient
    from active_monitor_amd.workflow import LocalWorkflowEngine

    async def go():
        client = MemoryClient(MemoryApiServer())
        engine = LocalWorkflowEngine(client)
        await engine.start()
        await client.create({
            "apiVersion": "argoproj.io/v1alpha1",
            "kind": "Workflow",
            "metadata": {"name": "real-wf", "namespace": "health"},
            "spec": {
                "entrypoint": "main",
                "templates": [
                    {"name": "main", "container": {"command": ["true"]}},
                ],
            },
        })
        import time
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            wf = await client.get("argoproj.io/v1alpha1", "Workflow", "health", "real-wf")
            if (wf.get("status") or {}).get("phase") in ("Succeeded", "Failed"):
                break
            await asyncio.sleep(0.05)
        await engine.stop()
        return wf["status"]["phase"]

    assert run(go()) == "Succeeded"


def test_native_extension_loaded_on_gpu_host():
    """On the benchmark host the native hot path must be the one running —
    no silent pure-Python fallback."""
    from active_monitor_amd.utils import fastcopy

    assert fastcopy.NATIVE, "_amcore.so not loaded on the GPU host"
    from active_monitor_amd import _amcore

    assert _amcore.deep_copy({"a": [1]}) == {"a": [1]}
