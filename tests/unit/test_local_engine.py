"""LocalWorkflowEngine unit tests: Argo-shaped workflows executed as local
subprocesses (container/script/steps templates, retries, deadline, outputs)."""
import asyncio
import time

from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.kube.registry import WF_API_VERSION, WF_KIND
from active_monitor_amd.workflow import LocalWorkflowEngine


async def run_workflow(spec, timeout=15.0):
    client = MemoryClient(MemoryApiServer())
    engine = LocalWorkflowEngine(client)
    await engine.start()
    await client.create({
        "apiVersion": WF_API_VERSION, "kind": WF_KIND,
        "metadata": {"name": "wf", "namespace": "health"},
        "spec": spec,
    })
    deadline = time.monotonic() + timeout
    wf = None
    while time.monotonic() < deadline:
        wf = await client.get(WF_API_VERSION, WF_KIND, "health", "wf")
        if (wf.get("status") or {}).get("phase") in ("Succeeded", "Failed"):
            break
        await asyncio.sleep(0.02)
    await engine.stop()
    return wf["status"]


def test_container_template_success(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{"name": "main", "container": {"command": ["true"]}}],
    }))
    assert st["phase"] == "Succeeded"


def test_container_template_failure_message(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{"name": "main",
                       "container": {"command": ["sh", "-c", "echo oops >&2; exit 3"]}}],
    }))
    assert st["phase"] == "Failed"
    assert "exit code 3" in st["message"]
    assert "oops" in st["message"]


def test_script_template(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{"name": "main",
                       "script": {"command": ["sh"], "source": "test 1 -eq 1"}}],
    }))
    assert st["phase"] == "Succeeded"


def test_steps_sequential_and_parallel(run):
    st = run(run_workflow({
        "entrypoint": "pipeline",
        "templates": [
            {"name": "pipeline", "steps": [
                [{"name": "a", "template": "ok"}],
                [{"name": "b", "template": "ok"}, {"name": "c", "template": "ok"}],
            ]},
            {"name": "ok", "container": {"command": ["true"]}},
        ],
    }))
    assert st["phase"] == "Succeeded"


def test_steps_failure_propagates(run):
    st = run(run_workflow({
        "entrypoint": "pipeline",
        "templates": [
            {"name": "pipeline", "steps": [[{"name": "a", "template": "bad"}]]},
            {"name": "bad", "container": {"command": ["false"]}},
        ],
    }))
    assert st["phase"] == "Failed"


def test_retry_strategy_retries_then_fails(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{"name": "main", "retryStrategy": {"limit": 2},
                       "container": {"command": ["false"]}}],
    }))
    assert st["phase"] == "Failed"


def test_active_deadline_enforced(run):
    t0 = time.monotonic()
    st = run(run_workflow({
        "entrypoint": "main",
        "activeDeadlineSeconds": 1,
        "templates": [{"name": "main", "container": {"command": ["sleep", "30"]}}],
    }))
    assert st["phase"] == "Failed"
    assert "deadline" in st["message"]
    assert time.monotonic() - t0 < 10


def test_output_parameters_surface_in_status(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{
            "name": "main",
            "container": {"command": ["true"]},
            "outputs": {"parameters": [
                {"name": "metrics", "globalName": "metrics",
                 "value": '{"metrics": [{"name": "g", "value": 5}]}'},
            ]},
        }],
    }))
    assert st["phase"] == "Succeeded"
    params = st["outputs"]["parameters"]
    assert params[0]["name"] == "metrics"


def test_missing_entrypoint_fails(run):
    st = run(run_workflow({"entrypoint": "nope", "templates": []}))
    assert st["phase"] == "Failed"
    assert "entrypoint" in st["message"]


def test_unknown_command_fails_gracefully(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{"name": "main",
                       "container": {"command": ["definitely-not-a-binary-xyz"]}}],
    }))
    assert st["phase"] == "Failed"


def test_suspend_template(run):
    st = run(run_workflow({
        "entrypoint": "main",
        "templates": [{"name": "main", "suspend": {"duration": 0.1}}],
    }))
    assert st["phase"] == "Succeeded"
