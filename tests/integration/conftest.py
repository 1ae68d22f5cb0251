"""Integration-test environment: in-memory apiserver + real Manager + a
workflow engine — the equivalent of the reference's envtest suite bootstrap
(internal/controllers/suite_test.go:67-134)."""
import asyncio
import time

import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.api import HealthCheck
from active_monitor_amd.engine import Manager
from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.workflow import ScriptedWorkflowEngine, always_succeed

HC = (API_VERSION, "HealthCheck")
WF = ("argoproj.io/v1alpha1", "Workflow")


class Env:
    """One controller + apiserver + optional workflow engine."""

    def __init__(self, workers=4, policy=always_succeed, engine=True, engine_delay=0.0,
                 latency=0.0):
        self.server = MemoryApiServer()
        self.client = MemoryClient(self.server, latency=latency)
        self.manager = Manager(self.client, max_workers=workers)
        self.engine = (
            ScriptedWorkflowEngine(self.client, policy=policy, delay=engine_delay)
            if engine
            else None
        )

    async def __aenter__(self):
        if self.engine is not None:
            await self.engine.start()
        await self.manager.start()
        return self

    async def __aexit__(self, *exc):
        await self.manager.stop()
        if self.engine is not None:
            await self.engine.stop()

    # -- helpers ----------------------------------------------------------

    async def create_hc(self, obj):
        if isinstance(obj, HealthCheck):
            obj = obj.to_dict()
        return await self.client.create(obj)

    async def get_hc(self, name, ns="health"):
        return HealthCheck.from_dict(await self.client.get(API_VERSION, "HealthCheck", ns, name))

    async def wait_for(self, pred, timeout=20.0, interval=0.02, msg="condition"):
        """Poll an async predicate until truthy."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            v = await pred()
            if v:
                return v
            await asyncio.sleep(interval)
        raise AssertionError(f"timed out waiting for {msg}")

    async def workflows(self, ns="health"):
        return await self.client.list(*WF, ns)


@pytest.fixture
def env_factory():
    return Env


INLINE_WF = """\
apiVersion: argoproj.io/v1alpha1
kind: Workflow
spec:
  entrypoint: start
  templates:
    - name: start
      container:
        image: busybox
        command: [echo, hello]
"""

INLINE_WF_WITH_LABELS = """\
apiVersion: argoproj.io/v1alpha1
kind: Workflow
metadata:
  labels:
    team: sre
    tier: "1"
spec:
  entrypoint: start
  templates:
    - name: start
      container:
        image: busybox
        command: [echo, hello]
"""


def make_hc(name="hello-check", ns="health", repeat=1, cron="", level="cluster",
            sa="check-sa", timeout=2, remedy=False, remedy_sa="remedy-sa",
            inline=INLINE_WF, remedy_inline=INLINE_WF, extra_spec=None):
    spec = {
        "workflow": {
            "generateName": f"{name}-wf-",
            "resource": {
                "namespace": ns,
                "serviceAccount": sa,
                "source": {"inline": inline},
            },
        },
        "level": level,
    }
    if timeout:
        spec["workflow"]["workflowtimeout"] = timeout
    if repeat:
        spec["repeatAfterSec"] = repeat
    if cron:
        spec["schedule"] = {"cron": cron}
    if remedy:
        spec["remedyworkflow"] = {
            "generateName": f"{name}-remedy-wf-",
            "workflowtimeout": timeout,
            "resource": {
                "namespace": ns,
                "serviceAccount": remedy_sa,
                "source": {"inline": remedy_inline},
            },
        }
    if extra_spec:
        spec.update(extra_spec)
    return {
        "apiVersion": API_VERSION,
        "kind": "HealthCheck",
        "metadata": {"name": name, "namespace": ns},
        "spec": spec,
    }
