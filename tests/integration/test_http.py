"""HTTP layer tests: HttpClient against the ApiServerFrontend — the full-loop
equivalent of running against a real kube-apiserver."""
import asyncio

import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.engine import Manager
from active_monitor_amd.kube import (
    ConflictError,
    MemoryApiServer,
    MemoryClient,
    NotFoundError,
)
from active_monitor_amd.kube.http import HttpClient
from active_monitor_amd.kube.server import ApiServerFrontend
from active_monitor_amd.workflow import ScriptedWorkflowEngine, always_succeed

from .conftest import make_hc


class HttpEnv:
    def __init__(self):
        self.server = MemoryApiServer()
        self.frontend = ApiServerFrontend(self.server)

    async def __aenter__(self):
        await self.frontend.start()
        self.client = HttpClient(self.frontend.url)
        await self.client.start()
        return self

    async def __aexit__(self, *exc):
        await self.client.close()
        await self.frontend.stop()


def test_http_crud_roundtrip(run):
    async def go():
        async with HttpEnv() as env:
            c = env.client
            created = await c.create(make_hc(name="h1"))
            assert created["metadata"]["uid"]

            got = await c.get(API_VERSION, "HealthCheck", "health", "h1")
            assert got["spec"]["repeatAfterSec"] == 1

            got["spec"]["repeatAfterSec"] = 99
            updated = await c.update(got)
            assert updated["spec"]["repeatAfterSec"] == 99

            # stale rv → conflict
            with pytest.raises(ConflictError):
                await c.update(got)

            fresh = await c.get(API_VERSION, "HealthCheck", "health", "h1")
            fresh["status"] = {"status": "Succeeded", "successCount": 2}
            out = await c.update_status(fresh)
            assert out["status"]["successCount"] == 2

            items = await c.list(API_VERSION, "HealthCheck", "health")
            assert len(items) == 1

            await c.delete(API_VERSION, "HealthCheck", "health", "h1")
            with pytest.raises(NotFoundError):
                await c.get(API_VERSION, "HealthCheck", "health", "h1")

    run(go(), timeout=30)


def test_http_cluster_scoped_and_selectors(run):
    async def go():
        async with HttpEnv() as env:
            c = env.client
            await c.create({
                "apiVersion": "rbac.authorization.k8s.io/v1",
                "kind": "ClusterRole",
                "metadata": {"name": "cr1", "labels": {"workflows.argoproj.io/managed-by": "active-monitor"}},
                "rules": [],
            })
            got = await c.get("rbac.authorization.k8s.io/v1", "ClusterRole", "", "cr1")
            assert got["metadata"]["name"] == "cr1"
            sel = await c.list(
                "rbac.authorization.k8s.io/v1", "ClusterRole",
                label_selector="workflows.argoproj.io/managed-by=active-monitor",
            )
            assert len(sel) == 1
            none = await c.list(
                "rbac.authorization.k8s.io/v1", "ClusterRole",
                label_selector="workflows.argoproj.io/managed-by=other",
            )
            assert none == []

    run(go(), timeout=30)


def test_http_watch_stream(run):
    async def go():
        async with HttpEnv() as env:
            c = env.client
            sub = c.watch(API_VERSION, "HealthCheck", "health")
            await asyncio.sleep(0.2)  # let the stream connect
            await c.create(make_hc(name="w1"))
            ev = await asyncio.wait_for(sub.__anext__(), 10)
            assert ev["type"] == "ADDED"
            assert ev["object"]["metadata"]["name"] == "w1"
            await c.delete(API_VERSION, "HealthCheck", "health", "w1")
            while True:
                ev = await asyncio.wait_for(sub.__anext__(), 10)
                if ev["type"] == "DELETED":
                    break
            sub.close()

    run(go(), timeout=30)


def test_full_controller_over_http(run):
    """The entire controller stack driven through HTTP — manager, reconciler,
    watch hub and RBAC all talk to the apiserver over the wire, while the
    workflow engine plays the in-cluster Argo controller."""

    async def go():
        async with HttpEnv() as env:
            # the engine runs inside the "cluster" (direct store access),
            # like Argo's controller would
            engine = ScriptedWorkflowEngine(MemoryClient(env.server), policy=always_succeed)
            await engine.start()
            manager = Manager(env.client, max_workers=4)
            await manager.start()
            try:
                await env.client.create(make_hc(name="over-http", repeat=1, timeout=2))
                deadline = asyncio.get_running_loop().time() + 20
                hc = None
                while asyncio.get_running_loop().time() < deadline:
                    obj = await env.client.get(API_VERSION, "HealthCheck", "health", "over-http")
                    if (obj.get("status") or {}).get("successCount", 0) >= 2:
                        hc = obj
                        break
                    await asyncio.sleep(0.05)
                assert hc is not None, "no completed cycles over HTTP"
                assert hc["status"]["status"] == "Succeeded"
                sa = await env.client.get("v1", "ServiceAccount", "health", "check-sa")
                assert sa["metadata"]["labels"]["workflows.argoproj.io/managed-by"] == "active-monitor"
            finally:
                await manager.stop()
                await engine.stop()

    run(go(), timeout=40)


def test_discovery_endpoints_for_kubectl(run):
    """/api, /apis, /version and per-group APIResourceLists — enough for
    `kubectl --server=<frontend-url>` to resolve `hc` and list resources."""
    import aiohttp

    async def go():
        async with HttpEnv() as env:
            async with aiohttp.ClientSession() as s:
                async def get(path):
                    async with s.get(env.frontend.url + path) as r:
                        assert r.status == 200, path
                        return await r.json()

                api = await get("/api")
                assert api["versions"] == ["v1"]

                apis = await get("/apis")
                names = {g["name"] for g in apis["groups"]}
                assert {"activemonitor.keikoproj.io", "argoproj.io",
                        "rbac.authorization.k8s.io"} <= names

                core = await get("/api/v1")
                core_names = {r["name"] for r in core["resources"]}
                assert {"serviceaccounts", "events", "namespaces"} <= core_names

                am = await get("/apis/activemonitor.keikoproj.io/v1alpha1")
                hc = next(r for r in am["resources"] if r["name"] == "healthchecks")
                assert hc["shortNames"] == ["hc", "hcs"]
                assert any(r["name"] == "healthchecks/status" for r in am["resources"])

                ver = await get("/version")
                assert "gitVersion" in ver

    run(go(), timeout=30)


def test_http_error_paths(run):
    """Frontend negative paths: unknown resources 404 with a k8s Status body,
    malformed JSON bodies surface as errors, unsupported methods 405."""
    import aiohttp

    async def go():
        async with HttpEnv() as env:
            async with aiohttp.ClientSession() as s:
                base = env.frontend.url
                async with s.get(base + "/apis/nope.example/v1/widgets") as r:
                    assert r.status == 404
                    body = await r.json()
                    assert body["kind"] == "Status" and body["status"] == "Failure"
                async with s.post(
                    base + "/apis/activemonitor.keikoproj.io/v1alpha1/namespaces/health/healthchecks",
                    data=b"{not json", headers={"Content-Type": "application/json"},
                ) as r:
                    assert r.status == 400
                    assert (await r.json())["kind"] == "Status"
                # PATCH with an empty body is a 400; a valid merge patch on
                # a missing object is a 404 (PATCH itself is supported now)
                async with s.patch(
                    base + "/apis/activemonitor.keikoproj.io/v1alpha1/namespaces/health/healthchecks/x"
                ) as r:
                    assert r.status == 400
                async with s.patch(
                    base + "/apis/activemonitor.keikoproj.io/v1alpha1/namespaces/health/healthchecks/x",
                    data=b'{"spec":{"repeatAfterSec":9}}',
                    headers={"Content-Type": "application/merge-patch+json"},
                ) as r:
                    assert r.status == 404

    run(go(), timeout=30)


def test_table_responses_for_kubectl_get(run):
    """kubectl get requests Accept: ...;as=Table — the frontend renders the
    CRD's printcolumns server-side (reference healthcheck_types.go:71-76)."""
    import aiohttp

    async def go():
        async with HttpEnv() as env:
            await env.client.create(make_hc(name="tbl-1"))
            obj = await env.client.get(API_VERSION, "HealthCheck", "health", "tbl-1")
            obj["status"] = {"status": "Succeeded", "successCount": 4}
            await env.client.update_status(obj)
            accept = ("application/json;as=Table;v=v1;g=meta.k8s.io, "
                      "application/json")
            async with aiohttp.ClientSession() as s:
                base = (env.frontend.url
                        + "/apis/activemonitor.keikoproj.io/v1alpha1/"
                          "namespaces/health/healthchecks")
                async with s.get(base, headers={"Accept": accept}) as r:
                    table = await r.json()
                assert table["kind"] == "Table"
                names = [c["name"] for c in table["columnDefinitions"]]
                assert names[0] == "Name" and "LATEST STATUS" in names
                row = table["rows"][0]
                assert row["cells"][0] == "tbl-1"
                assert row["cells"][1] == "Succeeded"
                assert row["cells"][2] == 4
                assert row["object"]["kind"] == "PartialObjectMetadata"

                # single-object GET renders too; plain Accept stays JSON
                async with s.get(base + "/tbl-1", headers={"Accept": accept}) as r:
                    one = await r.json()
                assert one["kind"] == "Table" and len(one["rows"]) == 1
                async with s.get(base) as r:
                    plain = await r.json()
                assert plain["kind"] == "HealthCheckList"

    run(go(), timeout=30)
