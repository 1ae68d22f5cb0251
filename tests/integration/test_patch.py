"""PATCH conformance: kubectl `patch` (merge-patch), `apply` (apply-patch
upsert) and the status-subresource isolation rules, against both the store
and the wire frontend."""
import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.kube import MemoryApiServer, NotFoundError
from active_monitor_amd.kube.http import HttpClient
from active_monitor_amd.kube.server import ApiServerFrontend

from .conftest import make_hc


def test_store_merge_patch_semantics(run):
    s = MemoryApiServer()
    s.create(make_hc(name="p1", repeat=5))
    out = s.patch(API_VERSION, "HealthCheck", "health", "p1",
                  {"spec": {"repeatAfterSec": 60},
                   "metadata": {"labels": {"a": "b"}}})
    assert out["spec"]["repeatAfterSec"] == 60
    assert out["spec"]["workflow"]["generateName"]  # merge kept siblings
    assert out["metadata"]["labels"]["a"] == "b"

    # null deletes (RFC 7386)
    out = s.patch(API_VERSION, "HealthCheck", "health", "p1",
                  {"metadata": {"labels": {"a": None}}})
    assert "a" not in (out["metadata"].get("labels") or {})

    # plain patch cannot touch status (subresource isolation)
    s.update_status({**s.get(API_VERSION, "HealthCheck", "health", "p1"),
                     "status": {"successCount": 3}})
    out = s.patch(API_VERSION, "HealthCheck", "health", "p1",
                  {"status": {"successCount": 99}})
    assert out["status"]["successCount"] == 3

    # status-subresource patch touches only status
    out = s.patch(API_VERSION, "HealthCheck", "health", "p1",
                  {"status": {"successCount": 7}, "spec": {"repeatAfterSec": 1}},
                  subresource="status")
    assert out["status"]["successCount"] == 7
    assert out["spec"]["repeatAfterSec"] == 60  # spec untouched via /status

    with pytest.raises(NotFoundError):
        s.patch(API_VERSION, "HealthCheck", "health", "missing", {"spec": {}})

    # upsert (server-side-apply shape) creates when absent
    out = s.patch(API_VERSION, "HealthCheck", "health", "applied",
                  make_hc(name="applied", repeat=4), upsert=True)
    assert out["metadata"]["uid"]


def test_patch_over_the_wire(run):
    async def go():
        server = MemoryApiServer()
        fe = ApiServerFrontend(server)
        await fe.start()
        client = HttpClient(fe.url, qps=0)
        await client.start()
        try:
            await client.create(make_hc(name="wp", repeat=5))
            out = await client.patch(API_VERSION, "HealthCheck", "health", "wp",
                                     {"spec": {"repeatAfterSec": 42}})
            assert out["spec"]["repeatAfterSec"] == 42
            got = await client.get(API_VERSION, "HealthCheck", "health", "wp")
            assert got["spec"]["repeatAfterSec"] == 42
            assert got["spec"]["workflow"]["resource"]["serviceAccount"]

            # apply-patch+yaml upserts a new object (kubectl apply shape)
            import aiohttp

            async with aiohttp.ClientSession() as s:
                body = (
                    "apiVersion: activemonitor.keikoproj.io/v1alpha1\n"
                    "kind: HealthCheck\n"
                    "metadata: {name: applied-wire, namespace: health}\n"
                    "spec: {repeatAfterSec: 3, level: cluster}\n"
                )
                async with s.patch(
                    fe.url + "/apis/activemonitor.keikoproj.io/v1alpha1/"
                             "namespaces/health/healthchecks/applied-wire",
                    data=body.encode(),
                    headers={"Content-Type": "application/apply-patch+yaml"},
                ) as r:
                    assert r.status == 200, await r.text()
            got = await client.get(API_VERSION, "HealthCheck", "health", "applied-wire")
            assert got["spec"]["repeatAfterSec"] == 3

            # json-patch op lists are rejected with 415
            async with aiohttp.ClientSession() as s:
                async with s.patch(
                    fe.url + "/apis/activemonitor.keikoproj.io/v1alpha1/"
                             "namespaces/health/healthchecks/wp",
                    data=b'[{"op":"replace","path":"/spec/repeatAfterSec","value":1}]',
                    headers={"Content-Type": "application/json-patch+json"},
                ) as r:
                    assert r.status == 415
        finally:
            await client.close()
            await fe.stop()

    run(go(), timeout=30)


def test_invalid_selectors_return_k8s_errors(run):
    """Malformed label/field selectors surface as 422 Status bodies over the
    wire, not connection drops."""
    import aiohttp

    from active_monitor_amd.kube.memory import MemoryApiServer
    from active_monitor_amd.kube.server import ApiServerFrontend

    async def go():
        fe = ApiServerFrontend(MemoryApiServer())
        await fe.start()
        try:
            base = (fe.url + "/apis/activemonitor.keikoproj.io/v1alpha1/"
                             "namespaces/health/healthchecks")
            async with aiohttp.ClientSession() as s:
                async with s.get(base + "?fieldSelector=bogus") as r:
                    assert r.status == 422
                    body = await r.json()
                    assert body["kind"] == "Status"
                async with s.get(base + "?labelSelector=no-equals-sign") as r:
                    assert r.status == 422
        finally:
            await fe.stop()

    run(go(), timeout=30)


def test_table_transform_edges(run):
    """Table rendering: empty lists, kinds without printcolumns, Status
    bodies pass through untouched."""
    from active_monitor_amd.kube.memory import MemoryApiServer
    from active_monitor_amd.kube.server import ApiServerFrontend

    fe = ApiServerFrontend(MemoryApiServer())
    empty = fe._to_table({"kind": "HealthCheckList", "items": [],
                          "metadata": {"resourceVersion": "5"}})
    assert empty["kind"] == "Table" and empty["rows"] == []

    generic = fe._to_table({
        "kind": "ServiceAccountList",
        "items": [{"metadata": {"name": "sa1", "creationTimestamp": "t"}}],
    })
    assert generic["rows"][0]["cells"] == ["sa1", "t"]  # default columns

    status = {"kind": "Status", "status": "Failure", "code": 404}
    assert fe._to_table(status) is status  # untouched
