"""In-memory apiserver tests (the envtest-equivalent backend)."""
import asyncio

import pytest

from active_monitor_amd.kube import (
    AlreadyExistsError,
    ConflictError,
    MemoryApiServer,
    MemoryClient,
    NotFoundError,
    ignore_not_found,
    is_storage_error,
)


def hc(name="check-1", ns="health", **meta):
    return {
        "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
        "kind": "HealthCheck",
        "metadata": {"name": name, "namespace": ns, **meta},
        "spec": {"repeatAfterSec": 30, "workflow": {}},
    }


def test_create_get_roundtrip():
    s = MemoryApiServer()
    created = s.create(hc())
    assert created["metadata"]["uid"].startswith("uid-")
    assert created["metadata"]["resourceVersion"] == "1"
    got = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    assert got["spec"]["repeatAfterSec"] == 30


def test_create_duplicate_fails():
    s = MemoryApiServer()
    s.create(hc())
    with pytest.raises(AlreadyExistsError):
        s.create(hc())


def test_generate_name():
    s = MemoryApiServer()
    obj = {"apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
           "metadata": {"generateName": "wf-", "namespace": "health"}, "spec": {}}
    a = s.create(obj)
    b = s.create(obj)
    assert a["metadata"]["name"].startswith("wf-")
    assert a["metadata"]["name"] != b["metadata"]["name"]


def test_not_found_message_shape():
    s = MemoryApiServer()
    with pytest.raises(NotFoundError, match=r'healthchecks\.activemonitor\.keikoproj\.io "nope" not found'):
        s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "nope")


def test_update_conflict_on_stale_rv():
    s = MemoryApiServer()
    s.create(hc())
    fresh = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    fresh["spec"]["repeatAfterSec"] = 60
    s.update(fresh)  # ok
    with pytest.raises(ConflictError):
        s.update(fresh)  # stale rv now


def test_status_subresource_separation():
    s = MemoryApiServer()
    s.create(hc())
    obj = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")

    # plain update cannot set status
    obj["status"] = {"status": "Succeeded"}
    s.update(obj)
    got = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    assert "status" not in got

    # update_status sets only status
    got["status"] = {"status": "Succeeded", "successCount": 1}
    got["spec"]["repeatAfterSec"] = 999
    s.update_status(got)
    final = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    assert final["status"]["successCount"] == 1
    assert final["spec"]["repeatAfterSec"] == 30  # spec untouched by status write


def test_generation_bumps_only_on_spec_change():
    s = MemoryApiServer()
    s.create(hc())
    o = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    o["metadata"]["labels"] = {"x": "y"}
    o = s.update(o)
    assert o["metadata"]["generation"] == 1
    o["spec"]["repeatAfterSec"] = 7
    o = s.update(o)
    assert o["metadata"]["generation"] == 2


def test_owner_reference_cascade_gc():
    s = MemoryApiServer()
    owner = s.create(hc())
    wf = {
        "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
        "metadata": {"name": "wf-1", "namespace": "health",
                     "ownerReferences": [{"uid": owner["metadata"]["uid"],
                                          "kind": "HealthCheck", "name": "check-1",
                                          "controller": True}]},
        "spec": {},
    }
    s.create(wf)
    s.delete("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    with pytest.raises(NotFoundError):
        s.get("argoproj.io/v1alpha1", "Workflow", "health", "wf-1")


def test_label_selector_list():
    s = MemoryApiServer()
    s.create(hc("a", labels={"managed-by": "active-monitor"}))
    s.create(hc("b", labels={"managed-by": "other"}))
    s.create(hc("c"))
    names = {o["metadata"]["name"]
             for o in s.list("activemonitor.keikoproj.io/v1alpha1", "HealthCheck",
                             "health", "managed-by=active-monitor")}
    assert names == {"a"}


def test_finalizer_defers_delete():
    s = MemoryApiServer()
    s.create(hc(finalizers=["keep"]))
    s.delete("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    obj = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    assert obj["metadata"]["deletionTimestamp"]
    obj["metadata"]["finalizers"] = []
    s.update(obj)
    with pytest.raises(NotFoundError):
        s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")


def test_watch_stream(run):
    async def go():
        s = MemoryApiServer()
        sub = s.watch("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health")
        s.create(hc())
        obj = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
        obj["spec"]["repeatAfterSec"] = 5
        s.update(obj)
        s.delete("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
        events = []
        for _ in range(3):
            events.append(await asyncio.wait_for(sub.__anext__(), 5))
        sub.close()
        return [e["type"] for e in events]

    assert run(go()) == ["ADDED", "MODIFIED", "DELETED"]


def test_watch_filters_namespace_and_kind(run):
    async def go():
        s = MemoryApiServer()
        sub = s.watch("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health")
        s.create(hc("other-ns-check", ns="default"))
        s.create({"apiVersion": "v1", "kind": "ServiceAccount",
                  "metadata": {"name": "sa", "namespace": "health"}})
        s.create(hc("mine"))
        ev = await asyncio.wait_for(sub.__anext__(), 5)
        sub.close()
        return ev["object"]["metadata"]["name"]

    assert run(go()) == "mine"


def test_memory_client_facade(run):
    async def go():
        c = MemoryClient()
        await c.create(hc())
        got = await c.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
        got["status"] = {"status": "Succeeded"}
        await c.update_status(got)
        out = await c.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
        await c.delete("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
        return out["status"]["status"]

    assert run(go()) == "Succeeded"


def test_error_helpers():
    assert ignore_not_found(NotFoundError("x")) is None
    e = ConflictError("y")
    assert ignore_not_found(e) is e
    assert is_storage_error(RuntimeError("StorageError: invalid object in etcd"))
    assert not is_storage_error(RuntimeError("other"))


def test_update_without_resource_version_is_last_write_wins():
    """Updates omitting resourceVersion skip the conflict check (apiserver
    semantics: rv-less update = last write wins)."""
    s = MemoryApiServer()
    s.create(hc())
    a = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    b = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    a["spec"]["repeatAfterSec"] = 10
    s.update(a)
    b["spec"]["repeatAfterSec"] = 20
    del b["metadata"]["resourceVersion"]
    s.update(b)  # no conflict despite being stale
    final = s.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health", "check-1")
    assert final["spec"]["repeatAfterSec"] == 20


def test_field_selector_filtering(run):
    """Equality fieldSelectors (the kubectl-describe Event filter shape)."""
    from active_monitor_amd.kube.memory import MemoryApiServer

    s = MemoryApiServer()
    for i, phase in enumerate(["Succeeded", "Failed", "Succeeded"]):
        s.create({
            "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
            "metadata": {"name": f"fs-{i}", "namespace": "health"},
            "spec": {}, "status": {"phase": phase},
        })
    got = s.list("argoproj.io/v1alpha1", "Workflow", "health",
                 field_selector="status.phase=Succeeded")
    assert {o["metadata"]["name"] for o in got} == {"fs-0", "fs-2"}
    got = s.list("argoproj.io/v1alpha1", "Workflow", "health",
                 field_selector="status.phase!=Succeeded")
    assert {o["metadata"]["name"] for o in got} == {"fs-1"}
    got = s.list("argoproj.io/v1alpha1", "Workflow", "health",
                 field_selector="metadata.name=fs-1,status.phase=Failed")
    assert len(got) == 1

    # the Event-by-involvedObject shape kubectl describe uses
    s.create({
        "apiVersion": "v1", "kind": "Event",
        "metadata": {"name": "e1", "namespace": "health"},
        "involvedObject": {"name": "fs-1", "kind": "Workflow"},
        "reason": "R",
    })
    got = s.list("v1", "Event", "health",
                 field_selector="involvedObject.name=fs-1")
    assert len(got) == 1 and got[0]["metadata"]["name"] == "e1"
    got = s.list("v1", "Event", "health",
                 field_selector="involvedObject.name=other")
    assert got == []
