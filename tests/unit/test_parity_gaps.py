"""Small parity scenarios mirrored from the reference's unit tests."""
import asyncio

import pytest

from active_monitor_amd.api import (
    ArtifactLocation,
    HealthCheck,
    HealthCheckSpec,
    ObjectMeta,
    ResourceObject,
    Workflow,
)
from active_monitor_amd.engine.reconciler import HealthCheckReconciler
from active_monitor_amd.kube import FakeRecorder, MemoryApiServer, MemoryClient
from active_monitor_amd.kube.errors import contains_equal_fold_substring
from active_monitor_amd.store.artifacts import ArtifactReadError


def test_contains_equal_fold_substring_matrix():
    # reference healthcheck_controller_unit_test.go:57-84
    assert contains_equal_fold_substring("StorageError: invalid object", "storageerror")
    assert contains_equal_fold_substring("ABCdef", "aBcD")
    assert not contains_equal_fold_substring("hello", "world")
    assert contains_equal_fold_substring("anything", "")   # empty substr matches
    assert contains_equal_fold_substring("", "")


def test_parse_unknown_artifact_errors():
    # reference :116-130 — source with no inline/url/file
    hc = HealthCheck(
        metadata=ObjectMeta(name="x", namespace="health"),
        spec=HealthCheckSpec(
            repeat_after_sec=30,
            workflow=Workflow(
                generate_name="w-",
                resource=ResourceObject(namespace="health", service_account="sa",
                                        source=ArtifactLocation()),
            ),
        ),
    )
    from active_monitor_amd.engine.parse import parse_workflow_from_healthcheck

    with pytest.raises(ArtifactReadError, match="unknown artifact location"):
        parse_workflow_from_healthcheck(hc)


def test_non_string_status_message_is_safe(run):
    """Workflow message of a non-string type must not corrupt status
    (reference TestSafeStatusMessageExtraction :260-306)."""

    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        rec = HealthCheckReconciler(client, FakeRecorder())
        hc_obj = {
            "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
            "kind": "HealthCheck",
            "metadata": {"name": "m", "namespace": "health"},
            "spec": {"repeatAfterSec": 3600, "level": "cluster", "workflow": {
                "generateName": "m-wf-", "workflowtimeout": 5,
                "resource": {"namespace": "health", "serviceAccount": "sa",
                             "source": {"inline": "spec:\n  entrypoint: e\n"}},
            }},
        }
        await client.create(hc_obj)
        hc = HealthCheck.from_dict(
            await client.get("activemonitor.keikoproj.io/v1alpha1", "HealthCheck",
                             "health", "m"))
        # a workflow already failed with a non-string message
        await client.create({
            "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
            "metadata": {"name": "m-wf-1", "namespace": "health"},
            "spec": {},
            "status": {"phase": "Failed", "message": {"not": "a string"}},
        })
        await rec.watch_workflow_reschedule("health", "m-wf-1", hc)
        assert hc.status.status == "Failed"
        assert hc.status.error_message == ""  # safe extraction, no crash
        rec.stop_all()

    run(go())


def test_custom_metric_without_value_defaults_zero():
    # reference TestCollectNoValueMetric :49-58
    import json

    from prometheus_client import CollectorRegistry

    from active_monitor_amd.metrics import create_dynamic_prometheus_metric

    reg = CollectorRegistry()
    payload = json.dumps({"metrics": [{"name": "novalue", "metrictype": "gauge"}]})
    updated = create_dynamic_prometheus_metric(
        "hcx", {"outputs": {"parameters": [{"name": "m", "value": payload}]}}, reg
    )
    assert updated == ["hcx_novalue"]
    assert reg.get_sample_value("hcx_novalue", {"healthcheck_name": "hcx"}) == 0.0
