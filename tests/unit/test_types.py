"""API type tests — JSON wire-format parity with the reference CRD
(reference: api/v1alpha1/healthcheck_types.go, healthcheck_types_unit_test.go)."""
import json

from active_monitor_amd.api import (
    ArtifactLocation,
    HealthCheck,
    HealthCheckSpec,
    HealthCheckStatus,
    ObjectMeta,
    PolicyRule,
    RemedyWorkflow,
    ResourceObject,
    ScheduleSpec,
    URLArtifact,
    Workflow,
    parse_k8s_time,
)


def make_spec_dict():
    return {
        "repeatAfterSec": 60,
        "description": "test check",
        "level": "cluster",
        "workflow": {
            "generateName": "test-wf-",
            "workflowtimeout": 120,
            "resource": {
                "namespace": "health",
                "serviceAccount": "activemonitor-controller-sa",
                "source": {"inline": "apiVersion: argoproj.io/v1alpha1\nkind: Workflow\n"},
            },
        },
        "schedule": {"cron": "@every 1m"},
        "remedyworkflow": {
            "generateName": "remedy-wf-",
            "resource": {
                "namespace": "health",
                "serviceAccount": "remedy-sa",
                "source": {"url": {"path": "https://example.com/wf.yaml", "verifyCert": False}},
            },
        },
        "backoffFactor": "0.7",
        "backoffMax": 30,
        "backoffMin": 2,
        "remedyRunsLimit": 3,
        "remedyResetInterval": 300,
    }


def test_spec_round_trip_preserves_wire_names():
    d = make_spec_dict()
    spec = HealthCheckSpec.from_dict(d)
    assert spec.repeat_after_sec == 60
    assert spec.workflow.timeout == 120  # json tag "workflowtimeout"
    assert spec.remedy_workflow.resource.source.url.verify_cert is False
    back = spec.to_dict()
    assert back == d


def test_status_wire_names_include_remedy_triggered_at_quirk():
    st = HealthCheckStatus(
        status="Succeeded",
        success_count=3,
        remedy_started_at="2026-01-02T03:04:05Z",
        total_healthcheck_runs=5,
    )
    d = st.to_dict()
    # the historical tag mismatch (healthcheck_types.go:53)
    assert d["remedyTriggeredAt"] == "2026-01-02T03:04:05Z"
    assert "remedyStartedAt" not in d
    assert d["totalHealthCheckRuns"] == 5
    assert HealthCheckStatus.from_dict(d) == st


def test_status_omitempty():
    assert HealthCheckStatus().to_dict() == {}
    d = HealthCheckStatus(success_count=0, failed_count=2).to_dict()
    assert "successCount" not in d and d["failedCount"] == 2


def test_status_all_18_fields_round_trip():
    full = HealthCheckStatus(
        error_message="e", remedy_error_message="re",
        started_at="2026-01-01T00:00:00Z", finished_at="2026-01-01T00:00:01Z",
        last_failed_at="2026-01-01T00:00:02Z", remedy_started_at="2026-01-01T00:00:03Z",
        remedy_finished_at="2026-01-01T00:00:04Z", remedy_last_failed_at="2026-01-01T00:00:05Z",
        last_failed_workflow="wf-f", last_successful_workflow="wf-s",
        success_count=1, failed_count=2, remedy_success_count=3, remedy_failed_count=4,
        remedy_total_runs=5, total_healthcheck_runs=6, status="Failed", remedy_status="Succeeded",
    )
    d = full.to_dict()
    assert len(d) == 18
    assert set(d) == {
        "errorMessage", "remedyErrorMessage", "startedAt", "finishedAt", "lastFailedAt",
        "remedyTriggeredAt", "remedyFinishedAt", "remedyLastFailedAt", "lastFailedWorkflow",
        "lastSuccessfulWorkflow", "successCount", "failedCount", "remedySuccessCount",
        "remedyFailedCount", "remedyTotalRuns", "totalHealthCheckRuns", "status", "remedyStatus",
    }
    assert HealthCheckStatus.from_dict(json.loads(json.dumps(d))) == full


def test_remedy_is_empty_semantics():
    # reference: any single set field makes the remedy non-empty
    # (healthcheck_types.go:104-106, healthcheck_types_unit_test.go:24-38)
    assert RemedyWorkflow().is_empty()
    assert not RemedyWorkflow(timeout=1).is_empty()
    assert not RemedyWorkflow(generate_name="x").is_empty()
    assert not RemedyWorkflow(resource=ResourceObject()).is_empty()
    assert not RemedyWorkflow(rbac_rules=[PolicyRule(verbs=["get"])]).is_empty()


def test_url_artifact_secure_by_default():
    assert URLArtifact(path="x").should_verify
    assert URLArtifact(path="x", verify_cert=True).should_verify
    assert not URLArtifact(path="x", verify_cert=False).should_verify


def test_schedule_go_string():
    assert ScheduleSpec(cron="@every 1m").go_string() == "{Cron:@every 1m}"
    assert ScheduleSpec().go_string() == "{Cron:}"


def test_healthcheck_full_object_round_trip():
    hc = HealthCheck(
        metadata=ObjectMeta(name="hello", namespace="health", labels={"a": "b"}),
        spec=HealthCheckSpec.from_dict(make_spec_dict()),
        status=HealthCheckStatus(status="Succeeded", success_count=1),
    )
    d = hc.to_dict()
    assert d["apiVersion"] == "activemonitor.keikoproj.io/v1alpha1"
    assert d["kind"] == "HealthCheck"
    hc2 = HealthCheck.from_dict(json.loads(json.dumps(d)))
    assert hc2.to_dict() == d


def test_policy_rule_wire_format():
    r = PolicyRule(api_groups=[""], resources=["pods"], verbs=["get", "list"])
    assert r.to_dict() == {"verbs": ["get", "list"], "apiGroups": [""], "resources": ["pods"]}
    assert PolicyRule.from_dict(r.to_dict()) == r


def test_parse_k8s_time():
    t = parse_k8s_time("2026-01-02T03:04:05Z")
    assert t.year == 2026 and t.second == 5
    assert parse_k8s_time(None) is None
    assert parse_k8s_time("") is None


def test_reset_remedy_zeroes_all_remedy_fields():
    st = HealthCheckStatus(
        remedy_success_count=2, remedy_failed_count=1, remedy_total_runs=3,
        remedy_started_at="2026-01-01T00:00:00Z", remedy_finished_at="2026-01-01T00:00:01Z",
        remedy_last_failed_at="2026-01-01T00:00:02Z", remedy_error_message="x",
        remedy_status="Failed",
    )
    st.reset_remedy()
    d = st.to_dict()
    assert set(d) == {"remedyStatus"}  # caller overwrites remedy_status afterwards
