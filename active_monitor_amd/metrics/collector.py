"""Prometheus collectors (reference: internal/metrics/collector.go).

Exact series-name parity with the reference's 5 static collectors:

- ``healthcheck_success_count``   (counter)  :18-24
- ``healthcheck_error_count``     (counter)  :25-30
- ``healthcheck_runtime_seconds`` (gauge)    :31-36
- ``healthcheck_starttime``       (gauge)    :37-42 (Unix-seconds epoch despite
  the "Time taken" help string — a reference quirk kept for scrape parity)
- ``healthcheck_finishedtime``    (gauge)    :43-48

all labeled ``{healthcheck_name, workflow}`` with workflow ∈ {"healthCheck",
"remedy"} (collector.go:14-15; controller constants
healthcheck_controller.go:60-61).

The Go prometheus client exposes counters under the exact configured name,
while prometheus_client appends ``_total``; to keep the wire format identical a
small custom collector is used for the two counters.

Custom metrics from workflow output parameters
(``workflow.status.outputs.parameters`` JSON of shape
``{"metrics": [{name, value, metrictype, help}]}``, README.md:275-285) are
registered dynamically by :func:`create_dynamic_prometheus_metric`
(collector.go:68-115) — and, unlike the reference (which documents the feature
but never calls it — SURVEY.md §2.3, quirk list), the reconciler here invokes
it on the workflow-success path.
"""
from __future__ import annotations

import json
import logging
import threading
from typing import Any, Dict, Iterable, List, Optional, Tuple

from prometheus_client import CollectorRegistry, Gauge, generate_latest
from prometheus_client.core import Metric

log = logging.getLogger("active_monitor_amd.metrics")

HC_NAME_LABEL = "healthcheck_name"
WF_LABEL = "workflow"

# The framework's own registry — the equivalent of controller-runtime's
# metrics.Registry (collector.go:62-65); served by the Manager's metrics
# endpoint.
REGISTRY = CollectorRegistry()


class _ExactNameCounter:
    """CounterVec exposed under its exact name (no ``_total`` suffix), matching
    the Go client's exposition of the reference series."""

    def __init__(self, name: str, help_text: str, labelnames: Tuple[str, ...]):
        self._name = name
        self._help = help_text
        self._labelnames = labelnames
        self._values: Dict[Tuple[str, ...], float] = {}
        self._lock = threading.Lock()

    def labels(self, *labelvalues: str) -> "_ExactNameCounterChild":
        if len(labelvalues) != len(self._labelnames):
            raise ValueError(f"{self._name}: expected {len(self._labelnames)} labels")
        return _ExactNameCounterChild(self, tuple(labelvalues))

    def _inc(self, key: Tuple[str, ...], amount: float) -> None:
        if amount < 0:
            raise ValueError("counters can only increase")
        with self._lock:
            self._values[key] = self._values.get(key, 0.0) + amount

    def value(self, *labelvalues: str) -> float:
        with self._lock:
            return self._values.get(tuple(labelvalues), 0.0)

    def collect(self) -> Iterable[Metric]:
        # An 'untyped' family keeps both the family name and sample names
        # exactly as configured (prometheus_client would append '_total' to a
        # counter-typed family, diverging from the Go client's exposition).
        fam = Metric(self._name, self._help, "untyped")
        with self._lock:
            for key, val in sorted(self._values.items()):
                fam.add_sample(self._name, dict(zip(self._labelnames, key)), val)
        return [fam]

    def describe(self) -> Iterable[Metric]:
        return [Metric(self._name, self._help, "untyped")]


class _ExactNameCounterChild:
    def __init__(self, parent: _ExactNameCounter, key: Tuple[str, ...]):
        self._parent = parent
        self._key = key

    def inc(self, amount: float = 1.0) -> None:
        self._parent._inc(self._key, amount)


_LABELS = (HC_NAME_LABEL, WF_LABEL)

MonitorSuccess = _ExactNameCounter(
    "healthcheck_success_count", "The total number of successful healthcheck resources", _LABELS
)
MonitorError = _ExactNameCounter(
    "healthcheck_error_count", "The total number of errored healthcheck resources", _LABELS
)
MonitorRuntime = Gauge(
    "healthcheck_runtime_seconds",
    "Time taken for the workflow to complete.",
    _LABELS,
    registry=REGISTRY,
)
MonitorStartedTime = Gauge(
    "healthcheck_starttime",
    "Time taken for the workflow to complete.",
    _LABELS,
    registry=REGISTRY,
)
MonitorFinishedTime = Gauge(
    "healthcheck_finishedtime",
    "Time taken for the workflow to complete.",
    _LABELS,
    registry=REGISTRY,
)

REGISTRY.register(MonitorSuccess)  # type: ignore[arg-type]
REGISTRY.register(MonitorError)  # type: ignore[arg-type]

# Dynamic custom-metric gauges, guarded by an RW-ish lock (the reference fixed
# a data race here with an RWMutex — collector.go:50-51,92-109, issue #288).
custom_gauge_metrics: Dict[str, Gauge] = {}
_custom_lock = threading.Lock()


def create_dynamic_prometheus_metric(
    name: str,
    workflow_status: Optional[Dict[str, Any]],
    registry: CollectorRegistry = REGISTRY,
) -> List[str]:
    """Parse ``workflow.status.outputs.parameters`` into dynamically registered
    gauges named ``<hc_name_with_underscores>_<metric name>`` labeled
    ``{healthcheck_name}`` (collector.go:68-115).

    ``workflow_status`` is the unstructured ``status`` dict of a Workflow.
    Returns the list of metric names updated (for tests/observability).
    """
    updated: List[str] = []
    if not workflow_status:
        return updated
    outputs = workflow_status.get("outputs")
    if not isinstance(outputs, dict):
        return updated
    parameters = outputs.get("parameters")
    if not isinstance(parameters, list):
        return updated

    for parameter in parameters:
        if not isinstance(parameter, dict):
            continue
        raw = parameter.get("value")
        if not isinstance(raw, str):
            continue
        try:
            json_map = json.loads(raw)
        except (ValueError, TypeError):
            # The reference ignores unmarshal errors (collector.go:76) and
            # iterates an empty map; mirror that leniency.
            continue
        if not isinstance(json_map, dict):
            continue
        for metric_raw in json_map.get("metrics", []) or []:
            if not isinstance(metric_raw, dict):
                log.error("Failed to decode metric for %s: %r", name, metric_raw)
                continue
            # mapstructure.Decode is case-insensitive on keys (collector.go:81).
            lowered = {str(k).lower(): v for k, v in metric_raw.items()}
            metric_name = str(lowered.get("name", "") or "")
            if metric_name == "":
                log.error("Skipping metric collection. Invalid metric %s: %r", name, metric_raw)
                continue
            try:
                value = float(lowered.get("value", 0) or 0)
            except (TypeError, ValueError):
                log.error("Failed to decode metric for %s: %r", name, metric_raw)
                continue
            help_text = str(lowered.get("help", "") or "")
            # '-' → '_' for Prometheus-friendly names (collector.go:90).
            full_name = name.replace("-", "_") + "_" + metric_name
            with _custom_lock:
                gauge = custom_gauge_metrics.get(full_name)
                if gauge is None:
                    try:
                        gauge = Gauge(full_name, help_text, (HC_NAME_LABEL,), registry=registry)
                        custom_gauge_metrics[full_name] = gauge
                    except ValueError as e:
                        log.error("Error registering %s metric %s", full_name, e)
                        continue
                gauge.labels(name).set(value)
            updated.append(full_name)
            log.info("Successfully collected metric for %s, metric: %s=%s", name, full_name, value)
    return updated


def exposition(registry: CollectorRegistry = REGISTRY) -> bytes:
    """Text exposition of the registry (for the manager's /metrics endpoint)."""
    return generate_latest(registry)
