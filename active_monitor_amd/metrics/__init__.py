"""Prometheus metrics with the reference's exact series names and labels."""
from .collector import (
    REGISTRY,
    MonitorError,
    MonitorFinishedTime,
    MonitorRuntime,
    MonitorStartedTime,
    MonitorSuccess,
    create_dynamic_prometheus_metric,
    custom_gauge_metrics,
    exposition,
)

__all__ = [
    "REGISTRY",
    "MonitorError",
    "MonitorFinishedTime",
    "MonitorRuntime",
    "MonitorStartedTime",
    "MonitorSuccess",
    "create_dynamic_prometheus_metric",
    "custom_gauge_metrics",
    "exposition",
]
