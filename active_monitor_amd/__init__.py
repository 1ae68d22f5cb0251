"""active-monitor-amd: a clean-room, Python-native HealthCheck/Remedy controller
framework with the capabilities of keikoproj/active-monitor.

The reference (Go, controller-runtime) runs "active" cluster health checks: each
``HealthCheck`` custom resource wraps an Argo Workflow that is submitted
periodically (fixed interval or cron), watched to completion, and reflected into
the CR's status; on failure an optional RemedyWorkflow self-heals, governed by a
run-limit / reset-interval state machine (reference: README.md:12-28).

This framework re-implements that capability set from scratch:

- :mod:`active_monitor_amd.api`      — the HealthCheck API types with the exact
  spec/status JSON field names of the reference CRD
  (reference: api/v1alpha1/healthcheck_types.go).
- :mod:`active_monitor_amd.store`    — artifact readers (inline / URL)
  (reference: internal/store/).
- :mod:`active_monitor_amd.metrics`  — Prometheus collectors with identical
  series names and labels (reference: internal/metrics/collector.go).
- :mod:`active_monitor_amd.kube`     — the Kubernetes client abstraction: an
  in-memory apiserver (envtest equivalent) and an HTTP client for real clusters.
- :mod:`active_monitor_amd.engine`   — the controller runtime: workqueue,
  manager, reconciler, cron, inverse-exponential backoff
  (reference: internal/controllers/healthcheck_controller.go, cmd/main.go).
- :mod:`active_monitor_amd.workflow` — workflow execution backends: Argo CRs on
  a real cluster, plus a local in-process engine for standalone use and tests.

Unlike the reference, the watch loop never blocks a reconcile worker and repeat
timers feed back through the workqueue, so ``MaxConcurrentReconciles`` genuinely
bounds reconcile work ("the single biggest architectural
decision").
"""

__version__ = "0.2.0"

GROUP = "activemonitor.keikoproj.io"
VERSION = "v1alpha1"
API_VERSION = f"{GROUP}/{VERSION}"
