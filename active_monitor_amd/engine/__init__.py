"""Controller runtime: workqueue, manager, reconciler, cron, backoff, RBAC."""
from .backoff import IEBTimeoutError, InverseExponentialBackoff, compute_backoff_params
from .cronx import CronParseError, parse_go_duration, parse_standard, seconds_until_next
from .manager import Manager
from .parse import (
    WorkflowParseError,
    parse_remedy_workflow_from_healthcheck,
    parse_remedy_workflow_from_healthcheck_async,
    parse_workflow_from_healthcheck,
    parse_workflow_from_healthcheck_async,
)
from .rbac import (
    DEFAULT_HEALTHCHECK_RULES,
    DEFAULT_REMEDY_RULES,
    RBACProvisioner,
    resolve_rbac_rules,
)
from .reconciler import HealthCheckReconciler, ReconcileResult
from .workqueue import RateLimiter, WorkQueue

__all__ = [
    "CronParseError",
    "DEFAULT_HEALTHCHECK_RULES",
    "DEFAULT_REMEDY_RULES",
    "HealthCheckReconciler",
    "IEBTimeoutError",
    "InverseExponentialBackoff",
    "Manager",
    "RateLimiter",
    "RBACProvisioner",
    "ReconcileResult",
    "WorkQueue",
    "WorkflowParseError",
    "compute_backoff_params",
    "parse_go_duration",
    "parse_remedy_workflow_from_healthcheck",
    "parse_remedy_workflow_from_healthcheck_async",
    "parse_standard",
    "parse_workflow_from_healthcheck",
    "parse_workflow_from_healthcheck_async",
    "resolve_rbac_rules",
    "seconds_until_next",
]
