"""HealthCheck API types.

Byte-compatible with the reference CRD's spec/status JSON
(reference: api/v1alpha1/healthcheck_types.go:32-151). Every field keeps the
exact JSON tag of the reference, including the quirky ones:

- ``HealthCheckStatus.RemedyStartedAt`` serializes as ``remedyTriggeredAt``
  (healthcheck_types.go:53),
- ``Workflow.Timeout`` / ``RemedyWorkflow.Timeout`` serialize as
  ``workflowtimeout`` (healthcheck_types.go:100,112).

Objects round-trip through plain dicts (the "unstructured" form) so they can be
stored in any Kubernetes-shaped backend. ``omitempty`` semantics are honored:
zero values are dropped on serialization, matching Go's encoding/json.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Dict, List, Optional

from .. import API_VERSION

# ---------------------------------------------------------------------------
# Kubernetes time helpers (metav1.Time equivalent: RFC3339, second precision)
# ---------------------------------------------------------------------------


_NOW_CACHE = [0, ""]


def k8s_now() -> str:
    """Current UTC time in Kubernetes metav1.Time wire format (second
    precision, so the formatted string is cached per second)."""
    import time as _time

    sec = int(_time.time())
    if _NOW_CACHE[0] != sec:
        _NOW_CACHE[0] = sec
        _NOW_CACHE[1] = format_k8s_time(datetime.fromtimestamp(sec, timezone.utc))
    return _NOW_CACHE[1]


def format_k8s_time(dt: datetime) -> str:
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=timezone.utc)
    return dt.astimezone(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


_PARSE_TIME_CACHE: Dict[str, datetime] = {}


def parse_k8s_time(s: Optional[str]) -> Optional[datetime]:
    if not s:
        return None
    cached = _PARSE_TIME_CACHE.get(s)
    if cached is None:
        cached = datetime.strptime(s, "%Y-%m-%dT%H:%M:%SZ").replace(tzinfo=timezone.utc)
        if len(_PARSE_TIME_CACHE) > 4096:
            _PARSE_TIME_CACHE.clear()
        _PARSE_TIME_CACHE[s] = cached
    return cached


# ---------------------------------------------------------------------------
# RBAC policy rule (rbacv1.PolicyRule equivalent)
# ---------------------------------------------------------------------------


@dataclass
class PolicyRule:
    """A single RBAC rule (k8s.io/api/rbac/v1 PolicyRule shape)."""

    api_groups: List[str] = field(default_factory=list)
    resources: List[str] = field(default_factory=list)
    verbs: List[str] = field(default_factory=list)
    resource_names: List[str] = field(default_factory=list)
    non_resource_urls: List[str] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"verbs": list(self.verbs)}
        if self.api_groups:
            d["apiGroups"] = list(self.api_groups)
        if self.resources:
            d["resources"] = list(self.resources)
        if self.resource_names:
            d["resourceNames"] = list(self.resource_names)
        if self.non_resource_urls:
            d["nonResourceURLs"] = list(self.non_resource_urls)
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "PolicyRule":
        return cls(
            api_groups=list(d.get("apiGroups", []) or []),
            resources=list(d.get("resources", []) or []),
            verbs=list(d.get("verbs", []) or []),
            resource_names=list(d.get("resourceNames", []) or []),
            non_resource_urls=list(d.get("nonResourceURLs", []) or []),
        )


# ---------------------------------------------------------------------------
# Artifact sources (reference: healthcheck_types.go:127-145)
# ---------------------------------------------------------------------------


@dataclass
class FileArtifact:
    """Filesystem artifact source. Declared in the reference API but left
    unimplemented by its store (store.go:15-22); we keep the type for schema
    parity and implement it for real (see store/)."""

    path: str = ""

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.path:
            d["path"] = self.path
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "FileArtifact":
        return cls(path=d.get("path", "") or "")


@dataclass
class URLArtifact:
    """HTTP(S) artifact source. ``verify_cert=None`` (omitted) or True means TLS
    certificates ARE verified — secure by default (healthcheck_types.go:139-145).
    """

    path: str = ""
    verify_cert: Optional[bool] = None

    @property
    def should_verify(self) -> bool:
        return self.verify_cert is not False

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.path:
            d["path"] = self.path
        if self.verify_cert is not None:
            d["verifyCert"] = self.verify_cert
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "URLArtifact":
        return cls(path=d.get("path", "") or "", verify_cert=d.get("verifyCert"))


@dataclass
class ArtifactLocation:
    """Source location for a workflow definition (healthcheck_types.go:127-131)."""

    inline: Optional[str] = None
    file: Optional[FileArtifact] = None
    url: Optional[URLArtifact] = None

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.inline is not None:
            d["inline"] = self.inline
        if self.file is not None:
            d["file"] = self.file.to_dict()
        if self.url is not None:
            d["url"] = self.url.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ArtifactLocation":
        return cls(
            inline=d.get("inline"),
            file=FileArtifact.from_dict(d["file"]) if isinstance(d.get("file"), dict) else None,
            url=URLArtifact.from_dict(d["url"]) if isinstance(d.get("url"), dict) else None,
        )


@dataclass
class ResourceObject:
    """The resource object to create on Kubernetes (healthcheck_types.go:117-124)."""

    namespace: str = ""
    service_account: str = ""
    source: ArtifactLocation = field(default_factory=ArtifactLocation)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"namespace": self.namespace, "source": self.source.to_dict()}
        if self.service_account:
            d["serviceAccount"] = self.service_account
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ResourceObject":
        return cls(
            namespace=d.get("namespace", "") or "",
            service_account=d.get("serviceAccount", "") or "",
            source=ArtifactLocation.from_dict(d.get("source", {}) or {}),
        )


# ---------------------------------------------------------------------------
# Workflow / RemedyWorkflow (healthcheck_types.go:97-114)
# ---------------------------------------------------------------------------


def _workflow_to_dict(w: "Workflow") -> Dict[str, Any]:
    d: Dict[str, Any] = {}
    if w.generate_name:
        d["generateName"] = w.generate_name
    if w.resource is not None:
        d["resource"] = w.resource.to_dict()
    if w.timeout:
        d["workflowtimeout"] = w.timeout
    if w.rbac_rules:
        d["rbacRules"] = [r.to_dict() for r in w.rbac_rules]
    return d


@dataclass
class Workflow:
    """Describes the Argo (health-check) workflow (healthcheck_types.go:109-114).

    ``timeout`` carries the quirky JSON tag ``workflowtimeout``."""

    generate_name: str = ""
    resource: Optional[ResourceObject] = None
    timeout: int = 0
    rbac_rules: List[PolicyRule] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return _workflow_to_dict(self)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Workflow":
        return cls(
            generate_name=d.get("generateName", "") or "",
            resource=ResourceObject.from_dict(d["resource"]) if isinstance(d.get("resource"), dict) else None,
            timeout=int(d.get("workflowtimeout", 0) or 0),
            rbac_rules=[PolicyRule.from_dict(r) for r in d.get("rbacRules", []) or []],
        )


@dataclass
class RemedyWorkflow(Workflow):
    """Describes the remedy workflow (healthcheck_types.go:97-106)."""

    def is_empty(self) -> bool:
        """True when every field is at its zero value — any single set field
        (even ``timeout``) makes the remedy non-empty
        (healthcheck_types.go:104-106)."""
        return (
            not self.generate_name
            and self.resource is None
            and self.timeout == 0
            and not self.rbac_rules
        )


@dataclass
class ScheduleSpec:
    """Cron schedule (healthcheck_types.go:148-151), robfig/cron standard syntax
    including descriptors such as ``@every 1m``."""

    cron: str = ""

    def to_dict(self) -> Dict[str, Any]:
        return {"cron": self.cron} if self.cron else {}

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ScheduleSpec":
        return cls(cron=d.get("cron", "") or "")

    def go_string(self) -> str:
        """Render as Go's ``%+v`` of the struct, used verbatim in the Stopped
        error message (healthcheck_controller.go:241)."""
        return "{Cron:%s}" % self.cron


# ---------------------------------------------------------------------------
# Spec / Status (healthcheck_types.go:32-66)
# ---------------------------------------------------------------------------


@dataclass
class HealthCheckSpec:
    """Desired state. Either ``repeat_after_sec`` or ``schedule`` must be set
    for the check to run (healthcheck_types.go:30-44)."""

    repeat_after_sec: int = 0
    description: str = ""
    workflow: Workflow = field(default_factory=Workflow)
    level: str = ""  # "cluster" or "namespace"
    schedule: ScheduleSpec = field(default_factory=ScheduleSpec)
    remedy_workflow: RemedyWorkflow = field(default_factory=RemedyWorkflow)
    backoff_factor: str = ""  # string in the reference API
    backoff_max: int = 0
    backoff_min: int = 0
    remedy_runs_limit: int = 0
    remedy_reset_interval: int = 0

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"workflow": self.workflow.to_dict()}
        if self.repeat_after_sec:
            d["repeatAfterSec"] = self.repeat_after_sec
        if self.description:
            d["description"] = self.description
        if self.level:
            d["level"] = self.level
        sched = self.schedule.to_dict()
        if sched:
            d["schedule"] = sched
        if not self.remedy_workflow.is_empty():
            d["remedyworkflow"] = self.remedy_workflow.to_dict()
        if self.backoff_factor:
            d["backoffFactor"] = self.backoff_factor
        if self.backoff_max:
            d["backoffMax"] = self.backoff_max
        if self.backoff_min:
            d["backoffMin"] = self.backoff_min
        if self.remedy_runs_limit:
            d["remedyRunsLimit"] = self.remedy_runs_limit
        if self.remedy_reset_interval:
            d["remedyResetInterval"] = self.remedy_reset_interval
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "HealthCheckSpec":
        rw = d.get("remedyworkflow")
        return cls(
            repeat_after_sec=int(d.get("repeatAfterSec", 0) or 0),
            description=d.get("description", "") or "",
            workflow=Workflow.from_dict(d.get("workflow", {}) or {}),
            level=d.get("level", "") or "",
            schedule=ScheduleSpec.from_dict(d.get("schedule", {}) or {}),
            remedy_workflow=(
                RemedyWorkflow.from_dict(rw) if isinstance(rw, dict) else RemedyWorkflow()
            ),
            backoff_factor=str(d.get("backoffFactor", "") or ""),
            backoff_max=int(d.get("backoffMax", 0) or 0),
            backoff_min=int(d.get("backoffMin", 0) or 0),
            remedy_runs_limit=int(d.get("remedyRunsLimit", 0) or 0),
            remedy_reset_interval=int(d.get("remedyResetInterval", 0) or 0),
        )


# (python_name, json_name, kind) — kind: "str" | "int" | "time"
_STATUS_FIELDS = [
    ("error_message", "errorMessage", "str"),
    ("remedy_error_message", "remedyErrorMessage", "str"),
    ("started_at", "startedAt", "time"),
    ("finished_at", "finishedAt", "time"),
    ("last_failed_at", "lastFailedAt", "time"),
    # NOTE the historical tag mismatch, kept for wire compatibility:
    ("remedy_started_at", "remedyTriggeredAt", "time"),
    ("remedy_finished_at", "remedyFinishedAt", "time"),
    ("remedy_last_failed_at", "remedyLastFailedAt", "time"),
    ("last_failed_workflow", "lastFailedWorkflow", "str"),
    ("last_successful_workflow", "lastSuccessfulWorkflow", "str"),
    ("success_count", "successCount", "int"),
    ("failed_count", "failedCount", "int"),
    ("remedy_success_count", "remedySuccessCount", "int"),
    ("remedy_failed_count", "remedyFailedCount", "int"),
    ("remedy_total_runs", "remedyTotalRuns", "int"),
    ("total_healthcheck_runs", "totalHealthCheckRuns", "int"),
    ("status", "status", "str"),
    ("remedy_status", "remedyStatus", "str"),
]


@dataclass
class HealthCheckStatus:
    """Observed state — the durable checkpoint (healthcheck_types.go:47-66).

    The 18 fields and their JSON names (including ``remedyTriggeredAt`` for
    ``remedy_started_at``) are pinned by BASELINE.json's "status/checkpoint
    format" requirement. Time fields hold the metav1.Time wire string."""

    error_message: str = ""
    remedy_error_message: str = ""
    started_at: Optional[str] = None
    finished_at: Optional[str] = None
    last_failed_at: Optional[str] = None
    remedy_started_at: Optional[str] = None
    remedy_finished_at: Optional[str] = None
    remedy_last_failed_at: Optional[str] = None
    last_failed_workflow: str = ""
    last_successful_workflow: str = ""
    success_count: int = 0
    failed_count: int = 0
    remedy_success_count: int = 0
    remedy_failed_count: int = 0
    remedy_total_runs: int = 0
    total_healthcheck_runs: int = 0
    status: str = ""
    remedy_status: str = ""

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        for py, js, kind in _STATUS_FIELDS:
            v = getattr(self, py)
            if kind == "int":
                if v:
                    d[js] = v
            elif v:  # str / time: omit empty/None
                d[js] = v
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "HealthCheckStatus":
        kw: Dict[str, Any] = {}
        for py, js, kind in _STATUS_FIELDS:
            if js in d and d[js] is not None:
                kw[py] = int(d[js]) if kind == "int" else d[js]
        return cls(**kw)

    def reset_remedy(self) -> None:
        """Zero every Remedy* counter/timestamp (the reset sets used by both the
        pass-reset and interval-reset paths, healthcheck_controller.go:650-660,
        695-703). The caller sets ``remedy_status`` to the applicable string."""
        self.remedy_success_count = 0
        self.remedy_failed_count = 0
        self.remedy_total_runs = 0
        self.remedy_started_at = None
        self.remedy_finished_at = None
        self.remedy_last_failed_at = None
        self.remedy_error_message = ""


# ---------------------------------------------------------------------------
# ObjectMeta + HealthCheck
# ---------------------------------------------------------------------------


@dataclass
class ObjectMeta:
    """The subset of metav1.ObjectMeta the controller uses."""

    name: str = ""
    generate_name: str = ""
    namespace: str = ""
    uid: str = ""
    resource_version: str = ""
    generation: int = 0
    creation_timestamp: Optional[str] = None
    deletion_timestamp: Optional[str] = None
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    owner_references: List[Dict[str, Any]] = field(default_factory=list)
    finalizers: List[str] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.name:
            d["name"] = self.name
        if self.generate_name:
            d["generateName"] = self.generate_name
        if self.namespace:
            d["namespace"] = self.namespace
        if self.uid:
            d["uid"] = self.uid
        if self.resource_version:
            d["resourceVersion"] = self.resource_version
        if self.generation:
            d["generation"] = self.generation
        if self.creation_timestamp:
            d["creationTimestamp"] = self.creation_timestamp
        if self.deletion_timestamp:
            d["deletionTimestamp"] = self.deletion_timestamp
        if self.labels:
            d["labels"] = dict(self.labels)
        if self.annotations:
            d["annotations"] = dict(self.annotations)
        if self.owner_references:
            d["ownerReferences"] = copy.deepcopy(self.owner_references)
        if self.finalizers:
            d["finalizers"] = list(self.finalizers)
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ObjectMeta":
        return cls(
            name=d.get("name", "") or "",
            generate_name=d.get("generateName", "") or "",
            namespace=d.get("namespace", "") or "",
            uid=d.get("uid", "") or "",
            resource_version=str(d.get("resourceVersion", "") or ""),
            generation=int(d.get("generation", 0) or 0),
            creation_timestamp=d.get("creationTimestamp"),
            deletion_timestamp=d.get("deletionTimestamp"),
            labels=dict(d.get("labels", {}) or {}),
            annotations=dict(d.get("annotations", {}) or {}),
            owner_references=copy.deepcopy(d.get("ownerReferences", []) or []),
            finalizers=list(d.get("finalizers", []) or []),
        )


@dataclass
class HealthCheck:
    """The HealthCheck custom resource (healthcheck_types.go:78-85)."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: HealthCheckSpec = field(default_factory=HealthCheckSpec)
    status: HealthCheckStatus = field(default_factory=HealthCheckStatus)

    KIND = "HealthCheck"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "apiVersion": API_VERSION,
            "kind": self.KIND,
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
        }
        status = self.status.to_dict()
        if status:
            d["status"] = status
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "HealthCheck":
        return cls(
            metadata=ObjectMeta.from_dict(d.get("metadata", {}) or {}),
            spec=HealthCheckSpec.from_dict(d.get("spec", {}) or {}),
            status=HealthCheckStatus.from_dict(d.get("status", {}) or {}),
        )
