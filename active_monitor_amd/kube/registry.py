"""Resource registry: kind ↔ (group/version, plural, scope).

The equivalent of scheme registration (reference:
api/v1alpha1/groupversion_info.go:26-35) plus the Argo GVK/GVR constants
(healthcheck_controller.go:72-82).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Tuple

from .. import API_VERSION

WF_API_VERSION = "argoproj.io/v1alpha1"
WF_KIND = "Workflow"
WF_PLURAL = "workflows"


@dataclass(frozen=True)
class ResourceInfo:
    api_version: str
    kind: str
    plural: str
    namespaced: bool = True


_BUILTINS = [
    ResourceInfo(API_VERSION, "HealthCheck", "healthchecks", True),
    ResourceInfo(WF_API_VERSION, WF_KIND, WF_PLURAL, True),
    ResourceInfo("v1", "ServiceAccount", "serviceaccounts", True),
    ResourceInfo("v1", "Event", "events", True),
    ResourceInfo("v1", "Namespace", "namespaces", False),
    ResourceInfo("v1", "Pod", "pods", True),
    ResourceInfo("v1", "ConfigMap", "configmaps", True),
    ResourceInfo("v1", "Secret", "secrets", True),
    ResourceInfo("rbac.authorization.k8s.io/v1", "Role", "roles", True),
    ResourceInfo("rbac.authorization.k8s.io/v1", "RoleBinding", "rolebindings", True),
    ResourceInfo("rbac.authorization.k8s.io/v1", "ClusterRole", "clusterroles", False),
    ResourceInfo(
        "rbac.authorization.k8s.io/v1", "ClusterRoleBinding", "clusterrolebindings", False
    ),
    ResourceInfo("coordination.k8s.io/v1", "Lease", "leases", True),
]


class Registry:
    def __init__(self):
        self._by_kind: Dict[Tuple[str, str], ResourceInfo] = {}
        self._by_plural: Dict[Tuple[str, str], ResourceInfo] = {}
        for info in _BUILTINS:
            self.register(info)

    def register(self, info: ResourceInfo) -> None:
        self._by_kind[(info.api_version, info.kind)] = info
        self._by_plural[(info.api_version, info.plural)] = info

    def by_kind(self, api_version: str, kind: str) -> ResourceInfo:
        info = self._by_kind.get((api_version, kind))
        if info is None:
            # Unknown kinds are treated as namespaced custom resources with a
            # best-effort plural, so the store stays generic.
            info = ResourceInfo(api_version, kind, kind.lower() + "s", True)
            self.register(info)
        return info

    def by_plural(self, api_version: str, plural: str) -> ResourceInfo:
        info = self._by_plural.get((api_version, plural))
        if info is None:
            raise KeyError(f"unknown resource {plural} in {api_version}")
        return info


DEFAULT_REGISTRY = Registry()

# Kinds whose /status is a distinct subresource: a plain update cannot change
# status, and update_status cannot change spec/metadata (matching the
# reference's CRD, which enables the status subresource —
# healthcheck_types.go:69).
STATUS_SUBRESOURCE_KINDS = {(API_VERSION, "HealthCheck")}
