"""RBAC provisioning for health-check and remedy workflows.

Re-implements the reference's SA/Role/ClusterRole/Binding lifecycle
(healthcheck_controller.go:302-474, 1127-1443):

- default least-privilege rule sets: read-only for health checks, scoped CRUD
  for remedies (:85-120), overridable per-CR via ``spec.*.rbacRules`` (:124-129),
- derived names ``<sa>-cluster-role``, ``<sa>-cluster-role-binding``,
  ``<sa>-ns-role``, ``<sa>-ns-role-binding`` (:306-309,321-324),
- get-then-create idempotency (existing objects are reused, never updated),
- every created object labeled ``workflows.argoproj.io/managed-by:
  active-monitor``; deletes only touch objects carrying that label
  (:1169,1242,1317,1375,1433),
- remedy SA collision rename ``<sa>-remedy`` when it matches the health-check
  SA (:316-319).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..api.types import HealthCheck, PolicyRule
from ..kube.client import EventRecorder, KubeClient
from ..kube.errors import AlreadyExistsError, NotFoundError

RBAC_API_VERSION = "rbac.authorization.k8s.io/v1"

WF_MANAGED_BY_LABEL_KEY = "workflows.argoproj.io/managed-by"
WF_MANAGED_BY_VALUE = "active-monitor"

CLUSTER_LEVEL = "cluster"
NAMESPACE_LEVEL = "namespace"

# Least-privilege read-only rules for monitoring workflows
# (healthcheck_controller.go:85-101).
DEFAULT_HEALTHCHECK_RULES: List[PolicyRule] = [
    PolicyRule(
        api_groups=[""],
        resources=["pods", "nodes", "events", "services", "configmaps", "namespaces", "endpoints"],
        verbs=["get", "list", "watch"],
    ),
    PolicyRule(
        api_groups=["apps"],
        resources=["deployments", "replicasets", "statefulsets", "daemonsets"],
        verbs=["get", "list", "watch"],
    ),
    PolicyRule(
        api_groups=["argoproj.io"],
        resources=["workflows"],
        verbs=["get", "list", "watch"],
    ),
]

# Scoped write rules for remedy workflows (healthcheck_controller.go:104-120).
DEFAULT_REMEDY_RULES: List[PolicyRule] = [
    PolicyRule(
        api_groups=[""],
        resources=["pods", "events", "services", "configmaps", "endpoints"],
        verbs=["get", "list", "watch", "create", "update", "patch", "delete"],
    ),
    PolicyRule(
        api_groups=["apps"],
        resources=["deployments", "replicasets", "statefulsets"],
        verbs=["get", "list", "watch", "create", "update", "patch", "delete"],
    ),
    PolicyRule(
        api_groups=["argoproj.io"],
        resources=["workflows"],
        verbs=["get", "list", "watch", "create", "update", "patch", "delete"],
    ),
]


def resolve_rbac_rules(
    custom: List[PolicyRule], defaults: List[PolicyRule]
) -> List[PolicyRule]:
    """Custom rules win when non-empty (healthcheck_controller.go:124-129)."""
    return custom if custom else defaults


def _managed_labels() -> Dict[str, str]:
    return {WF_MANAGED_BY_LABEL_KEY: WF_MANAGED_BY_VALUE}


def _is_managed(obj: Dict[str, Any]) -> bool:
    labels = (obj.get("metadata") or {}).get("labels") or {}
    return labels.get(WF_MANAGED_BY_LABEL_KEY) == WF_MANAGED_BY_VALUE


class RBACProvisioner:
    """Creates and deletes the workflow RBAC objects through a KubeClient
    (the reference threads a typed clientset through each helper so unit
    tests can pass a fake — here the client itself is injectable)."""

    #: how long a successful ensure suppresses re-checking the same object.
    #: The reference re-gets every RBAC object on every cycle (:333-408); with
    #: a fleet at 1 Hz that is ~4 redundant apiserver round-trips per cycle.
    #: External deletions are still repaired within this window; 0 disables.
    ENSURE_TTL = 30.0

    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None,
                 ensure_ttl: Optional[float] = None):
        import os

        self.client = client
        self.recorder = recorder
        if ensure_ttl is None:
            env = os.environ.get("AM_RBAC_ENSURE_TTL")
            ensure_ttl = float(env) if env else self.ENSURE_TTL
        self.ensure_ttl = ensure_ttl
        self._ensured: Dict[tuple, float] = {}

    def _fresh(self, key: tuple) -> bool:
        import time

        if self.ensure_ttl <= 0:
            return False
        exp = self._ensured.get(key)
        return exp is not None and exp > time.monotonic()

    def _mark(self, key: tuple) -> None:
        import time

        if self.ensure_ttl > 0:
            if len(self._ensured) > 50000:
                now = time.monotonic()
                self._ensured = {k: v for k, v in self._ensured.items() if v > now}
            self._ensured[key] = time.monotonic() + self.ensure_ttl

    def _invalidate(self, kind: str, namespace: str, name: str) -> None:
        self._ensured.pop((kind, namespace, name), None)

    async def _event(self, hc_obj: Dict[str, Any], ev_type: str, message: str) -> None:
        if self.recorder is not None:
            await self.recorder.event(hc_obj, ev_type, ev_type, message)

    # -- creates (get-then-create, reuse if present) -----------------------

    async def create_service_account(self, name: str, namespace: str) -> str:
        cache_key = ("ServiceAccount", namespace, name)
        if self._fresh(cache_key):
            return name
        try:
            sa = await self.client.get("v1", "ServiceAccount", namespace, name)
            self._mark(cache_key)
            return sa["metadata"]["name"]
        except NotFoundError:
            pass
        sa = {
            "apiVersion": "v1",
            "kind": "ServiceAccount",
            "metadata": {"name": name, "namespace": namespace, "labels": _managed_labels()},
        }
        try:
            created = await self.client.create(sa, transfer=True)
        except AlreadyExistsError:
            # concurrent reconciles sharing an SA race get-then-create;
            # losing the race means the object exists — reuse it
            self._mark(cache_key)
            return name
        self._mark(cache_key)
        return created["metadata"]["name"]

    async def create_cluster_role(self, name: str, rules: List[PolicyRule]) -> str:
        cache_key = ("ClusterRole", "", name)
        if self._fresh(cache_key):
            return name
        try:
            cr = await self.client.get(RBAC_API_VERSION, "ClusterRole", "", name)
            self._mark(cache_key)
            return cr["metadata"]["name"]
        except NotFoundError:
            pass
        cr = {
            "apiVersion": RBAC_API_VERSION,
            "kind": "ClusterRole",
            "metadata": {"name": name, "labels": _managed_labels()},
            "rules": [r.to_dict() for r in rules],
        }
        try:
            created = await self.client.create(cr, transfer=True)
        except AlreadyExistsError:
            # concurrent reconciles sharing an SA race get-then-create;
            # losing the race means the object exists — reuse it
            self._mark(cache_key)
            return name
        self._mark(cache_key)
        return created["metadata"]["name"]

    async def create_cluster_role_binding(
        self, name: str, role_name: str, sa_name: str, sa_namespace: str
    ) -> str:
        cache_key = ("ClusterRoleBinding", "", name)
        if self._fresh(cache_key):
            return name
        try:
            crb = await self.client.get(RBAC_API_VERSION, "ClusterRoleBinding", "", name)
            self._mark(cache_key)
            return crb["metadata"]["name"]
        except NotFoundError:
            pass
        crb = {
            "apiVersion": RBAC_API_VERSION,
            "kind": "ClusterRoleBinding",
            "metadata": {"name": name, "labels": _managed_labels()},
            "roleRef": {
                "apiGroup": "rbac.authorization.k8s.io",
                "kind": "ClusterRole",
                "name": role_name,
            },
            "subjects": [
                {"kind": "ServiceAccount", "name": sa_name, "namespace": sa_namespace}
            ],
        }
        try:
            created = await self.client.create(crb, transfer=True)
        except AlreadyExistsError:
            # concurrent reconciles sharing an SA race get-then-create;
            # losing the race means the object exists — reuse it
            self._mark(cache_key)
            return name
        self._mark(cache_key)
        return created["metadata"]["name"]

    async def create_namespace_role(
        self, name: str, namespace: str, rules: List[PolicyRule]
    ) -> str:
        cache_key = ("Role", namespace, name)
        if self._fresh(cache_key):
            return name
        try:
            role = await self.client.get(RBAC_API_VERSION, "Role", namespace, name)
            self._mark(cache_key)
            return role["metadata"]["name"]
        except NotFoundError:
            pass
        role = {
            "apiVersion": RBAC_API_VERSION,
            "kind": "Role",
            "metadata": {"name": name, "namespace": namespace, "labels": _managed_labels()},
            "rules": [r.to_dict() for r in rules],
        }
        try:
            created = await self.client.create(role, transfer=True)
        except AlreadyExistsError:
            # concurrent reconciles sharing an SA race get-then-create;
            # losing the race means the object exists — reuse it
            self._mark(cache_key)
            return name
        self._mark(cache_key)
        return created["metadata"]["name"]

    async def create_namespace_role_binding(
        self, name: str, role_name: str, sa_name: str, namespace: str
    ) -> str:
        cache_key = ("RoleBinding", namespace, name)
        if self._fresh(cache_key):
            return name
        try:
            rb = await self.client.get(RBAC_API_VERSION, "RoleBinding", namespace, name)
            self._mark(cache_key)
            return rb["metadata"]["name"]
        except NotFoundError:
            pass
        rb = {
            "apiVersion": RBAC_API_VERSION,
            "kind": "RoleBinding",
            "metadata": {"name": name, "namespace": namespace, "labels": _managed_labels()},
            "roleRef": {
                "apiGroup": "rbac.authorization.k8s.io",
                "kind": "Role",
                "name": role_name,
            },
            "subjects": [
                {"kind": "ServiceAccount", "name": sa_name, "namespace": namespace}
            ],
        }
        try:
            created = await self.client.create(rb, transfer=True)
        except AlreadyExistsError:
            # concurrent reconciles sharing an SA race get-then-create;
            # losing the race means the object exists — reuse it
            self._mark(cache_key)
            return name
        self._mark(cache_key)
        return created["metadata"]["name"]

    # -- deletes (only objects carrying the managed-by label) --------------

    async def _delete_if_managed(self, api_version: str, kind: str, namespace: str, name: str) -> None:
        # Already-gone objects count as deleted. (The reference propagates the
        # Get error here (:1164-1166), which makes concurrent remedy teardowns
        # of a shared SA fail each other — an availability fix, not a
        # semantic change.)
        try:
            obj = await self.client.get(api_version, kind, namespace, name)
        except NotFoundError:
            return
        if _is_managed(obj):
            try:
                await self.client.delete(api_version, kind, namespace, name)
            except NotFoundError:
                return

    async def delete_service_account(self, name: str, namespace: str) -> None:
        self._invalidate("ServiceAccount", namespace, name)
        await self._delete_if_managed("v1", "ServiceAccount", namespace, name)

    async def delete_cluster_role(self, name: str) -> None:
        self._invalidate("ClusterRole", "", name)
        await self._delete_if_managed(RBAC_API_VERSION, "ClusterRole", "", name)

    async def delete_cluster_role_binding(self, name: str) -> None:
        self._invalidate("ClusterRoleBinding", "", name)
        await self._delete_if_managed(RBAC_API_VERSION, "ClusterRoleBinding", "", name)

    async def delete_namespace_role(self, name: str, namespace: str) -> None:
        self._invalidate("Role", namespace, name)
        await self._delete_if_managed(RBAC_API_VERSION, "Role", namespace, name)

    async def delete_namespace_role_binding(self, name: str, namespace: str) -> None:
        self._invalidate("RoleBinding", namespace, name)
        await self._delete_if_managed(RBAC_API_VERSION, "RoleBinding", namespace, name)

    # -- orchestration ------------------------------------------------------

    async def create_rbac_for_workflow(self, hc: HealthCheck, workflow_type: str) -> None:
        """``workflow_type`` ∈ {"healthCheck", "remedy"}
        (healthcheck_controller.go:302-415). May rename the remedy SA on
        collision — a spec mutation visible for the rest of the reconcile,
        as in the reference (:316-319)."""
        hc_obj = hc.to_dict()
        level = hc.spec.level
        hc_sa = hc.spec.workflow.resource.service_account
        wf_namespace = hc.spec.workflow.resource.namespace

        remedy_sa = ""
        wf_remedy_namespace = ""
        if not hc.spec.remedy_workflow.is_empty():
            if hc.spec.remedy_workflow.resource is None:
                await self._event(hc.to_dict(), "Warning",
                                  "RemedyWorkflow is set but Resource is nil")
                raise ValueError("RemedyWorkflow is set but Resource is nil")
            if hc.spec.remedy_workflow.resource.service_account == "":
                await self._event(
                    hc.to_dict(), "Warning",
                    "ServiceAccount for the RemedyWorkflow is not specified"
                )
                raise ValueError("ServiceAccount for the RemedyWorkflow is not specified")
            if hc_sa == hc.spec.remedy_workflow.resource.service_account:
                hc.spec.remedy_workflow.resource.service_account = hc_sa + "-remedy"
            remedy_sa = hc.spec.remedy_workflow.resource.service_account
            wf_remedy_namespace = hc.spec.remedy_workflow.resource.namespace

        is_remedy = workflow_type == "remedy"
        if is_remedy:
            await self.create_service_account(remedy_sa, wf_remedy_namespace)
        else:
            await self.create_service_account(hc_sa, wf_namespace)

        hc_rules = resolve_rbac_rules(hc.spec.workflow.rbac_rules, DEFAULT_HEALTHCHECK_RULES)
        remedy_rules = resolve_rbac_rules(hc.spec.remedy_workflow.rbac_rules, DEFAULT_REMEDY_RULES)

        if level == CLUSTER_LEVEL:
            if not is_remedy:
                await self.create_cluster_role(hc_sa + "-cluster-role", hc_rules)
                await self.create_cluster_role_binding(
                    hc_sa + "-cluster-role-binding", hc_sa + "-cluster-role", hc_sa, wf_namespace
                )
            else:
                await self.create_cluster_role(remedy_sa + "-cluster-role", remedy_rules)
                await self.create_cluster_role_binding(
                    remedy_sa + "-cluster-role-binding",
                    remedy_sa + "-cluster-role",
                    remedy_sa,
                    wf_remedy_namespace,
                )
        elif level == NAMESPACE_LEVEL:
            if not is_remedy:
                await self.create_namespace_role(hc_sa + "-ns-role", wf_namespace, hc_rules)
                await self.create_namespace_role_binding(
                    hc_sa + "-ns-role-binding", hc_sa + "-ns-role", hc_sa, wf_namespace
                )
            else:
                await self.create_namespace_role(
                    remedy_sa + "-ns-role", wf_remedy_namespace, remedy_rules
                )
                await self.create_namespace_role_binding(
                    remedy_sa + "-ns-role-binding",
                    remedy_sa + "-ns-role",
                    remedy_sa,
                    wf_remedy_namespace,
                )
        else:
            await self._event(hc.to_dict(), "Warning", "level is not set")
            raise ValueError("level is not set")

    async def delete_rbac_for_workflow(self, hc: HealthCheck) -> None:
        """Tear down the remedy RBAC after a remedy run
        (healthcheck_controller.go:417-474). Only remedy objects are deleted,
        and only when labeled managed-by active-monitor."""
        if hc.spec.remedy_workflow.resource is None:
            return
        level = hc.spec.level
        remedy_sa = hc.spec.remedy_workflow.resource.service_account
        ns = hc.spec.remedy_workflow.resource.namespace
        await self.delete_service_account(remedy_sa, ns)
        if level == CLUSTER_LEVEL:
            await self.delete_cluster_role(remedy_sa + "-cluster-role")
            await self.delete_cluster_role_binding(remedy_sa + "-cluster-role-binding")
        elif level == NAMESPACE_LEVEL:
            await self.delete_namespace_role(remedy_sa + "-ns-role", ns)
            await self.delete_namespace_role_binding(remedy_sa + "-ns-role-binding", ns)
        else:
            await self._event(hc.to_dict(), "Warning", "level is not set")
            raise ValueError("level is not set")
