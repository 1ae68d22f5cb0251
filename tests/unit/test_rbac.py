"""RBAC provisioner unit tests — the reference's fake-clientset tier
(healthcheck_controller_unit_test.go:310-502)."""
import pytest

from active_monitor_amd.api import (
    ArtifactLocation,
    HealthCheck,
    HealthCheckSpec,
    ObjectMeta,
    PolicyRule,
    RemedyWorkflow,
    ResourceObject,
    Workflow,
)
from active_monitor_amd.engine.rbac import (
    DEFAULT_HEALTHCHECK_RULES,
    DEFAULT_REMEDY_RULES,
    RBACProvisioner,
    resolve_rbac_rules,
)
from active_monitor_amd.kube import MemoryApiServer, MemoryClient, NotFoundError

RBAC = "rbac.authorization.k8s.io/v1"


def make_hc(level="cluster", sa="sa-x", remedy_sa=None, hc_rules=None, remedy_rules=None):
    spec = HealthCheckSpec(
        repeat_after_sec=30,
        level=level,
        workflow=Workflow(
            generate_name="w-",
            rbac_rules=hc_rules or [],
            resource=ResourceObject(namespace="health", service_account=sa,
                                    source=ArtifactLocation(inline="spec: {}")),
        ),
    )
    if remedy_sa is not None:
        spec.remedy_workflow = RemedyWorkflow(
            generate_name="r-",
            rbac_rules=remedy_rules or [],
            resource=ResourceObject(namespace="health", service_account=remedy_sa,
                                    source=ArtifactLocation(inline="spec: {}")),
        )
    return HealthCheck(metadata=ObjectMeta(name="x", namespace="health"), spec=spec)


@pytest.fixture
def client():
    return MemoryClient(MemoryApiServer())


def test_default_rule_scoping():
    """Health defaults read-only, remedy defaults CRUD, no wildcards
    (reference :85-120, unit test :310-457)."""
    for rule in DEFAULT_HEALTHCHECK_RULES:
        assert set(rule.verbs) == {"get", "list", "watch"}
        assert "*" not in rule.verbs and "*" not in rule.resources
    for rule in DEFAULT_REMEDY_RULES:
        assert set(rule.verbs) == {"get", "list", "watch", "create", "update",
                                   "patch", "delete"}
        assert "*" not in rule.verbs and "*" not in rule.resources
    # remedy defaults must not touch nodes/namespaces (reference :104-110)
    core = next(r for r in DEFAULT_REMEDY_RULES if r.api_groups == [""])
    assert "nodes" not in core.resources and "namespaces" not in core.resources


def test_resolve_rules_prefers_custom():
    custom = [PolicyRule(api_groups=[""], resources=["secrets"], verbs=["get"])]
    assert resolve_rbac_rules(custom, DEFAULT_HEALTHCHECK_RULES) is custom
    assert resolve_rbac_rules([], DEFAULT_HEALTHCHECK_RULES) is DEFAULT_HEALTHCHECK_RULES


def test_cluster_level_objects_and_names(client, run):
    async def go():
        prov = RBACProvisioner(client)
        await prov.create_rbac_for_workflow(make_hc(), "healthCheck")
        sa = await client.get("v1", "ServiceAccount", "health", "sa-x")
        cr = await client.get(RBAC, "ClusterRole", "", "sa-x-cluster-role")
        crb = await client.get(RBAC, "ClusterRoleBinding", "", "sa-x-cluster-role-binding")
        assert crb["roleRef"]["name"] == "sa-x-cluster-role"
        assert crb["subjects"] == [
            {"kind": "ServiceAccount", "name": "sa-x", "namespace": "health"}
        ]
        for obj in (sa, cr, crb):
            assert obj["metadata"]["labels"]["workflows.argoproj.io/managed-by"] == "active-monitor"

    run(go())


def test_create_is_idempotent_and_preserves_existing(client, run):
    """Get-then-create: an existing object is reused, never updated
    (reference behavior — no update of drifted rules)."""

    async def go():
        await client.create({
            "apiVersion": RBAC, "kind": "ClusterRole",
            "metadata": {"name": "sa-x-cluster-role"},  # pre-existing, unlabeled
            "rules": [{"verbs": ["*"]}],
        })
        prov = RBACProvisioner(client)
        await prov.create_rbac_for_workflow(make_hc(), "healthCheck")
        cr = await client.get(RBAC, "ClusterRole", "", "sa-x-cluster-role")
        assert cr["rules"] == [{"verbs": ["*"]}]  # untouched

    run(go())


def test_delete_guarded_by_managed_by_label(client, run):
    """Only objects labeled managed-by active-monitor are ever deleted
    (reference :1169,1242; unit test :366-407)."""

    async def go():
        await client.create({
            "apiVersion": "v1", "kind": "ServiceAccount",
            "metadata": {"name": "rsa", "namespace": "health"},  # foreign SA
        })
        prov = RBACProvisioner(client)
        hc = make_hc(remedy_sa="rsa")
        await prov.delete_rbac_for_workflow(hc)
        # survived the teardown
        await client.get("v1", "ServiceAccount", "health", "rsa")

    run(go())


def test_remedy_cycle_creates_then_deletes(client, run):
    async def go():
        prov = RBACProvisioner(client)
        hc = make_hc(remedy_sa="rsa")
        await prov.create_rbac_for_workflow(hc, "remedy")
        await client.get("v1", "ServiceAccount", "health", "rsa")
        await client.get(RBAC, "ClusterRole", "", "rsa-cluster-role")
        await prov.delete_rbac_for_workflow(hc)
        with pytest.raises(NotFoundError):
            await client.get("v1", "ServiceAccount", "health", "rsa")
        with pytest.raises(NotFoundError):
            await client.get(RBAC, "ClusterRole", "", "rsa-cluster-role")
        with pytest.raises(NotFoundError):
            await client.get(RBAC, "ClusterRoleBinding", "", "rsa-cluster-role-binding")

    run(go())


def test_sa_collision_renames_remedy(client, run):
    """Remedy SA == check SA → '<sa>-remedy' (reference :316-319)."""

    async def go():
        prov = RBACProvisioner(client)
        hc = make_hc(sa="shared", remedy_sa="shared")
        await prov.create_rbac_for_workflow(hc, "remedy")
        assert hc.spec.remedy_workflow.resource.service_account == "shared-remedy"
        await client.get("v1", "ServiceAccount", "health", "shared-remedy")

    run(go())


def test_namespace_level_uses_roles(client, run):
    async def go():
        prov = RBACProvisioner(client)
        await prov.create_rbac_for_workflow(make_hc(level="namespace"), "healthCheck")
        await client.get(RBAC, "Role", "health", "sa-x-ns-role")
        rb = await client.get(RBAC, "RoleBinding", "health", "sa-x-ns-role-binding")
        assert rb["roleRef"]["kind"] == "Role"
        with pytest.raises(NotFoundError):
            await client.get(RBAC, "ClusterRole", "", "sa-x-cluster-role")

    run(go())


def test_level_unset_errors(client, run):
    async def go():
        prov = RBACProvisioner(client)
        with pytest.raises(ValueError, match="level is not set"):
            await prov.create_rbac_for_workflow(make_hc(level=""), "healthCheck")

    run(go())


def test_remedy_validation_errors(client, run):
    async def go():
        prov = RBACProvisioner(client)
        hc = make_hc()
        hc.spec.remedy_workflow = RemedyWorkflow(generate_name="r-")  # no resource
        with pytest.raises(ValueError, match="RemedyWorkflow is set but Resource is nil"):
            await prov.create_rbac_for_workflow(hc, "healthCheck")
        hc2 = make_hc(remedy_sa="")
        with pytest.raises(ValueError, match="ServiceAccount for the RemedyWorkflow"):
            await prov.create_rbac_for_workflow(hc2, "healthCheck")

    run(go())


def test_custom_rules_applied(client, run):
    async def go():
        rules = [PolicyRule(api_groups=[""], resources=["secrets"], verbs=["get"])]
        prov = RBACProvisioner(client)
        await prov.create_rbac_for_workflow(make_hc(hc_rules=rules), "healthCheck")
        cr = await client.get(RBAC, "ClusterRole", "", "sa-x-cluster-role")
        assert cr["rules"] == [{"verbs": ["get"], "apiGroups": [""], "resources": ["secrets"]}]

    run(go())
