"""Controller integration tests — the envtest/BDD tier
(reference: internal/controllers/healthcheck_controller_test.go,
healthcheck_controller_edge_test.go)."""
import asyncio

import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.workflow import (
    ScriptedWorkflowEngine,
    always_fail,
    always_succeed,
    never_complete,
)

from .conftest import Env, INLINE_WF_WITH_LABELS, make_hc

RBAC = "rbac.authorization.k8s.io/v1"


def test_success_cycle_end_to_end(run):
    """The 'one model running end-to-end' milestone: CR → RBAC → submit →
    watch → Succeeded status → re-armed timer."""

    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(make_hc(repeat=1, timeout=2))

            async def succeeded():
                hc = await env.get_hc("hello-check")
                return hc.status.success_count >= 1 and hc
            hc = await env.wait_for(succeeded, msg="first successful run")

            assert hc.status.status == "Succeeded"
            assert hc.status.last_successful_workflow.startswith("hello-check-wf-")
            assert hc.status.total_healthcheck_runs >= 1
            assert hc.status.started_at and hc.status.finished_at

            # RBAC provisioned with managed-by labels (cluster level)
            sa = await env.client.get("v1", "ServiceAccount", "health", "check-sa")
            assert sa["metadata"]["labels"]["workflows.argoproj.io/managed-by"] == "active-monitor"
            cr = await env.client.get(RBAC, "ClusterRole", "", "check-sa-cluster-role")
            verbs = {v for rule in cr["rules"] for v in rule["verbs"]}
            assert verbs == {"get", "list", "watch"}  # read-only defaults
            await env.client.get(RBAC, "ClusterRoleBinding", "", "check-sa-cluster-role-binding")

            # the submitted workflow carries ownerRef + injections
            wfs = await env.workflows()
            wf = wfs[0]
            assert wf["metadata"]["ownerReferences"][0]["kind"] == "HealthCheck"
            assert wf["metadata"]["ownerReferences"][0]["controller"] is True
            assert (
                wf["metadata"]["labels"]["workflows.argoproj.io/controller-instanceid"]
                == "activemonitor-workflows"
            )
            assert wf["spec"]["podGC"] == {"strategy": "OnPodCompletion"}
            assert wf["spec"]["serviceAccountName"] == "check-sa"
            assert wf["spec"]["activeDeadlineSeconds"] == 2

            # repeat timer armed for the next run
            assert env.manager.reconciler.get_timer_by_name("hello-check") is not None

    run(go(), timeout=30)


def test_repeat_executes_multiple_runs(run):
    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(make_hc(name="rep", repeat=1, timeout=2))

            async def three_runs():
                hc = await env.get_hc("rep")
                return hc.status.total_healthcheck_runs >= 3 and hc

            hc = await env.wait_for(three_runs, timeout=25, msg=">=3 runs")
            assert hc.status.success_count >= 3

    run(go(), timeout=40)


def test_no_engine_timeout_forces_failed(run):
    """The reference's envtest linchpin: without a workflow controller the IEB
    timeout synthesizes a Failed phase (healthcheck_controller.go:627-632)."""

    async def go():
        async with Env(engine=False) as env:
            await env.create_hc(make_hc(name="lonely", repeat=2, timeout=1))

            async def failed():
                hc = await env.get_hc("lonely")
                return hc.status.failed_count >= 1 and hc

            hc = await env.wait_for(failed, msg="timeout-forced failure")
            assert hc.status.status == "Failed"
            assert hc.status.error_message == "Failed"  # synthesized message
            assert hc.status.last_failed_workflow.startswith("lonely-wf-")
            assert hc.status.last_failed_at

    run(go(), timeout=30)


def test_pause_via_repeat_after_sec_zero(run):
    """repeatAfterSec<=0 with no cron → Stopped with the reference's exact
    message (healthcheck_controller.go:238-250)."""

    async def go():
        async with Env() as env:
            await env.create_hc(make_hc(name="paused", repeat=0, timeout=0))

            async def stopped():
                hc = await env.get_hc("paused")
                return hc.status.status == "Stopped" and hc

            hc = await env.wait_for(stopped, msg="Stopped status")
            assert hc.status.error_message == (
                "workflow execution is stopped; either spec.RepeatAfterSec or "
                "spec.Schedule must be provided. spec.RepeatAfterSec set to 0. "
                "spec.Schedule set to {Cron:}"
            )
            assert hc.status.finished_at
            assert (await env.workflows()) == []  # nothing submitted

    run(go(), timeout=30)


def test_cron_schedule_runs(run):
    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(make_hc(name="cronny", repeat=0, cron="@every 1s", timeout=2))

            async def ran_twice():
                hc = await env.get_hc("cronny")
                return hc.status.success_count >= 2 and hc

            hc = await env.wait_for(ran_twice, timeout=25, msg="two cron runs")
            assert hc.status.status == "Succeeded"

    run(go(), timeout=40)


def test_invalid_cron_no_crash(run):
    """Invalid cron must not panic the controller; it errors + requeues
    (reference edge test :107-150)."""

    async def go():
        async with Env() as env:
            await env.create_hc(make_hc(name="badcron", repeat=0, cron="not a cron", timeout=1))
            await asyncio.sleep(0.5)
            hc = await env.get_hc("badcron")
            assert hc.status.success_count == 0
            assert (await env.workflows()) == []
            # manager still alive and reconciling others
            await env.create_hc(make_hc(name="ok", repeat=1, timeout=2))

            async def ok_ran():
                hc2 = await env.get_hc("ok")
                return hc2.status.success_count >= 1

            await env.wait_for(ok_ran, msg="healthy CR still reconciled")

    run(go(), timeout=30)


def test_delete_stops_rescheduling(run):
    """Deleting the CR stops the repeat timer and cascade-GCs workflows
    (reference edge test :152-198, controller :180-184)."""

    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(make_hc(name="doomed", repeat=1, timeout=2))

            async def ran():
                hc = await env.get_hc("doomed")
                return hc.status.success_count >= 1

            await env.wait_for(ran, msg="first run")
            await env.client.delete(API_VERSION, "HealthCheck", "health", "doomed")

            async def timer_gone():
                return env.manager.reconciler.get_timer_by_name("doomed") is None

            await env.wait_for(timer_gone, msg="timer cancelled")
            # workflows GC'd via ownerReference cascade
            assert all(
                not w["metadata"]["name"].startswith("doomed-")
                for w in await env.workflows()
            )
            # and no new submissions appear afterwards
            await asyncio.sleep(1.5)
            assert all(
                not w["metadata"]["name"].startswith("doomed-")
                for w in await env.workflows()
            )

    run(go(), timeout=40)


def test_nil_workflow_resource_is_noop(run):
    """Workflow.Resource unset → reconcile is a no-op (reference :227,
    edge test :47-74)."""

    async def go():
        async with Env() as env:
            await env.create_hc({
                "apiVersion": API_VERSION,
                "kind": "HealthCheck",
                "metadata": {"name": "no-resource", "namespace": "health"},
                "spec": {"repeatAfterSec": 1, "workflow": {"generateName": "x-"}},
            })
            await asyncio.sleep(0.5)
            hc = await env.get_hc("no-resource")
            assert hc.status.to_dict() == {}
            assert (await env.workflows()) == []

    run(go(), timeout=30)


def test_empty_level_errors(run):
    """level unset → 'level is not set' error, no workflow submitted
    (reference :412, edge test :76-105)."""

    async def go():
        async with Env() as env:
            await env.create_hc(make_hc(name="no-level", level="", repeat=1, timeout=1))
            await asyncio.sleep(0.5)
            assert (await env.workflows()) == []
            evs = await env.client.list("v1", "Event", "health")
            assert any("level is not set" in e.get("message", "") for e in evs)

    run(go(), timeout=30)


def test_namespace_level_rbac(run):
    """namespace level provisions Role/RoleBinding, not ClusterRole
    (reference :383-408)."""

    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(make_hc(name="ns-check", level="namespace", repeat=1, timeout=2))

            async def ran():
                hc = await env.get_hc("ns-check")
                return hc.status.success_count >= 1

            await env.wait_for(ran, msg="namespace-level run")
            role = await env.client.get(RBAC, "Role", "health", "check-sa-ns-role")
            assert role["metadata"]["labels"]["workflows.argoproj.io/managed-by"] == "active-monitor"
            await env.client.get(RBAC, "RoleBinding", "health", "check-sa-ns-role-binding")
            with pytest.raises(Exception):
                await env.client.get(RBAC, "ClusterRole", "", "check-sa-cluster-role")

    run(go(), timeout=30)


def test_custom_rbac_rules_override(run):
    """spec.workflow.rbacRules overrides the default rule set
    (reference :124-129, unit test :411-445)."""

    async def go():
        async with Env(policy=always_succeed) as env:
            hc = make_hc(name="custom-rbac", repeat=1, timeout=2)
            hc["spec"]["workflow"]["rbacRules"] = [
                {"apiGroups": [""], "resources": ["secrets"], "verbs": ["get"]}
            ]
            await env.create_hc(hc)

            async def ran():
                h = await env.get_hc("custom-rbac")
                return h.status.success_count >= 1

            await env.wait_for(ran, msg="run with custom rules")
            cr = await env.client.get(RBAC, "ClusterRole", "", "check-sa-cluster-role")
            assert cr["rules"] == [
                {"verbs": ["get"], "apiGroups": [""], "resources": ["secrets"]}
            ]

    run(go(), timeout=30)


def test_workflow_own_labels_scoped_per_submission(run):
    """Labels from one CR's workflow must NOT leak onto other CRs' workflows
    (fixing the reference's shared-map leak), while the
    instance-id label is still guaranteed."""

    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(
                make_hc(name="labeled", repeat=1, timeout=2, inline=INLINE_WF_WITH_LABELS)
            )
            await env.create_hc(make_hc(name="plain", repeat=1, timeout=2))

            async def both_ran():
                a = await env.get_hc("labeled")
                b = await env.get_hc("plain")
                return a.status.success_count >= 1 and b.status.success_count >= 1

            await env.wait_for(both_ran, msg="both CRs ran")
            for wf in await env.workflows():
                labels = wf["metadata"]["labels"]
                assert (
                    labels["workflows.argoproj.io/controller-instanceid"]
                    == "activemonitor-workflows"
                )
                if wf["metadata"]["name"].startswith("plain-"):
                    assert "team" not in labels  # no leak
                if wf["metadata"]["name"].startswith("labeled-"):
                    assert labels["team"] == "sre" and labels["tier"] == "1"

    run(go(), timeout=30)


def test_custom_metrics_wired_on_success(run):
    """Workflow output parameters become Prometheus gauges on the success path
    (the feature the reference documents but never wires — README.md:275-285)."""
    import json

    from active_monitor_amd.metrics import REGISTRY

    def policy(wf):
        payload = json.dumps({"metrics": [{"name": "checked_pods", "value": 42,
                                           "metrictype": "gauge", "help": "pods checked"}]})
        return ("Succeeded", "", {"parameters": [{"name": "metrics", "value": payload}]})

    async def go():
        async with Env(policy=policy) as env:
            await env.create_hc(make_hc(name="metric-check", repeat=1, timeout=2))

            async def ran():
                hc = await env.get_hc("metric-check")
                return hc.status.success_count >= 1

            await env.wait_for(ran, msg="success with outputs")
            assert REGISTRY.get_sample_value(
                "metric_check_checked_pods", {"healthcheck_name": "metric-check"}
            ) == 42

    run(go(), timeout=30)


def test_same_name_crs_in_different_namespaces(run):
    """Same-name CRs in two namespaces run independently — the reference's
    name-only timer map makes them fight over one slot (its :139); fixed by
    keying timers and watches by (namespace, name)."""

    async def go():
        async with Env(policy=always_succeed) as env:
            for ns in ("health", "default"):
                await env.create_hc(make_hc(name="dup", ns=ns, repeat=1, timeout=2))

            async def both_repeat():
                a = await env.client.get(
                    "activemonitor.keikoproj.io/v1alpha1", "HealthCheck",
                    "health", "dup")
                b = await env.client.get(
                    "activemonitor.keikoproj.io/v1alpha1", "HealthCheck",
                    "default", "dup")
                return (
                    (a.get("status") or {}).get("successCount", 0) >= 2
                    and (b.get("status") or {}).get("successCount", 0) >= 2
                )

            deadline = asyncio.get_running_loop().time() + 25
            while asyncio.get_running_loop().time() < deadline:
                if await both_repeat():
                    break
                await asyncio.sleep(0.1)
            assert await both_repeat(), "one namespace starved the other"
            rec = env.manager.reconciler
            assert rec.get_timer_by_name("dup", "health") is not None
            assert rec.get_timer_by_name("dup", "default") is not None
            assert rec.get_timer_by_name("dup", "health") is not rec.get_timer_by_name("dup", "default")

    run(go(), timeout=45)


def test_pure_polling_without_watch_hub(run):
    """With the watch hub disabled the controller degrades to the reference's
    pure IEB polling and still completes cycles (higher detection latency,
    identical semantics)."""
    from active_monitor_amd.engine import Manager
    from active_monitor_amd.kube import MemoryApiServer, MemoryClient

    async def go():
        client = MemoryClient(MemoryApiServer())
        engine = ScriptedWorkflowEngine(client, policy=always_succeed)
        await engine.start()
        manager = Manager(client, max_workers=2, enable_wf_hub=False)
        await manager.start()
        assert manager.reconciler.wf_hub is None
        try:
            await client.create(make_hc(name="polled", repeat=1, timeout=4))
            deadline = asyncio.get_running_loop().time() + 25
            ok = False
            while asyncio.get_running_loop().time() < deadline:
                obj = await client.get(API_VERSION, "HealthCheck", "health", "polled")
                if (obj.get("status") or {}).get("successCount", 0) >= 2:
                    ok = True
                    break
                await asyncio.sleep(0.1)
            assert ok, "no cycles completed on the pure-polling path"
        finally:
            await manager.stop()
            await engine.stop()

    run(go(), timeout=45)


def test_repeat_takes_precedence_over_cron(run):
    """When both repeatAfterSec>0 and schedule.cron are set, the interval wins
    and the cron is ignored (the reference's branch order, :238-263)."""

    async def go():
        async with Env(policy=always_succeed) as env:
            # cron says every hour; repeat says every second — repeat must win
            await env.create_hc(make_hc(name="both", repeat=1, cron="0 * * * *",
                                        timeout=2))

            async def ran_twice():
                hc = await env.get_hc("both")
                return hc.status.success_count >= 2

            await env.wait_for(ran_twice, timeout=20, msg="interval-driven repeats")

    run(go(), timeout=40)


def test_pause_then_resume_lifecycle(run):
    """Active → paused (repeatAfterSec:0) → Stopped; then resumed → runs
    again. Covers live spec edits across schedule states."""

    async def go():
        async with Env(policy=always_succeed) as env:
            await env.create_hc(make_hc(name="toggle", repeat=1, timeout=2))

            async def ran():
                hc = await env.get_hc("toggle")
                return hc.status.success_count >= 1

            await env.wait_for(ran, msg="initial run")

            # pause it
            obj = await env.client.get(API_VERSION, "HealthCheck", "health", "toggle")
            obj["spec"].pop("repeatAfterSec", None)
            await env.client.update(obj)

            async def stopped():
                hc = await env.get_hc("toggle")
                return hc.status.status == "Stopped"

            await env.wait_for(stopped, timeout=15, msg="paused → Stopped")
            count_at_pause = (await env.get_hc("toggle")).status.success_count
            await asyncio.sleep(2.0)
            assert (await env.get_hc("toggle")).status.success_count == count_at_pause

            # resume it
            obj = await env.client.get(API_VERSION, "HealthCheck", "health", "toggle")
            obj["spec"]["repeatAfterSec"] = 1
            await env.client.update(obj)

            async def resumed():
                hc = await env.get_hc("toggle")
                return hc.status.success_count > count_at_pause

            await env.wait_for(resumed, timeout=15, msg="resumed and running")

    run(go(), timeout=60)
