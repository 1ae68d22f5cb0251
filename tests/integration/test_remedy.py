"""Remedy state-machine integration tests
(reference: healthcheck_controller.go:649-660,677-721,759-874;
BDD specs healthcheck_controller_test.go:63-117)."""
import asyncio

import pytest

from active_monitor_amd import API_VERSION

from .conftest import Env, make_hc

RBAC = "rbac.authorization.k8s.io/v1"


def fail_check_pass_remedy(wf):
    """Health-check workflows fail; remedy workflows succeed."""
    name = wf["metadata"]["name"]
    if "-remedy-wf-" in name:
        return ("Succeeded", "")
    return ("Failed", "check failed")


def test_remedy_runs_on_failure_and_rbac_cycle(run):
    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            await env.create_hc(make_hc(name="rem", repeat=1, timeout=2, remedy=True))

            async def remedy_ran():
                hc = await env.get_hc("rem")
                return hc.status.remedy_total_runs >= 1 and hc

            hc = await env.wait_for(remedy_ran, msg="remedy run")
            assert hc.status.status == "Failed"
            assert hc.status.error_message == "check failed"
            assert hc.status.remedy_status == "Succeeded"
            assert hc.status.remedy_success_count >= 1
            assert hc.status.remedy_started_at and hc.status.remedy_finished_at
            # wire format: remedyTriggeredAt tag
            raw = await env.client.get(API_VERSION, "HealthCheck", "health", "rem")
            assert "remedyTriggeredAt" in raw["status"]

            # remedy RBAC was torn down after the run (create→use→delete,
            # reference :759-786); the health-check RBAC stays
            await env.client.get("v1", "ServiceAccount", "health", "check-sa")
            with pytest.raises(Exception):
                await env.client.get("v1", "ServiceAccount", "health", "remedy-sa")
            with pytest.raises(Exception):
                await env.client.get(RBAC, "ClusterRole", "", "remedy-sa-cluster-role")

    run(go(), timeout=40)


def test_remedy_runs_limit_enforced(run):
    """With RemedyRunsLimit=2 and a long reset interval, the remedy stops
    after 2 runs while the check keeps failing (reference :681-694)."""

    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            hc = make_hc(name="limited", repeat=1, timeout=2, remedy=True)
            hc["spec"]["remedyRunsLimit"] = 2
            hc["spec"]["remedyResetInterval"] = 3600
            await env.create_hc(hc)

            async def failed_thrice():
                h = await env.get_hc("limited")
                return h.status.failed_count >= 4 and h

            h = await env.wait_for(failed_thrice, timeout=30, msg="4 failed runs")
            assert h.status.remedy_total_runs == 2  # capped at the limit

    run(go(), timeout=45)


def test_remedy_reset_interval_elapsed_resets_and_runs(run):
    """At the limit, once now-RemedyFinishedAt exceeds RemedyResetInterval the
    counters reset and the remedy runs again (reference :695-710)."""

    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            hc = make_hc(name="resetting", repeat=1, timeout=2, remedy=True)
            hc["spec"]["remedyRunsLimit"] = 1
            hc["spec"]["remedyResetInterval"] = 2
            await env.create_hc(hc)

            async def limit_reached():
                h = await env.get_hc("resetting")
                return h.status.remedy_total_runs >= 1 and h

            await env.wait_for(limit_reached, msg="first remedy run")

            # keep failing; after the 2s reset interval the counters must
            # reset (run count drops back) — observe the reset event
            async def reset_happened():
                evs = await env.client.list("v1", "Event", "health")
                return any(
                    "RemedyResetInterval elapsed so Remedy is reset" in e.get("message", "")
                    for e in evs
                )

            await env.wait_for(reset_happened, timeout=30, msg="reset-interval reset")

    run(go(), timeout=45)


def test_healthcheck_pass_resets_remedy(run):
    """After remedy runs, a passing health check zeroes all remedy state with
    the exact RemedyStatus string (reference :649-660)."""
    mode = {"fail": True}

    def policy(wf):
        name = wf["metadata"]["name"]
        if "-remedy-wf-" in name:
            return ("Succeeded", "")
        return ("Failed", "boom") if mode["fail"] else ("Succeeded", "")

    async def go():
        async with Env(policy=policy) as env:
            await env.create_hc(make_hc(name="healer", repeat=1, timeout=2, remedy=True))

            async def remedied():
                h = await env.get_hc("healer")
                return h.status.remedy_total_runs >= 1

            await env.wait_for(remedied, msg="remedy ran")
            mode["fail"] = False  # check starts passing

            async def reset():
                h = await env.get_hc("healer")
                return h.status.remedy_status == "HealthCheck Passed so Remedy is reset" and h

            h = await env.wait_for(reset, timeout=30, msg="remedy reset on pass")
            assert h.status.remedy_total_runs == 0
            assert h.status.remedy_success_count == 0
            assert h.status.remedy_failed_count == 0
            assert h.status.remedy_started_at is None
            assert h.status.remedy_finished_at is None
            assert h.status.status == "Succeeded"

    run(go(), timeout=45)


def test_remedy_without_limits_always_runs(run):
    """RunsLimit/ResetInterval unset → remedy runs on every failure
    (reference :712-720)."""

    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            await env.create_hc(make_hc(name="unlimited", repeat=1, timeout=2, remedy=True))

            async def many_remedies():
                h = await env.get_hc("unlimited")
                return h.status.remedy_total_runs >= 3 and h

            h = await env.wait_for(many_remedies, timeout=30, msg="3 remedy runs")
            assert h.status.remedy_success_count >= 3

    run(go(), timeout=45)


def test_remedy_nil_resource_errors_no_crash(run):
    """RemedyWorkflow set but Resource nil → error event, controller survives
    (issue #313; reference :312-315, edge test :238-270)."""

    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            hc = make_hc(name="nil-remedy", repeat=1, timeout=1)
            hc["spec"]["remedyworkflow"] = {"generateName": "x-", "workflowtimeout": 5}
            await env.create_hc(hc)
            await asyncio.sleep(0.5)
            evs = await env.client.list("v1", "Event", "health")
            assert any(
                "RemedyWorkflow is set but Resource is nil" in e.get("message", "")
                for e in evs
            )
            # controller still functional
            await env.create_hc(make_hc(name="fine", repeat=1, timeout=2))

            async def fine_ran():
                h = await env.get_hc("fine")
                return h.status.total_healthcheck_runs >= 1

            await env.wait_for(fine_ran, msg="other CR reconciled")

    run(go(), timeout=40)


def test_remedy_sa_missing_errors(run):
    """Remedy resource without serviceAccount → explicit error
    (reference :327-330)."""

    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            hc = make_hc(name="no-sa", repeat=1, timeout=1, remedy=True, remedy_sa="")
            await env.create_hc(hc)
            await asyncio.sleep(0.5)
            evs = await env.client.list("v1", "Event", "health")
            assert any(
                "ServiceAccount for the RemedyWorkflow is not specified" in e.get("message", "")
                for e in evs
            )

    run(go(), timeout=40)


def test_remedy_sa_collision_renamed(run):
    """Remedy SA == health-check SA → renamed '<sa>-remedy'
    (reference :316-319, unit test :461-502)."""

    async def go():
        async with Env(policy=fail_check_pass_remedy) as env:
            hc = make_hc(name="collide", repeat=1, timeout=2, remedy=True,
                         sa="shared-sa", remedy_sa="shared-sa")
            await env.create_hc(hc)

            async def remedy_ran():
                h = await env.get_hc("collide")
                return h.status.remedy_total_runs >= 1

            await env.wait_for(remedy_ran, msg="remedy with renamed SA")
            # during the remedy run the renamed SA existed; afterwards the
            # teardown removed it — assert via the events trail
            evs = await env.client.list("v1", "Event", "health")
            assert any("Successfully created remedyWorkflow" in e.get("message", "")
                       for e in evs)
            # the shared (health-check) SA must still exist, un-deleted
            await env.client.get("v1", "ServiceAccount", "health", "shared-sa")
            with pytest.raises(Exception):
                await env.client.get("v1", "ServiceAccount", "health", "shared-sa-remedy")

    run(go(), timeout=40)


def test_remedy_default_rules_are_write_scoped(run):
    """Remedy ClusterRole carries CRUD verbs, health-check role read-only
    (reference :85-120; unit test :310-457)."""
    seen = {}

    def policy(wf):
        name = wf["metadata"]["name"]
        if "-remedy-wf-" in name:
            return None  # keep remedy pending so its RBAC stays up
        return ("Failed", "boom")

    async def go():
        async with Env(policy=policy) as env:
            await env.create_hc(make_hc(name="verbs", repeat=5, timeout=2, remedy=True))

            async def remedy_role_up():
                try:
                    seen["role"] = await env.client.get(
                        RBAC, "ClusterRole", "", "remedy-sa-cluster-role"
                    )
                    return True
                except Exception:
                    return False

            await env.wait_for(remedy_role_up, timeout=30, msg="remedy ClusterRole")
            verbs = {v for rule in seen["role"]["rules"] for v in rule["verbs"]}
            assert verbs == {"get", "list", "watch", "create", "update", "patch", "delete"}
            assert not any("*" in rule["verbs"] for rule in seen["role"]["rules"])

    run(go(), timeout=45)
