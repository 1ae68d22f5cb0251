"""Observability assertions: event trail and metric values after real cycles
(the reference emits events on nearly every transition and epoch-valued
gauges — SURVEY §5)."""
import time

from active_monitor_amd.metrics import (
    MonitorError,
    MonitorFinishedTime,
    MonitorRuntime,
    MonitorStartedTime,
    MonitorSuccess,
)

from .conftest import Env, make_hc


def test_event_trail_of_success_cycle(run):
    async def go():
        async with Env() as env:
            await env.create_hc(make_hc(name="observed", repeat=1, timeout=2))

            async def ran():
                hc = await env.get_hc("observed")
                return hc.status.success_count >= 1

            await env.wait_for(ran, msg="one run")
            await env.manager.recorder.flush()
            evs = await env.client.list("v1", "Event", "health")
            messages = [e.get("message", "") for e in evs]
            for expected in (
                "Successfully created workflow",
                "Workflow status is Succeeded",
                "Rescheduled workflow for next run",
                "workflow is parsed from healthcheck",
            ):
                assert any(expected in m for m in messages), (expected, messages)
            # events carry the involved HealthCheck
            ev = next(e for e in evs if "Successfully created workflow" in e["message"])
            assert ev["involvedObject"]["kind"] == "HealthCheck"
            assert ev["involvedObject"]["name"] == "observed"
            assert ev["source"]["component"] == "active-monitor"

    run(go(), timeout=30)


def test_metric_values_after_cycles(run):
    async def go():
        async with Env() as env:
            await env.create_hc(make_hc(name="metered", repeat=1, timeout=2))

            async def two_runs():
                hc = await env.get_hc("metered")
                return hc.status.success_count >= 2

            await env.wait_for(two_runs, timeout=20, msg="two runs")

            assert MonitorSuccess.value("metered", "healthCheck") >= 2
            assert MonitorError.value("metered", "healthCheck") == 0
            # start/finished gauges carry Unix epochs (reference quirk §2.3.6)
            started = MonitorStartedTime.labels("metered", "healthCheck")._value.get()
            finished = MonitorFinishedTime.labels("metered", "healthCheck")._value.get()
            now = time.time()
            assert now - 300 < started <= now + 5
            assert started <= finished <= now + 5
            runtime = MonitorRuntime.labels("metered", "healthCheck")._value.get()
            assert 0 <= runtime < 60

    run(go(), timeout=40)


def test_failure_metrics_and_error_message(run):
    async def go():
        async with Env(policy=lambda wf: ("Failed", "probe exploded")) as env:
            await env.create_hc(make_hc(name="failing", repeat=1, timeout=2))

            async def failed():
                hc = await env.get_hc("failing")
                return hc.status.failed_count >= 1 and hc

            hc = await env.wait_for(failed, msg="failure observed")
            assert hc.status.error_message == "probe exploded"
            assert MonitorError.value("failing", "healthCheck") >= 1
            assert MonitorSuccess.value("failing", "healthCheck") == 0

    run(go(), timeout=30)
