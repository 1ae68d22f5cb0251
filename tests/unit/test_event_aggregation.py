"""EventRecorder aggregation/rate-limiting.

client-go's EventCorrelator dedups repeats of the same event into one Event
object with an incrementing ``count`` and caps the write rate per key with a
token bucket; the recorder reproduces that shape so a fleet doesn't drown a
real apiserver in Event creates (~6-8 per reconcile cycle otherwise).
"""
import asyncio

import pytest

from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.kube.client import EventRecorder


def _hc(name="hc-1", ns="health"):
    return {
        "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
        "kind": "HealthCheck",
        "metadata": {"name": name, "namespace": ns, "uid": "u1"},
    }


def test_repeats_dedup_into_one_event_with_count(run):
    async def go():
        server = MemoryApiServer()
        rec = EventRecorder(MemoryClient(server))
        for _ in range(5):
            await rec.event(_hc(), "Normal", "WorkflowCreated",
                            "Successfully created workflow")
        await rec.flush()
        events = server.list("v1", "Event", "health")
        assert len(events) == 1, [e["metadata"]["name"] for e in events]
        assert events[0]["count"] == 5
        assert events[0]["reason"] == "WorkflowCreated"
        assert events[0]["firstTimestamp"] and events[0]["lastTimestamp"]
        await rec.stop()

    run(go(), timeout=20)


def test_distinct_reasons_and_objects_stay_separate(run):
    async def go():
        server = MemoryApiServer()
        rec = EventRecorder(MemoryClient(server))
        await rec.event(_hc("a"), "Normal", "WorkflowCreated", "m1")
        await rec.event(_hc("a"), "Warning", "WorkflowFailed", "m2")
        await rec.event(_hc("b"), "Normal", "WorkflowCreated", "m1")
        await rec.flush()
        events = server.list("v1", "Event", "health")
        assert len(events) == 3
        await rec.stop()

    run(go(), timeout=20)


def test_spam_filter_caps_write_rate(run):
    """Beyond the burst, repeats stop hitting the wire; the local count keeps
    accruing and the next allowed write carries it."""

    async def go():
        server = MemoryApiServer()
        rec = EventRecorder(MemoryClient(server))
        n = rec.SPAM_BURST + 200
        for _ in range(n):
            await rec.event(_hc(), "Normal", "R", "msg")
        await rec.flush()
        writes = server.op_counts["create"] + server.op_counts["update"]
        assert writes <= rec.SPAM_BURST + 1, f"{writes} wire writes for {n} events"
        assert rec.suppressed >= 200 - 1
        events = server.list("v1", "Event", "health")
        assert len(events) == 1
        # the last allowed write carried the then-current aggregate count
        assert events[0]["count"] >= rec.SPAM_BURST
        await rec.stop()

    run(go(), timeout=20)


def test_aggregation_window_rollover(run):
    """After AGG_TTL the key starts a fresh aggregation (client-go interval)."""

    async def go():
        server = MemoryApiServer()
        rec = EventRecorder(MemoryClient(server))
        await rec.event(_hc(), "Normal", "R", "msg")
        await rec.flush()
        # expire the window
        key = next(iter(rec._agg))
        rec._agg[key]["t0"] -= rec.AGG_TTL + 1
        await rec.event(_hc(), "Normal", "R", "msg")
        await rec.flush()
        events = server.list("v1", "Event", "health")
        # same deterministic name → still one object, count restarted at 1
        assert len(events) == 1
        assert events[0]["count"] == 1
        await rec.stop()

    run(go(), timeout=20)


def test_recreate_after_server_side_ttl(run):
    """If the apiserver TTL'd the Event object away, a later update recreates
    it instead of erroring into the void."""

    async def go():
        server = MemoryApiServer()
        rec = EventRecorder(MemoryClient(server))
        await rec.event(_hc(), "Normal", "R", "msg")
        await rec.flush()
        ev = server.list("v1", "Event", "health")[0]
        server.delete("v1", "Event", "health", ev["metadata"]["name"])
        await rec.event(_hc(), "Normal", "R", "msg")
        await rec.flush()
        events = server.list("v1", "Event", "health")
        assert len(events) == 1
        assert events[0]["count"] == 2
        await rec.stop()

    run(go(), timeout=20)
