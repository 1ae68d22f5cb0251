"""Real-apiserver watch conformance for the HTTP backend.

The reference inherits client-go's reflector, which survives watch expiry by
re-listing (healthcheck_controller.go:133-137 via controller-runtime). These
tests force the same failure modes through the wire — stream kicks, 410 Gone
on stale resourceVersions, oversized events — and assert the HttpClient
recovers without losing events or wedging.
"""
import asyncio

import pytest

from active_monitor_amd import API_VERSION
from active_monitor_amd.kube.errors import ExpiredError
from active_monitor_amd.kube.http import HttpClient, _TokenBucket
from active_monitor_amd.kube.memory import MemoryApiServer
from active_monitor_amd.kube.server import ApiServerFrontend

from .conftest import make_hc


class Env:
    async def __aenter__(self):
        self.server = MemoryApiServer()
        self.frontend = ApiServerFrontend(self.server)
        await self.frontend.start()
        self.client = HttpClient(self.frontend.url, qps=0)  # unthrottled tests
        await self.client.start()
        return self

    async def __aexit__(self, *exc):
        await self.client.close()
        await self.frontend.stop()


async def _collect_until(sub, want_names, timeout=10.0):
    """Drain the subscription until every name in want_names was seen."""
    seen = set()
    deadline = asyncio.get_running_loop().time() + timeout
    while want_names - seen:
        remaining = deadline - asyncio.get_running_loop().time()
        assert remaining > 0, f"timed out; saw {seen}, wanted {want_names}"
        ev = await asyncio.wait_for(sub.__anext__(), remaining)
        seen.add(ev["object"]["metadata"]["name"])
    return seen


def test_watch_survives_kick_and_resumes(run):
    """A dropped stream reconnects with its resourceVersion and replays the
    events that fired while disconnected (reflector resume, no re-list)."""

    async def go():
        async with Env() as env:
            sub = env.client.watch(API_VERSION, "HealthCheck", "health")
            await env.client.create(make_hc(name="before"))
            await _collect_until(sub, {"before"})

            env.frontend.kick_watches()
            # events landing around the reconnect window must still arrive
            await env.client.create(make_hc(name="during"))
            await _collect_until(sub, {"during"})
            sub.close()

    run(go(), timeout=30)


def test_watch_410_triggers_relist_recovery(run):
    """With the server's event history compacted away, the resumed watch gets
    410 Gone; the client must clear its resourceVersion, re-list, surface the
    full current state, and keep streaming — not reconnect-loop forever
    (kube/http.py:93)."""

    async def go():
        async with Env() as env:
            sub = env.client.watch(API_VERSION, "HealthCheck", "health")
            old = await env.client.create(make_hc(name="old"))
            await _collect_until(sub, {"old"})

            # compact the entire history so any stale resume rv is too old
            env.server.history_window = 0
            await env.client.create(make_hc(name="compacted"))
            await _collect_until(sub, {"compacted"})
            # pin the client's resume point behind the compaction horizon —
            # the deterministic stand-in for a client that fell behind while
            # disconnected (real wall-clock races can't be scripted)
            sub._resource_version = old["metadata"]["resourceVersion"]
            env.frontend.kick_watches()

            # recovery = re-list: both live objects surface as ADDED
            await _collect_until(sub, {"old", "compacted"})
            assert sub.relists >= 1, "client recovered without re-listing?"

            # and the stream keeps working after recovery
            await env.client.create(make_hc(name="after"))
            await _collect_until(sub, {"after"})
            sub.close()

    run(go(), timeout=30)


def test_oversized_watch_event_streams_intact(run):
    """A watch event larger than aiohttp's 64 KB readline ceiling must be
    framed by the incremental reader, not raise and replay forever
    (kube/http.py:93 second half)."""

    async def go():
        async with Env() as env:
            sub = env.client.watch(API_VERSION, "HealthCheck", "health")
            await asyncio.sleep(0.2)
            big = make_hc(name="big")
            big["metadata"]["annotations"] = {"blob": "x" * (256 * 1024)}
            await env.client.create(big)
            ev = await asyncio.wait_for(sub.__anext__(), 10)
            assert ev["object"]["metadata"]["name"] == "big"
            assert len(ev["object"]["metadata"]["annotations"]["blob"]) == 256 * 1024
            sub.close()

    run(go(), timeout=30)


def test_memory_events_since_semantics(run):
    """The store-level contract the frontend serves: replay after rv, 410
    below the compaction horizon, DELETED events carry a fresh rv."""

    async def go():
        server = MemoryApiServer()
        created = server.create(make_hc(name="a"))
        rv0 = created["metadata"]["resourceVersion"]
        server.create(make_hc(name="b"))
        server.delete(API_VERSION, "HealthCheck", "health", "a")

        evs = server.events_since(API_VERSION, "HealthCheck", "health", rv0)
        kinds = [(e["type"], e["object"]["metadata"]["name"]) for e in evs]
        assert ("ADDED", "b") in kinds
        assert ("DELETED", "a") in kinds  # delete bumped the rv → replayable

        server.history_window = 0
        server.create(make_hc(name="c"))  # compacts everything
        with pytest.raises(ExpiredError):
            server.events_since(API_VERSION, "HealthCheck", "health", rv0)

    run(go(), timeout=10)


def test_client_side_rate_limiter(run):
    """Token bucket paces requests at qps once the burst is spent, FIFO."""

    async def go():
        bucket = _TokenBucket(qps=100.0, burst=2)
        loop = asyncio.get_running_loop()
        t0 = loop.time()
        for _ in range(6):
            await bucket.acquire()
        elapsed = loop.time() - t0
        # 2 free (burst) + 4 paced at 10ms each ≈ 40ms minimum
        assert elapsed >= 0.035, f"limiter did not pace: {elapsed:.4f}s"

        unlimited = _TokenBucket(qps=0, burst=0)
        t0 = loop.time()
        for _ in range(100):
            await unlimited.acquire()
        assert loop.time() - t0 < 0.05

    run(go(), timeout=10)


def test_watch_timeout_seconds_closes_stream(run):
    """kubectl sends timeoutSeconds on watches; the server must end the
    stream cleanly at the budget (clients resume from their rv)."""
    import aiohttp

    async def go():
        async with Env() as env:
            await env.client.create(make_hc(name="twatch"))
            url = (env.frontend.url
                   + "/apis/activemonitor.keikoproj.io/v1alpha1/namespaces/"
                     "health/healthchecks?watch=true&timeoutSeconds=0.7")
            t0 = asyncio.get_running_loop().time()
            lines = []
            async with aiohttp.ClientSession() as s:
                async with s.get(url, timeout=aiohttp.ClientTimeout(total=10)) as r:
                    assert r.status == 200
                    async for line in r.content:
                        if line.strip():
                            lines.append(line)
            elapsed = asyncio.get_running_loop().time() - t0
            assert lines, "initial ADDED replay missing"
            assert elapsed < 5, f"stream did not close at the budget ({elapsed:.1f}s)"

    run(go(), timeout=30)


def test_watch_field_selector_filters_stream(run):
    """kubectl get hc X -w watches the collection with
    fieldSelector=metadata.name=X — only that object's events may stream."""
    import aiohttp
    import json as _json

    async def go():
        async with Env() as env:
            await env.client.create(make_hc(name="fs-a"))
            await env.client.create(make_hc(name="fs-b"))
            url = (env.frontend.url
                   + "/apis/activemonitor.keikoproj.io/v1alpha1/namespaces/"
                     "health/healthchecks?watch=true&timeoutSeconds=1.2"
                     "&fieldSelector=metadata.name%3Dfs-a")
            names = set()
            async with aiohttp.ClientSession() as s:
                async with s.get(url, timeout=aiohttp.ClientTimeout(total=10)) as r:
                    assert r.status == 200
                    async for line in r.content:
                        if line.strip():
                            ev = _json.loads(line)
                            names.add(ev["object"]["metadata"]["name"])
            assert names == {"fs-a"}, names

    run(go(), timeout=30)
