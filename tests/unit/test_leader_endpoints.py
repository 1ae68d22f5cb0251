"""Leader election + HTTP endpoint unit tests
(reference: cmd/main.go:74-126 manager wiring)."""
import asyncio
import urllib.request

from active_monitor_amd.engine.endpoints import serve_endpoints
from active_monitor_amd.engine.leader import LeaderElector
from active_monitor_amd.kube import MemoryApiServer, MemoryClient


def test_leader_acquire_and_mutual_exclusion(run):
    async def go():
        client = MemoryClient(MemoryApiServer())
        a = LeaderElector(client, "689451f8.keikoproj.io", "health", "replica-a",
                          lease_duration=5)
        b = LeaderElector(client, "689451f8.keikoproj.io", "health", "replica-b",
                          lease_duration=5)
        assert await a.try_acquire() is True
        assert await b.try_acquire() is False  # held by a
        assert await a.try_acquire() is True   # renewal by holder OK
        await a.release()
        assert await b.try_acquire() is True   # freed lease acquired

    run(go())


def test_leader_takes_over_expired_lease(run):
    async def go():
        client = MemoryClient(MemoryApiServer())
        a = LeaderElector(client, "lease-x", "health", "a", lease_duration=0.1)
        b = LeaderElector(client, "lease-x", "health", "b", lease_duration=0.1)
        assert await a.try_acquire()
        await asyncio.sleep(0.25)  # a's lease expires
        assert await b.try_acquire() is True

    run(go())


def test_endpoints_healthz_readyz_metrics(run):
    class FakeManager:
        ready = True

    async def go():
        servers = await serve_endpoints(FakeManager(), health=("127.0.0.1", 0),
                                        metrics=("127.0.0.1", 0))
        try:
            results = {}
            for srv, path in zip(servers, ["/healthz", "/metrics"]):
                port = srv.sockets[0].getsockname()[1]
                for p in ({"/healthz", "/readyz", "/metrics", "/nope"}):
                    url = f"http://127.0.0.1:{port}{p}"
                    body, code = await asyncio.get_running_loop().run_in_executor(
                        None, _fetch, url
                    )
                    results[p] = (code, body)
            assert results["/healthz"][0] == 200
            assert results["/readyz"][0] == 200
            assert results["/metrics"][0] == 200
            assert b"healthcheck_success_count" in results["/metrics"][1]
            assert results["/nope"][0] == 404
        finally:
            for s in servers:
                s.close()

    run(go())


def _fetch(url):
    try:
        with urllib.request.urlopen(url, timeout=5) as r:
            return r.read(), r.status
    except urllib.error.HTTPError as e:
        return e.read(), e.code


def test_shared_address_starts_one_server(run):
    class FakeManager:
        ready = False

    async def go():
        servers = await serve_endpoints(FakeManager(), health=("127.0.0.1", 0),
                                        metrics=("127.0.0.1", 0))
        # distinct ports requested via 0 → two servers; same tuple → one
        for s in servers:
            s.close()
        servers2 = await serve_endpoints(
            FakeManager(), health=("127.0.0.1", 18931), metrics=("127.0.0.1", 18931)
        )
        try:
            assert len(servers2) == 1
            body, code = await asyncio.get_running_loop().run_in_executor(
                None, _fetch, "http://127.0.0.1:18931/readyz"
            )
            assert code == 503  # not ready
        finally:
            for s in servers2:
                s.close()

    run(go())


def test_cli_parse_bind_address():
    from active_monitor_amd.cmd.main import build_parser, parse_bind_address

    assert parse_bind_address(":8443") == ("0.0.0.0", 8443)
    assert parse_bind_address("127.0.0.1:9000") == ("127.0.0.1", 9000)
    assert parse_bind_address("0") is None
    assert parse_bind_address("") is None

    args = build_parser().parse_args([])
    # reference flag defaults (cmd/main.go:138-144)
    assert args.metrics_bind_address == ":8443"
    assert args.health_probe_bind_address == ":8081"
    assert args.max_workers == 10
    assert args.leader_elect is False


def test_json_log_formatter():
    import json as _json
    import logging

    from active_monitor_amd.cmd.logfmt import JsonFormatter

    rec = logging.LogRecord("active_monitor_amd.x", logging.WARNING, "f.py", 1,
                            "reconcile %s failed", ("hc-1",), None)
    out = _json.loads(JsonFormatter().format(rec))
    assert out["level"] == "warning"
    assert out["logger"] == "active_monitor_amd.x"
    assert out["msg"] == "reconcile hc-1 failed"
    assert isinstance(out["ts"], float)


def test_cli_log_format_flag():
    from active_monitor_amd.cmd.main import build_parser

    args = build_parser().parse_args(["--log-format", "json"])
    assert args.log_format == "json"
    assert build_parser().parse_args([]).log_format == "console"


def test_statusz_endpoint(run):
    class FakeRec:
        reconcile_count = 7
        completed_runs = 5
        repeat_timers_by_name = {("health", "a"): object()}

        def active_watches(self):
            return 2

    class FakeManager:
        ready = True
        max_workers = 4
        shard_index = 1
        shard_count = 3
        queue = []
        reconciler = FakeRec()

    async def go():
        servers = await serve_endpoints(FakeManager(), health=("127.0.0.1", 0))
        try:
            port = servers[0].sockets[0].getsockname()[1]
            body, code = await asyncio.get_running_loop().run_in_executor(
                None, _fetch, f"http://127.0.0.1:{port}/statusz"
            )
            assert code == 200
            import json

            stats = json.loads(body)
            assert stats["ready"] is True
            assert stats["workers"] == 4
            assert stats["shard"] == [1, 3]
            assert stats["reconciles"] == 7
            assert stats["completed_runs"] == 5
            assert stats["active_watches"] == 2
            assert stats["armed_timers"] == 1
        finally:
            for s in servers:
                s.close()

    run(go())
