"""CLI process-level tests (reference: cmd/main_test.go run()
error/shutdown paths; cmd/main_envtest_test.go wiring)."""
import asyncio
import urllib.error
import urllib.request

from active_monitor_amd.cmd.main import build_parser
from active_monitor_amd.cmd.main import run as cli_run


def _fetch(url):
    try:
        with urllib.request.urlopen(url, timeout=5) as r:
            return r.status, r.read()
    except urllib.error.HTTPError as e:
        return e.code, e.read()


def test_run_memory_backend_startup_and_shutdown(run):
    """Full process lifecycle: memory backend + local engine + probe/metrics
    servers, clean shutdown via the stop event."""

    async def go():
        args = build_parser().parse_args([
            "--backend", "memory",
            "--max-workers", "3",
            "--metrics-bind-address", "127.0.0.1:18181",
            "--health-probe-bind-address", "127.0.0.1:18182",
        ])
        stop = asyncio.Event()
        task = asyncio.ensure_future(cli_run(args, stop))
        loop = asyncio.get_running_loop()

        # probes respond while running
        deadline = loop.time() + 10
        code = None
        while loop.time() < deadline:
            try:
                code, _ = await loop.run_in_executor(
                    None, _fetch, "http://127.0.0.1:18182/healthz"
                )
                break
            except Exception:
                await asyncio.sleep(0.1)
        assert code == 200
        code, body = await loop.run_in_executor(
            None, _fetch, "http://127.0.0.1:18181/metrics"
        )
        assert code == 200 and b"healthcheck_success_count" in body
        code, _ = await loop.run_in_executor(
            None, _fetch, "http://127.0.0.1:18182/readyz"
        )
        assert code == 200

        stop.set()
        rc = await asyncio.wait_for(task, 15)
        assert rc == 0

    run(go(), timeout=40)


def test_run_disabled_endpoints(run):
    """'0' disables an endpoint (reference flag semantics)."""

    async def go():
        args = build_parser().parse_args([
            "--backend", "memory",
            "--metrics-bind-address", "0",
            "--health-probe-bind-address", "0",
            "--workflow-engine", "none",
        ])
        stop = asyncio.Event()
        task = asyncio.ensure_future(cli_run(args, stop))
        await asyncio.sleep(0.3)
        stop.set()
        assert await asyncio.wait_for(task, 15) == 0

    run(go(), timeout=30)


def test_run_http_backend_unreachable_server_errors(run):
    """Invalid apiserver config → run() returns an error rather than running
    blind (reference cmd/main_test.go:35-49)."""

    async def go():
        args = build_parser().parse_args([
            "--backend", "http",
            "--server", "http://127.0.0.1:1",  # nothing listens there
            "--metrics-bind-address", "0",
            "--health-probe-bind-address", "0",
        ])
        rc = await asyncio.wait_for(cli_run(args, asyncio.Event()), 30)
        assert rc == 1

    run(go(), timeout=40)
