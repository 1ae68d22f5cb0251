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
            # plain-HTTP scrape: this test covers lifecycle; the secure
            # default path is covered by tests/unit/test_metrics_security.py
            "--no-metrics-secure",
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


def test_standalone_stack_serve_api_with_amctl(run, capsys, tmp_path):
    """The full standalone product: CLI controller (memory backend + local
    subprocess engine + served REST API) driven by amctl; a health check
    executes end-to-end."""
    import yaml

    from active_monitor_amd.cmd.ctl import build_parser as ctl_parser
    from active_monitor_amd.cmd.ctl import run as ctl_run

    async def go():
        args = build_parser().parse_args([
            "--backend", "memory",
            "--serve-api", "127.0.0.1:18281",
            "--metrics-bind-address", "0",
            "--health-probe-bind-address", "0",
            "--max-workers", "4",
        ])
        stop = asyncio.Event()
        task = asyncio.ensure_future(cli_run(args, stop))
        url = "http://127.0.0.1:18281"

        async def ctl(*argv):
            rc = await ctl_run(ctl_parser().parse_args(["--server", url, *argv]))
            return rc, capsys.readouterr().out

        # wait for the API to come up
        import aiohttp
        deadline = asyncio.get_running_loop().time() + 15
        async with aiohttp.ClientSession() as s:
            while asyncio.get_running_loop().time() < deadline:
                try:
                    async with s.get(url + "/version"):
                        break
                except aiohttp.ClientError:
                    await asyncio.sleep(0.1)

        doc = {
            "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
            "kind": "HealthCheck",
            "metadata": {"name": "standalone", "namespace": "health"},
            "spec": {
                "repeatAfterSec": 60, "level": "cluster",
                "workflow": {
                    "generateName": "standalone-wf-",
                    "workflowtimeout": 20,
                    "resource": {
                        "namespace": "health", "serviceAccount": "sa",
                        "source": {"inline": (
                            "spec:\n  entrypoint: main\n  templates:\n"
                            "    - name: main\n      container:\n"
                            "        command: [\"true\"]\n"
                        )},
                    },
                },
            },
        }
        f = tmp_path / "hc.yaml"
        f.write_text(yaml.safe_dump(doc))
        rc, out = await ctl("apply", "-f", str(f))
        assert rc == 0 and "created" in out

        # the local engine executes `true` and the controller records success
        deadline = asyncio.get_running_loop().time() + 25
        ok = False
        while asyncio.get_running_loop().time() < deadline:
            rc, out = await ctl("get", "hc", "standalone", "-n", "health")
            line = next(l for l in out.splitlines() if l.startswith("standalone"))
            if line.split()[1] == "Succeeded":
                ok = True
                break
            await asyncio.sleep(0.2)
        assert ok, f"standalone check never succeeded: {out}"

        stop.set()
        assert await asyncio.wait_for(task, 15) == 0

    run(go(), timeout=60)
