"""Secure metrics endpoint: TLS + bearer authn on /metrics,
default-on in the CLI like the reference (cmd/main.go:74-85,139 —
``metrics-secure`` true with an authn/z filter). Probes stay open."""
import asyncio
import ssl

import pytest

from active_monitor_amd.cmd.main import build_parser
from active_monitor_amd.engine.endpoints import (
    MetricsSecurity,
    build_metrics_security,
    serve_endpoints,
)


async def _https_get(port, path, token=None):
    ctx = ssl.create_default_context()
    ctx.check_hostname = False
    ctx.verify_mode = ssl.CERT_NONE  # self-signed server cert
    reader, writer = await asyncio.open_connection("127.0.0.1", port, ssl=ctx)
    auth = f"Authorization: Bearer {token}\r\n" if token else ""
    writer.write(
        f"GET {path} HTTP/1.1\r\nHost: x\r\n{auth}Connection: close\r\n\r\n".encode()
    )
    await writer.drain()
    data = await reader.read()
    writer.close()
    return data.decode("latin-1", "replace")


async def _http_get(port, path):
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    writer.write(f"GET {path} HTTP/1.1\r\nHost: x\r\nConnection: close\r\n\r\n".encode())
    await writer.drain()
    data = await reader.read()
    writer.close()
    return data.decode("latin-1", "replace")


def test_cli_defaults_to_secure_metrics():
    args = build_parser().parse_args([])
    assert args.metrics_secure is True
    args = build_parser().parse_args(["--no-metrics-secure"])
    assert args.metrics_secure is False


def test_secured_metrics_requires_bearer_token(run, tmp_path):
    sec = build_metrics_security(True, workdir=str(tmp_path))
    assert sec is not None and sec.token and sec.ssl_context is not None
    # the generated token is persisted for scrapers
    assert (tmp_path / "metrics-token").read_text() == sec.token

    async def go():
        servers = await serve_endpoints(
            None, health=("127.0.0.1", 0), metrics=("127.0.0.1", 0),
            metrics_security=sec,
        )
        try:
            health_port = servers[0].sockets[0].getsockname()[1]
            metrics_port = servers[1].sockets[0].getsockname()[1]

            # probes remain open plain HTTP (reference :8081)
            assert "200 OK" in await _http_get(health_port, "/healthz")

            # metrics: no token → 401; wrong token → 401; right token → 200
            assert "401" in await _https_get(metrics_port, "/metrics")
            assert "401" in await _https_get(metrics_port, "/metrics", "nope")
            body = await _https_get(metrics_port, "/metrics", sec.token)
            assert "200 OK" in body and "healthcheck_success_count" in body
            # statusz is also behind the filter
            assert "401" in await _https_get(metrics_port, "/statusz")
        finally:
            for s in servers:
                s.close()

    run(go(), timeout=30)


def test_insecure_mode_serves_plain_http(run):
    async def go():
        servers = await serve_endpoints(None, metrics=("127.0.0.1", 0))
        try:
            port = servers[0].sockets[0].getsockname()[1]
            assert "200 OK" in await _http_get(port, "/metrics")
        finally:
            for s in servers:
                s.close()

    run(go(), timeout=30)


def test_token_file_is_used(run, tmp_path):
    tok = tmp_path / "tok"
    tok.write_text("sekret\n")
    sec = build_metrics_security(True, token_file=str(tok), workdir=str(tmp_path))
    assert sec.token == "sekret"


def test_probe_port_does_not_leak_observability(run, tmp_path):
    """With a dedicated secured metrics server, the open probe port must not
    serve /metrics or /statusz — the authn filter can't be sidestepped."""
    sec = build_metrics_security(True, workdir=str(tmp_path))

    async def go():
        servers = await serve_endpoints(
            None, health=("127.0.0.1", 0), metrics=("127.0.0.1", 0),
            metrics_security=sec,
        )
        try:
            health_port = servers[0].sockets[0].getsockname()[1]
            assert "200 OK" in await _http_get(health_port, "/healthz")
            assert "404" in await _http_get(health_port, "/metrics")
            assert "404" in await _http_get(health_port, "/statusz")
        finally:
            for s in servers:
                s.close()

    run(go(), timeout=30)


def test_cli_default_wires_security_into_manager(run, monkeypatch):
    """The entrypoint's default path hands the manager a MetricsSecurity."""
    import active_monitor_amd.engine as engine_pkg
    from active_monitor_amd.cmd.main import run as cmd_run

    captured = {}

    class FakeManager:
        def __init__(self, *a, **kw):
            import asyncio

            captured.update(kw)
            self.fatal = asyncio.Event()
            self.fatal_reason = None
            self.ready = True

        async def start(self):
            pass

        async def stop(self):
            pass

    monkeypatch.setattr(engine_pkg, "Manager", FakeManager)

    async def go():
        import asyncio

        from active_monitor_amd.cmd.main import build_parser

        args = build_parser().parse_args(
            ["--backend", "memory", "--workflow-engine", "none",
             "--metrics-bind-address", "127.0.0.1:0",
             "--health-probe-bind-address", "0"]
        )
        stop = asyncio.Event()
        stop.set()  # exit immediately after start
        return await cmd_run(args, stop)

    rc = run(go(), timeout=30)
    assert rc == 0
    sec = captured.get("metrics_security")
    assert sec is not None and sec.token and sec.ssl_context is not None
