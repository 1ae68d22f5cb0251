"""HTTP endpoints: /healthz, /readyz probes and Prometheus /metrics.

The reference serves health probes at :8081 and metrics at :8443 through the
controller-runtime manager (cmd/main.go:74-85,121-126, README documents
:8080/metrics). A dependency-light asyncio HTTP/1.1 server is plenty here.

Metrics security parity (cmd/main.go:74-85,139 — ``metrics-secure`` defaults
to TRUE with an authn/z filter): :class:`MetricsSecurity` carries an optional
TLS context and bearer token; when set, /metrics and /statusz require
``Authorization: Bearer <token>`` and are served over HTTPS. Like
controller-runtime, a self-signed certificate is generated at startup when
none is provided (via the openssl CLI). Probes stay open (reference :8081 is
plain HTTP).
"""
from __future__ import annotations

import asyncio
import logging
import ssl as ssl_mod
from dataclasses import dataclass
from typing import List, Optional, Tuple

from ..metrics import exposition

log = logging.getLogger("active_monitor_amd.endpoints")

#: endpoints behind the authn filter when a bearer token is configured
_PROTECTED = ("/metrics", "/statusz")


@dataclass
class MetricsSecurity:
    """TLS + bearer-token protection for the metrics endpoint."""

    ssl_context: Optional[ssl_mod.SSLContext] = None
    token: Optional[str] = None


def generate_self_signed_cert(directory: str) -> Tuple[str, str]:
    """Write a self-signed cert/key pair (openssl CLI) and return their
    paths — the controller-runtime behavior when no cert is provided."""
    import os
    import subprocess

    cert = os.path.join(directory, "tls.crt")
    key = os.path.join(directory, "tls.key")
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-keyout", key,
         "-out", cert, "-days", "365", "-nodes",
         "-subj", "/CN=active-monitor-metrics"],
        check=True, capture_output=True,
    )
    return cert, key


def build_metrics_security(
    secure: bool,
    cert: Optional[str] = None,
    key: Optional[str] = None,
    token_file: Optional[str] = None,
    workdir: Optional[str] = None,
) -> Optional[MetricsSecurity]:
    """Resolve CLI flags into a MetricsSecurity (None = plain/open).

    secure=True with no cert generates a self-signed pair; with no token
    file, a random bearer token is generated and written next to the certs
    (its path is logged) so scrapers can authenticate."""
    if not secure:
        return None
    import os
    import secrets
    import tempfile

    if workdir is None:
        workdir = tempfile.mkdtemp(prefix="am-metrics-")
    if not cert or not key:
        cert, key = generate_self_signed_cert(workdir)
        log.info("metrics: generated self-signed certificate at %s", cert)
    ctx = ssl_mod.SSLContext(ssl_mod.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(cert, key)
    if token_file:
        with open(token_file) as f:
            token = f.read().strip()
    else:
        token = secrets.token_urlsafe(32)
        token_path = os.path.join(workdir, "metrics-token")
        with open(token_path, "w") as f:
            f.write(token)
        os.chmod(token_path, 0o600)
        log.info("metrics: generated bearer token at %s", token_path)
    return MetricsSecurity(ssl_context=ctx, token=token)


async def _handle(reader: asyncio.StreamReader, writer: asyncio.StreamWriter,
                  manager, auth_token: Optional[str] = None,
                  serve_observability: bool = True) -> None:
    try:
        request_line = await asyncio.wait_for(reader.readline(), 10)
        if not request_line:
            return
        parts = request_line.decode("latin-1").split()
        path = parts[1] if len(parts) >= 2 else "/"
        # drain headers, capturing Authorization for the authn filter
        bearer = None
        while True:
            line = await asyncio.wait_for(reader.readline(), 10)
            if line in (b"\r\n", b"\n", b""):
                break
            if line.lower().startswith(b"authorization:"):
                value = line.split(b":", 1)[1].strip().decode("latin-1")
                if value.lower().startswith("bearer "):
                    bearer = value[7:]

        if not serve_observability and path.startswith(_PROTECTED):
            # probe-only server: /metrics and /statusz live on the (secured)
            # metrics endpoint, not the open probe port
            body = b"not found"
            writer.write(
                (
                    "HTTP/1.1 404 Not Found\r\nContent-Type: text/plain\r\n"
                    f"Content-Length: {len(body)}\r\nConnection: close\r\n\r\n"
                ).encode("latin-1") + body
            )
            await writer.drain()
            return
        if auth_token is not None and path.startswith(_PROTECTED):
            if bearer != auth_token:
                body = b"Unauthorized"
                writer.write(
                    (
                        "HTTP/1.1 401 Unauthorized\r\n"
                        'WWW-Authenticate: Bearer realm="metrics"\r\n'
                        "Content-Type: text/plain\r\n"
                        f"Content-Length: {len(body)}\r\n"
                        "Connection: close\r\n\r\n"
                    ).encode("latin-1")
                    + body
                )
                await writer.drain()
                return

        if path.startswith("/statusz"):
            import json

            stats = {}
            if manager is not None:
                rec = getattr(manager, "reconciler", None)
                stats = {
                    "ready": manager.ready,
                    "workers": getattr(manager, "max_workers", None),
                    "shard": [getattr(manager, "shard_index", 0),
                              getattr(manager, "shard_count", 1)],
                    "queue_depth": len(getattr(manager, "queue", [])),
                    "reconciles": getattr(rec, "reconcile_count", 0),
                    "completed_runs": getattr(rec, "completed_runs", 0),
                    "active_watches": rec.active_watches() if rec else 0,
                    "armed_timers": len(getattr(rec, "repeat_timers_by_name", {})),
                    "apiserver_requests": getattr(
                        getattr(manager, "client", None), "request_count", None
                    ),
                    "owned_shards": sorted(
                        getattr(getattr(manager, "coordinator", None), "owned", ())
                    ),
                }
            body, ctype, code = json.dumps(stats).encode(), "application/json", 200
        elif path.startswith("/healthz"):
            body, ctype, code = b"ok", "text/plain", 200
        elif path.startswith("/readyz"):
            if manager is None or manager.ready:
                body, ctype, code = b"ok", "text/plain", 200
            else:
                body, ctype, code = b"not ready", "text/plain", 503
        elif path.startswith("/metrics"):
            body, ctype, code = exposition(), "text/plain; version=0.0.4; charset=utf-8", 200
        else:
            body, ctype, code = b"not found", "text/plain", 404

        status = {200: "OK", 404: "Not Found", 503: "Service Unavailable"}[code]
        writer.write(
            (
                f"HTTP/1.1 {code} {status}\r\n"
                f"Content-Type: {ctype}\r\n"
                f"Content-Length: {len(body)}\r\n"
                "Connection: close\r\n\r\n"
            ).encode("latin-1")
            + body
        )
        await writer.drain()
    except (asyncio.TimeoutError, ConnectionError):
        pass
    finally:
        try:
            writer.close()
        except Exception:
            pass


async def serve_endpoints(
    manager,
    health: Optional[Tuple[str, int]] = None,
    metrics: Optional[Tuple[str, int]] = None,
    metrics_security: Optional[MetricsSecurity] = None,
) -> List[asyncio.AbstractServer]:
    """Start probe/metrics servers; returns the server objects (close() them
    to shut down). When health == metrics only one server is started (and
    any metrics security applies to it as a whole)."""
    servers: List[asyncio.AbstractServer] = []

    token = metrics_security.token if metrics_security else None
    ssl_ctx = metrics_security.ssl_context if metrics_security else None
    # when a dedicated (possibly secured) metrics server exists, the probe
    # server is probes-only: /metrics and /statusz 404 there so the authn
    # filter cannot be sidestepped via the open port. (Equal non-ephemeral
    # addresses mean one genuinely shared server; port 0 is always distinct.)
    shared = health is not None and health == metrics and health[1] != 0
    probes_only = metrics is not None and not shared

    async def open_handler(r, w):
        await _handle(r, w, manager, serve_observability=not probes_only)

    async def secured_handler(r, w):
        await _handle(r, w, manager, auth_token=token)

    # port 0 is ephemeral — two (host, 0) requests are distinct servers
    if health is not None and health == metrics and health[1] != 0:
        servers.append(await asyncio.start_server(
            secured_handler, health[0], health[1], ssl=ssl_ctx))
        return servers
    if health is not None:
        servers.append(await asyncio.start_server(open_handler, health[0], health[1]))
    if metrics is not None:
        servers.append(await asyncio.start_server(
            secured_handler, metrics[0], metrics[1], ssl=ssl_ctx))
    return servers
