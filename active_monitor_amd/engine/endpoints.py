"""HTTP endpoints: /healthz, /readyz probes and Prometheus /metrics.

The reference serves health probes at :8081 and metrics at :8443 through the
controller-runtime manager (cmd/main.go:74-85,121-126, README documents
:8080/metrics). A dependency-light asyncio HTTP/1.1 server is plenty here.
"""
from __future__ import annotations

import asyncio
from typing import List, Optional, Tuple

from ..metrics import exposition


async def _handle(reader: asyncio.StreamReader, writer: asyncio.StreamWriter, manager) -> None:
    try:
        request_line = await asyncio.wait_for(reader.readline(), 10)
        if not request_line:
            return
        parts = request_line.decode("latin-1").split()
        path = parts[1] if len(parts) >= 2 else "/"
        # drain headers
        while True:
            line = await asyncio.wait_for(reader.readline(), 10)
            if line in (b"\r\n", b"\n", b""):
                break

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
) -> List[asyncio.AbstractServer]:
    """Start probe/metrics servers; returns the server objects (close() them
    to shut down). When health == metrics only one server is started."""
    servers: List[asyncio.AbstractServer] = []

    async def handler(r, w):
        await _handle(r, w, manager)

    seen = set()
    for addr in (health, metrics):
        if addr is None or addr in seen:
            continue
        seen.add(addr)
        servers.append(await asyncio.start_server(handler, addr[0], addr[1]))
    return servers
