"""Controller process entry point (reference: cmd/main.go).

Flag parity with the reference (cmd/main.go:138-144):

- ``--metrics-bind-address`` (default ``:8443``)
- ``--health-probe-bind-address`` (default ``:8081``)
- ``--leader-elect`` (default off; LeaderElectionID ``689451f8.keikoproj.io``)
- ``--max-workers`` (default 10 — MaxConcurrentReconciles)

plus the backend selection this framework adds:

- ``--backend memory`` (default): self-contained in-memory apiserver with the
  local subprocess workflow engine — run health checks standalone, no cluster.
- ``--backend http --server URL``: a real Kubernetes apiserver (workflows are
  executed by the cluster's Argo controller, exactly like the reference).
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import signal
import sys
from typing import Optional, Tuple


def parse_bind_address(addr: str) -> Optional[Tuple[str, int]]:
    """':8443' → ('0.0.0.0', 8443); '0' or '' disables."""
    if not addr or addr == "0":
        return None
    host, _, port = addr.rpartition(":")
    return (host or "0.0.0.0", int(port))


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="active-monitor-amd",
        description="HealthCheck/Remedy controller (clean-room active-monitor)",
    )
    p.add_argument("--metrics-bind-address", default=":8443",
                   help="metrics endpoint address ('0' disables)")
    p.add_argument("--metrics-secure", default=True,
                   action=argparse.BooleanOptionalAction,
                   help="serve /metrics over HTTPS with bearer-token authn "
                        "(reference default true, cmd/main.go:139); "
                        "--no-metrics-secure serves plain HTTP")
    p.add_argument("--metrics-cert", default=None,
                   help="TLS certificate for the metrics endpoint "
                        "(default: self-signed generated at startup)")
    p.add_argument("--metrics-key", default=None,
                   help="TLS key for the metrics endpoint")
    p.add_argument("--metrics-auth-token-file", default=None,
                   help="file holding the bearer token scrapers must present "
                        "(default: a random token generated at startup; its "
                        "path is logged)")
    p.add_argument("--health-probe-bind-address", default=":8081",
                   help="healthz/readyz endpoint address ('0' disables)")
    p.add_argument("--leader-elect", action="store_true",
                   help="enable leader election for controller manager")
    p.add_argument("--max-workers", type=int, default=10,
                   help="maximum number of concurrent reconciles")
    p.add_argument("--namespace", default=None,
                   help="restrict watch to one namespace (default: all)")
    p.add_argument("--backend", choices=["memory", "http"], default="memory")
    p.add_argument("--server", default="",
                   help="apiserver URL for --backend http (default: in-cluster "
                        "config, then $KUBECONFIG/~/.kube/config)")
    p.add_argument("--kubeconfig", default=None,
                   help="explicit kubeconfig path for --backend http")
    p.add_argument("--token", default="", help="bearer token for --backend http")
    p.add_argument("--insecure-skip-tls-verify", action="store_true")
    p.add_argument("--workflow-engine", choices=["local", "none"], default="local",
                   help="memory backend: execute workflows locally or leave to an "
                        "external engine")
    p.add_argument("--serve-api", default="",
                   help="memory backend: also serve the store over the "
                        "Kubernetes REST API at this address (kubectl-able)")
    p.add_argument("--shard-index", type=int, default=0,
                   help="this controller's shard (keyspace split by CR-name hash)")
    p.add_argument("--shard-count", type=int, default=1,
                   help="total cooperating controller shards")
    p.add_argument("--shard-ha", action="store_true",
                   help="lease-per-shard failure takeover: live shards adopt "
                        "a dead shard's keys; a restarted shard reclaims its "
                        "keys via the lease preferredHolder handshake")
    p.add_argument("--shard-lease-duration", type=float, default=15.0,
                   help="shard-ha lease TTL seconds (takeover latency bound)")
    p.add_argument("--shard-renew-interval", type=float, default=5.0,
                   help="shard-ha lease renew/scan cadence seconds")
    p.add_argument("--zap-log-level", default="info",
                   help="log level (debug/info/warn/error)")
    p.add_argument("--log-format", choices=["console", "json"], default="console",
                   help="console or structured JSON lines (the reference's "
                        "zap production encoding)")
    p.add_argument("--gc-threshold", type=int, default=50000,
                   help="gen-0 GC threshold; large fleets hold 100k+ live "
                        "objects and default CPython thresholds cost ~30%% "
                        "throughput (0 keeps the interpreter default)")
    return p


async def run(args, stop_event: Optional[asyncio.Event] = None) -> int:
    """Process main loop; runs until SIGINT/SIGTERM (or ``stop_event`` for
    tests — the reference covers run()'s error/shutdown paths the same way,
    cmd/main_test.go:35-64)."""
    return await _run(args, stop_event)


async def _run(args, stop_event: Optional[asyncio.Event] = None) -> int:
    if getattr(args, "gc_threshold", 0):
        import gc

        gc.set_threshold(args.gc_threshold, 50, 50)
    from ..engine import Manager
    from ..kube import MemoryApiServer, MemoryClient

    level = {"debug": logging.DEBUG, "info": logging.INFO,
             "warn": logging.WARNING, "error": logging.ERROR}.get(
        args.zap_log_level, logging.INFO)
    if getattr(args, "log_format", "console") == "json":
        from .logfmt import JsonFormatter

        handler = logging.StreamHandler()
        handler.setFormatter(JsonFormatter())
        logging.basicConfig(level=level, handlers=[handler], force=True)
    else:
        logging.basicConfig(
            level=level,
            format="%(asctime)s %(levelname)s %(name)s %(message)s",
        )
    log = logging.getLogger("active_monitor_amd.main")

    engine = None
    if args.backend == "http":
        from ..kube.config import get_config

        cfg = get_config(
            server=args.server, token=args.token,
            insecure=args.insecure_skip_tls_verify, kubeconfig=args.kubeconfig,
        )
        client = cfg.make_client()
        await client.start()
        try:
            await client.ping()
        except Exception as e:
            log.error("cannot reach apiserver %s: %s", cfg.server, e)
            await client.close()
            return 1
    else:
        client = MemoryClient(MemoryApiServer())
        if args.workflow_engine == "local":
            from ..workflow import LocalWorkflowEngine

            engine = LocalWorkflowEngine(client, args.namespace)
        if args.serve_api:
            from ..kube.server import ApiServerFrontend

            addr = parse_bind_address(args.serve_api)
            frontend = ApiServerFrontend(client.server, addr[0], addr[1])
            await frontend.start()
            log.info("serving Kubernetes REST API at %s", frontend.url)

    metrics_addr = parse_bind_address(args.metrics_bind_address)
    metrics_security = None
    if metrics_addr is not None and getattr(args, "metrics_secure", False):
        from ..engine.endpoints import build_metrics_security

        metrics_security = build_metrics_security(
            True,
            cert=getattr(args, "metrics_cert", None),
            key=getattr(args, "metrics_key", None),
            token_file=getattr(args, "metrics_auth_token_file", None),
        )

    manager = Manager(
        client,
        max_workers=args.max_workers,
        namespace=args.namespace,
        metrics_addr=metrics_addr,
        health_addr=parse_bind_address(args.health_probe_bind_address),
        leader_elect=args.leader_elect,
        shard_index=args.shard_index,
        shard_count=args.shard_count,
        shard_ha=getattr(args, "shard_ha", False),
        shard_lease_duration=getattr(args, "shard_lease_duration", 15.0),
        shard_renew_interval=getattr(args, "shard_renew_interval", 5.0),
        metrics_security=metrics_security,
    )

    stop = stop_event if stop_event is not None else asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        try:
            loop.add_signal_handler(sig, stop.set)
        except (NotImplementedError, RuntimeError):  # pragma: no cover - non-unix
            pass

    if engine is not None:
        await engine.start()
    await manager.start()
    log.info("starting manager: workers=%d backend=%s", args.max_workers, args.backend)
    # run until signalled — or until the manager reports a fatal condition
    # (lost leadership lease): the reference treats losing the lease as fatal
    # for the replica (cmd/main.go:87-88), so a deposed process exits nonzero
    # instead of reconciling without the lease.
    stop_w = asyncio.ensure_future(stop.wait())
    fatal_w = asyncio.ensure_future(manager.fatal.wait())
    done, pending = await asyncio.wait(
        {stop_w, fatal_w}, return_when=asyncio.FIRST_COMPLETED
    )
    for t in pending:
        t.cancel()
    rc = 0
    if fatal_w in done:
        log.error("manager fatal: %s", manager.fatal_reason)
        rc = 1
    log.info("shutting down")
    await manager.stop()
    if engine is not None:
        await engine.stop()
    if args.backend == "http":
        await client.close()
    if args.backend == "memory" and args.serve_api:
        await frontend.stop()
    return rc


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    try:
        return asyncio.run(run(args))
    except KeyboardInterrupt:
        return 0


if __name__ == "__main__":
    sys.exit(main())
