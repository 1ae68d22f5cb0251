"""Standalone local apiserver process.

Runs a MemoryApiServer behind the Kubernetes-REST ApiServerFrontend in its
own process, optionally with an in-"cluster" workflow engine playing the
Argo workflow-controller role (completing submitted Workflows the way the
real controller at deploy/deploy-argo.yaml would).

Two consumers:

- ``bench.py``'s default real-wire regime: the controller under test runs in
  a *different* process and reaches this one over 127.0.0.1 HTTP, so the
  measured reconcile path crosses the same process/serialization/TCP boundary
  the reference crosses to kube-apiserver — no shared event loop, no shared
  memory, no zero-RTT self-play.
- a kubectl-able local dev apiserver (``python -m active_monitor_amd.kube.standalone``).

Prints one ``READY {json}`` line on stdout once serving; runs until
SIGTERM/SIGINT.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import signal
import sys

from .memory import MemoryApiServer
from .server import ApiServerFrontend


def bench_policy(remedy_frac: float):
    """The bench fleet's workflow outcomes, decodable from workflow names
    (bench.py make_cr): CR ``hc-NNNNN`` fails its checks iff NNNNN%100 <
    remedy_frac*100; remedy workflows always succeed."""
    cut = remedy_frac * 100.0

    def policy(wf):
        name = (wf.get("metadata") or {}).get("name", "")
        if "-remedy-wf-" in name:
            return ("Succeeded", "")
        base = name.split("-wf-")[0]
        parts = base.split("-")
        if len(parts) == 2 and parts[0] == "hc" and parts[1].isdigit():
            if int(parts[1]) % 100 < cut:
                return ("Failed", "synthetic failure")
        return ("Succeeded", "")

    return policy


def parse_args(argv=None):
    p = argparse.ArgumentParser(prog="active-monitor-apiserver")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=0, help="0 = ephemeral")
    p.add_argument("--engine", choices=["scripted-bench", "local", "none"],
                   default="none",
                   help="in-cluster workflow engine: scripted-bench completes "
                        "workflows per the bench policy, local executes them "
                        "as subprocesses, none leaves them pending")
    p.add_argument("--remedy-frac", type=float, default=0.2,
                   help="scripted-bench: fraction of CRs whose checks fail")
    p.add_argument("--engine-delay", type=float, default=0.0,
                   help="scripted-bench: simulated workflow runtime (s)")
    p.add_argument("--engine-ttl", type=float, default=60.0,
                   help="completed-workflow TTL seconds (Argo ttlStrategy "
                        "equivalent; bounds server memory at fleet rates)")
    return p.parse_args(argv)


async def amain(args) -> int:
    from .client import MemoryClient

    server = MemoryApiServer()
    frontend = ApiServerFrontend(server, args.host, args.port)
    await frontend.start()

    engine = None
    if args.engine == "scripted-bench":
        from ..workflow import ScriptedWorkflowEngine

        engine = ScriptedWorkflowEngine(
            MemoryClient(server),
            policy=bench_policy(args.remedy_frac),
            delay=args.engine_delay,
            ttl_seconds=args.engine_ttl,
        )
    elif args.engine == "local":
        from ..workflow import LocalWorkflowEngine

        engine = LocalWorkflowEngine(MemoryClient(server), ttl_seconds=args.engine_ttl)
    if engine is not None:
        await engine.start()

    print("READY " + json.dumps({"url": frontend.url, "port": frontend.port}),
          flush=True)

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        try:
            loop.add_signal_handler(sig, stop.set)
        except (NotImplementedError, RuntimeError):  # pragma: no cover
            pass
    await stop.wait()

    if engine is not None:
        await engine.stop()
    await frontend.stop()
    return 0


def main(argv=None) -> int:
    return asyncio.run(amain(parse_args(argv)))


if __name__ == "__main__":
    sys.exit(main())
