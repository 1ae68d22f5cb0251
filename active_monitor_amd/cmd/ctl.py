"""amctl — operator CLI for the HealthCheck API.

The reference's operator UX is kubectl against the CRD (shortnames hc/hcs,
printcolumns — healthcheck_types.go:68-76); amctl provides the same surface
against any server speaking the Kubernetes REST API, including this
framework's standalone ``--serve-api`` endpoint:

    amctl --server http://127.0.0.1:8001 get hc -n health
    amctl --server ... describe hc inline-hello -n health
    amctl --server ... apply -f examples/inline-hello.yaml
    amctl --server ... delete hc inline-hello -n health
"""
from __future__ import annotations

import argparse
import asyncio
import sys
from datetime import datetime, timezone
from typing import Any, Dict, List, Optional

import yaml

from .. import API_VERSION
from ..api.types import parse_k8s_time
from ..kube.errors import AlreadyExistsError, ConflictError, NotFoundError
from ..kube.http import HttpClient
from ..kube.registry import WF_API_VERSION

ALIASES = {
    "hc": (API_VERSION, "HealthCheck"),
    "hcs": (API_VERSION, "HealthCheck"),
    "healthcheck": (API_VERSION, "HealthCheck"),
    "healthchecks": (API_VERSION, "HealthCheck"),
    "wf": (WF_API_VERSION, "Workflow"),
    "workflow": (WF_API_VERSION, "Workflow"),
    "workflows": (WF_API_VERSION, "Workflow"),
    "events": ("v1", "Event"),
    "event": ("v1", "Event"),
}


def _age(creation: Optional[str]) -> str:
    t = parse_k8s_time(creation)
    if t is None:
        return "<unknown>"
    delta = datetime.now(timezone.utc) - t
    secs = int(delta.total_seconds())
    if secs < 120:
        return f"{secs}s"
    if secs < 7200:
        return f"{secs // 60}m"
    if secs < 172800:
        return f"{secs // 3600}h"
    return f"{secs // 86400}d"


def _table(headers: List[str], rows: List[List[str]]) -> str:
    widths = [max(len(h), *(len(r[i]) for r in rows)) if rows else len(h)
              for i, h in enumerate(headers)]
    out = ["   ".join(h.ljust(w) for h, w in zip(headers, widths)).rstrip()]
    for r in rows:
        out.append("   ".join(c.ljust(w) for c, w in zip(r, widths)).rstrip())
    return "\n".join(out)


def _hc_row(o: Dict[str, Any]) -> List[str]:
    st = o.get("status") or {}
    return [
        o["metadata"]["name"],
        st.get("status", "") or "<none>",
        str(st.get("successCount", 0)),
        str(st.get("failedCount", 0)),
        str(st.get("remedySuccessCount", 0)),
        str(st.get("remedyFailedCount", 0)),
        _age(o["metadata"].get("creationTimestamp")),
    ]


async def cmd_get(client: HttpClient, args) -> int:
    av, kind = ALIASES[args.resource]
    if args.name:
        objs = [await client.get(av, kind, args.namespace, args.name)]
    else:
        objs = await client.list(av, kind, args.namespace or None)
    if args.output == "yaml":
        print(yaml.safe_dump_all(objs, sort_keys=False).rstrip())
        return 0
    if kind == "HealthCheck":
        # the CRD's printcolumns (healthcheck_types.go:71-76)
        print(_table(
            ["NAME", "LATEST STATUS", "SUCCESS CNT", "FAIL CNT",
             "REMEDY SUCCESS CNT", "REMEDY FAIL CNT", "AGE"],
            [_hc_row(o) for o in objs],
        ))
    elif kind == "Workflow":
        print(_table(
            ["NAME", "STATUS", "AGE"],
            [[o["metadata"]["name"],
              (o.get("status") or {}).get("phase", "") or "<pending>",
              _age(o["metadata"].get("creationTimestamp"))] for o in objs],
        ))
    else:
        print(_table(
            ["NAME", "TYPE", "REASON", "MESSAGE"],
            [[o["metadata"]["name"], o.get("type", ""), o.get("reason", ""),
              (o.get("message", "") or "")[:80]] for o in objs],
        ))
    return 0


async def cmd_describe(client: HttpClient, args) -> int:
    av, kind = ALIASES[args.resource]
    o = await client.get(av, kind, args.namespace, args.name)
    print(yaml.safe_dump(o, sort_keys=False).rstrip())
    return 0


async def cmd_apply(client: HttpClient, args) -> int:
    with open(args.filename) as f:
        docs = [d for d in yaml.safe_load_all(f) if d]
    for doc in docs:
        meta = doc.get("metadata") or {}
        name = meta.get("name", "")
        try:
            await client.create(doc)
            print(f'{doc.get("kind", "object").lower()}/{name} created')
        except (AlreadyExistsError, ConflictError):
            # exists: update with a fresh resourceVersion
            current = await client.get(
                doc.get("apiVersion", ""), doc.get("kind", ""),
                meta.get("namespace", ""), name,
            )
            doc.setdefault("metadata", {})["resourceVersion"] = (
                current["metadata"]["resourceVersion"]
            )
            await client.update(doc)
            print(f'{doc.get("kind", "object").lower()}/{name} configured')
    return 0


async def cmd_delete(client: HttpClient, args) -> int:
    av, kind = ALIASES[args.resource]
    await client.delete(av, kind, args.namespace, args.name)
    print(f"{args.resource}/{args.name} deleted")
    return 0


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="amctl", description=__doc__.splitlines()[0])
    p.add_argument("--server", required=True, help="apiserver URL")
    p.add_argument("--token", default="")
    p.add_argument("--insecure-skip-tls-verify", action="store_true")
    sub = p.add_subparsers(dest="command", required=True)

    g = sub.add_parser("get")
    g.add_argument("resource", choices=sorted(ALIASES))
    g.add_argument("name", nargs="?", default="")
    g.add_argument("-n", "--namespace", default="health")
    g.add_argument("-o", "--output", choices=["table", "yaml"], default="table")

    d = sub.add_parser("describe")
    d.add_argument("resource", choices=sorted(ALIASES))
    d.add_argument("name")
    d.add_argument("-n", "--namespace", default="health")

    a = sub.add_parser("apply")
    a.add_argument("-f", "--filename", required=True)

    rm = sub.add_parser("delete")
    rm.add_argument("resource", choices=sorted(ALIASES))
    rm.add_argument("name")
    rm.add_argument("-n", "--namespace", default="health")
    return p


async def run(args) -> int:
    client = HttpClient(args.server, token=args.token or None,
                        verify=not args.insecure_skip_tls_verify)
    await client.start()
    try:
        handler = {"get": cmd_get, "describe": cmd_describe,
                   "apply": cmd_apply, "delete": cmd_delete}[args.command]
        return await handler(client, args)
    except NotFoundError as e:
        print(f"Error: {e}", file=sys.stderr)
        return 1
    finally:
        await client.close()


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    return asyncio.run(run(args))


if __name__ == "__main__":
    sys.exit(main())
