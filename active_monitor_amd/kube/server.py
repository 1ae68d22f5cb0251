"""HTTP apiserver frontend: serves the in-memory store over the Kubernetes
REST API.

This is the framework's envtest-style local apiserver: the HttpClient (and
kubectl-shaped tooling) can talk to a MemoryApiServer over real HTTP —
typed paths, status subresource, label selectors, resourceVersion watch
resume with real 410 Gone semantics, and streaming watches (JSON lines).

Implementation note: the server is a hand-rolled asyncio HTTP/1.1 protocol
(keep-alive, Content-Length framing; watch streams use Connection: close
framing) rather than an aiohttp web app — at fleet rates the apiserver
process is request-bound and the web-framework dispatch chain cost ~200µs
of the ~300µs per request. The wire format is unchanged; the full HTTP test
suite (CRUD, discovery, error paths, watch conformance, kubectl-ability)
runs against this server.
"""
from __future__ import annotations

import asyncio
import json
import logging
from typing import Optional, Tuple
from urllib.parse import parse_qsl, unquote, urlsplit

from .errors import ApiError, ExpiredError, InvalidError
from .memory import MemoryApiServer

log = logging.getLogger("active_monitor_amd.kube.server")

_REASONS = {
    200: "OK", 201: "Created", 400: "Bad Request", 404: "Not Found",
    405: "Method Not Allowed", 409: "Conflict", 410: "Gone",
    422: "Unprocessable Entity", 500: "Internal Server Error",
}


def _status_body(err: ApiError) -> dict:
    return {
        "kind": "Status",
        "apiVersion": "v1",
        "status": "Failure",
        "message": err.message,
        "reason": err.reason,
        "code": err.code,
    }


class ApiServerFrontend:
    def __init__(self, server: MemoryApiServer, host: str = "127.0.0.1", port: int = 0):
        self.server = server
        self.host = host
        self.port = port
        self._srv: Optional[asyncio.AbstractServer] = None
        self._live_subs: list = []

    @property
    def url(self) -> str:
        return f"http://{self.host}:{self.port}"

    async def start(self) -> None:
        self._srv = await asyncio.start_server(self._handle_conn, self.host, self.port)
        if self.port == 0:
            self.port = self._srv.sockets[0].getsockname()[1]

    async def stop(self) -> None:
        self.kick_watches()
        if self._srv is not None:
            self._srv.close()
            try:
                await asyncio.wait_for(self._srv.wait_closed(), 2.0)
            except asyncio.TimeoutError:
                pass

    def kick_watches(self) -> None:
        """Terminate every open watch stream (server stays up). Clients see a
        clean end-of-stream and reconnect with their resourceVersion — the
        fault-injection hook for RV-expiry/resume conformance tests."""
        for sub in list(self._live_subs):
            sub.close()

    # -- HTTP protocol ------------------------------------------------------

    async def _handle_conn(self, reader: asyncio.StreamReader,
                           writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                request_line = await reader.readline()
                if not request_line or request_line in (b"\r\n", b"\n"):
                    return
                try:
                    method, target, _ = request_line.decode("latin-1").split(" ", 2)
                except ValueError:
                    return
                content_length = 0
                content_type = ""
                expect_continue = False
                want_table = False
                while True:
                    line = await reader.readline()
                    if line in (b"\r\n", b"\n", b""):
                        break
                    lower = line.lower()
                    if lower.startswith(b"content-length:"):
                        content_length = int(line.split(b":", 1)[1])
                    elif lower.startswith(b"content-type:"):
                        content_type = line.split(b":", 1)[1].strip().decode("latin-1")
                    elif lower.startswith(b"expect:") and b"100-continue" in lower:
                        expect_continue = True  # curl sends this for big bodies
                    elif lower.startswith(b"accept:") and b"as=table" in lower:
                        want_table = True
                if expect_continue:
                    writer.write(b"HTTP/1.1 100 Continue\r\n\r\n")
                    await writer.drain()
                body = await reader.readexactly(content_length) if content_length else b""

                parts = urlsplit(target)
                path = unquote(parts.path)
                query = dict(parse_qsl(parts.query))

                if query.get("watch") in ("true", "1"):
                    await self._serve_watch(writer, path, query)
                    return  # watch streams own the connection
                status, obj = self._serve_unary(method, path, query, body,
                                                content_type)
                if (want_table and method == "GET" and status == 200
                        and isinstance(obj, dict)):
                    obj = self._to_table(obj)
                if isinstance(obj, str):  # health probes are plain text
                    payload, ctype = obj.encode(), "text/plain"
                else:
                    payload = json.dumps(obj, separators=(",", ":")).encode() if obj is not None else b""
                    ctype = "application/json"
                writer.write(
                    (
                        f"HTTP/1.1 {status} {_REASONS.get(status, 'OK')}\r\n"
                        f"Content-Type: {ctype}\r\n"
                        f"Content-Length: {len(payload)}\r\n\r\n"
                    ).encode("latin-1") + payload
                )
                await writer.drain()
        except (ConnectionError, asyncio.IncompleteReadError, asyncio.CancelledError):
            pass
        except Exception:  # defensive: never kill the accept loop
            log.exception("frontend connection handler failed")
        finally:
            try:
                writer.close()
            except Exception:
                pass

    # -- routing ------------------------------------------------------------

    def _split(self, path: str) -> Optional[Tuple[str, str]]:
        """Return (api_version, tail) for /api/v1/... or /apis/g/v/...; None
        for non-resource paths."""
        segs = [s for s in path.split("/") if s]
        if not segs:
            return None
        if segs[0] == "api" and len(segs) >= 2:
            return segs[1], "/".join(segs[2:])
        if segs[0] == "apis" and len(segs) >= 3:
            return f"{segs[1]}/{segs[2]}", "/".join(segs[3:])
        return None

    def _serve_unary(self, method: str, path: str, query: dict,
                     body: bytes, content_type: str = "") -> Tuple[int, Optional[dict]]:
        # discovery endpoints (enough for kubectl --server=<url>)
        if path in ("/api", "/api/"):
            return 200, {"kind": "APIVersions", "versions": ["v1"]}
        if path in ("/apis", "/apis/"):
            return 200, self._discovery_apis()
        if path in ("/version", "/version/"):
            from .. import __version__

            return 200, {
                "major": "1", "minor": "33",
                "gitVersion": f"v1.33.0-active-monitor-amd+{__version__}",
            }
        if path in ("/healthz", "/readyz", "/livez"):
            return 200, "ok"  # apiserver health probes (kubectl checks these)
        split = self._split(path)
        if split is None:
            return 404, _status_body(ApiError(f"the server could not find {path}"))
        api_version, tail = split
        if not tail:  # /api/v1 or /apis/g/v → resource discovery
            return 200, self._resource_list(api_version)
        return self._dispatch(method, api_version, tail, query, body, content_type)

    def _discovery_apis(self) -> dict:
        groups_map: dict = {}
        for (av, _), info in self.server.registry._by_kind.items():
            if "/" in av:
                g, v = av.split("/", 1)
                groups_map.setdefault(g, set()).add(v)
        groups = []
        for g, versions in sorted(groups_map.items()):
            gv = sorted({f"{g}/{v}" for v in versions})
            groups.append({
                "name": g,
                "versions": [{"groupVersion": x, "version": x.split("/", 1)[1]} for x in gv],
                "preferredVersion": {"groupVersion": gv[0], "version": gv[0].split("/", 1)[1]},
            })
        return {"kind": "APIGroupList", "apiVersion": "v1", "groups": groups}

    def _resource_list(self, api_version: str) -> dict:
        resources = []
        for (av, kind), info in sorted(self.server.registry._by_kind.items()):
            if av != api_version:
                continue
            resources.append({
                "name": info.plural,
                "singularName": kind.lower(),
                "namespaced": info.namespaced,
                "kind": kind,
                "verbs": ["create", "delete", "get", "list", "patch", "update", "watch"],
            })
            if (av, kind) == ("activemonitor.keikoproj.io/v1alpha1", "HealthCheck"):
                resources[-1]["shortNames"] = ["hc", "hcs"]
                resources.append({
                    "name": f"{info.plural}/status",
                    "singularName": "",
                    "namespaced": True,
                    "kind": kind,
                    "verbs": ["get", "update", "patch"],
                })
        return {"kind": "APIResourceList", "apiVersion": "v1",
                "groupVersion": api_version, "resources": resources}

    def _dispatch(self, method: str, api_version: str, tail: str, query: dict,
                  body: bytes, content_type: str = "") -> Tuple[int, Optional[dict]]:
        # tail forms:
        #   {plural}[/{name}[/status]]
        #   namespaces/{ns}/{plural}[/{name}[/status]]
        parts = [p for p in tail.split("/") if p]
        namespace = ""
        # /api/v1/namespaces/{ns}/{plural}... is namespaced access;
        # /api/v1/namespaces[/{name}] (≤2 segments) is the Namespace resource
        if len(parts) >= 3 and parts[0] == "namespaces":
            namespace = parts[1]
            parts = parts[2:]
        plural = parts[0] if parts else ""
        name = parts[1] if len(parts) > 1 else ""
        subresource = parts[2] if len(parts) > 2 else ""

        try:
            info = self.server.registry.by_plural(api_version, plural)
        except KeyError:
            return 404, _status_body(ApiError(f"unknown resource {plural}"))
        kind = info.kind

        try:
            if method == "GET" and not name:
                items = self.server.list(
                    api_version, kind, namespace or None, query.get("labelSelector"),
                    field_selector=query.get("fieldSelector"),
                )
                return 200, {
                    "apiVersion": api_version, "kind": kind + "List",
                    "metadata": {"resourceVersion": self.server.resource_version()},
                    "items": items,
                }
            if method == "GET":
                return 200, self.server.get(api_version, kind, namespace, name)
            if method == "POST":
                obj = self._parse_body(body)
                meta = obj.setdefault("metadata", {})
                if namespace and not meta.get("namespace"):
                    meta["namespace"] = namespace
                return 201, self.server.create(obj)
            if method == "PUT":
                obj = self._parse_body(body)
                if subresource == "status":
                    return 200, self.server.update_status(obj)
                return 200, self.server.update(obj)
            if method == "DELETE" and not name:
                # deletecollection (kubectl delete --all / -l selector)
                victims = self.server.list(
                    api_version, kind, namespace or None, query.get("labelSelector")
                )
                for obj in victims:
                    m = obj.get("metadata") or {}
                    try:
                        self.server.delete(api_version, kind,
                                           m.get("namespace", ""), m.get("name", ""))
                    except ApiError:
                        pass  # raced another delete / cascade GC
                return 200, {"apiVersion": api_version, "kind": kind + "List",
                             "items": victims}
            if method == "DELETE":
                self.server.delete(api_version, kind, namespace, name)
                return 200, {"kind": "Status", "status": "Success"}
            if method == "PATCH":
                ct = content_type.split(";")[0].strip().lower()
                if ct == "application/json-patch+json":
                    # RFC 6902 op lists are not supported; kubectl defaults
                    # to strategic/merge for CRDs anyway
                    return 415, _status_body(
                        InvalidError("json-patch is not supported; use merge-patch")
                    )
                if ct == "application/apply-patch+yaml":
                    import yaml as _yaml

                    try:
                        patch_obj = _yaml.safe_load(body)
                    except _yaml.YAMLError:
                        raise InvalidError("request body is not valid JSON")
                    if not isinstance(patch_obj, dict):
                        raise InvalidError("request body is not valid JSON")
                    upsert = True  # server-side apply creates when absent
                else:  # merge-patch / strategic-merge-patch
                    patch_obj = self._parse_body(body)
                    upsert = False
                return 200, self.server.patch(
                    api_version, kind, namespace, name, patch_obj,
                    subresource=subresource, upsert=upsert,
                )
        except InvalidError as e:
            code = 400 if e.message == "request body is not valid JSON" else e.code
            return code, _status_body(e)
        except ApiError as e:
            return e.code, _status_body(e)
        return 405, _status_body(ApiError(f"unsupported method {method}"))

    @staticmethod
    def _parse_body(body: bytes) -> dict:
        try:
            obj = json.loads(body)
            if not isinstance(obj, dict):
                raise ValueError
            return obj
        except (ValueError, TypeError):
            raise InvalidError("request body is not valid JSON")

    # -- Table responses (kubectl get) --------------------------------------

    #: printcolumns per kind, mirroring the CRD's additionalPrinterColumns
    #: (api/crd.py; reference healthcheck_types.go:71-76 printcolumn markers)
    _PRINTCOLUMNS = {
        "HealthCheck": [
            ("LATEST STATUS", "string", ("status", "status")),
            ("SUCCESS CNT  ", "string", ("status", "successCount")),
            ("FAIL CNT", "string", ("status", "failedCount")),
            ("REMEDY SUCCESS CNT  ", "string", ("status", "remedySuccessCount")),
            ("REMEDY FAIL CNT", "string", ("status", "remedyFailedCount")),
            ("Age", "date", ("metadata", "creationTimestamp")),
        ],
        "Workflow": [
            ("Status", "string", ("status", "phase")),
            ("Age", "date", ("metadata", "creationTimestamp")),
        ],
    }
    _DEFAULT_COLUMNS = [("Age", "date", ("metadata", "creationTimestamp"))]

    def _to_table(self, obj: dict) -> dict:
        """meta.k8s.io/v1 Table transform — what kubectl get requests via
        Accept: ...;as=Table. Rows carry PartialObjectMetadata objects."""
        if obj.get("kind", "").endswith("List"):
            items = obj.get("items", [])
            kind = obj.get("kind", "")[:-4]
            meta = obj.get("metadata", {})
        elif obj.get("kind") and obj.get("metadata") is not None:
            items = [obj]
            kind = obj.get("kind", "")
            meta = {}
        else:
            return obj  # Status bodies etc. pass through
        cols = self._PRINTCOLUMNS.get(kind, self._DEFAULT_COLUMNS)
        defs = [{"name": "Name", "type": "string", "format": "name"}] + [
            {"name": n, "type": t} for (n, t, _) in cols
        ]
        rows = []
        for it in items:
            cells = [((it.get("metadata") or {}).get("name", ""))]
            for _, _, path in cols:
                cur = it
                for seg in path:
                    cur = cur.get(seg) if isinstance(cur, dict) else None
                    if cur is None:
                        break
                cells.append("" if cur is None else cur)
            rows.append({
                "cells": cells,
                "object": {
                    "kind": "PartialObjectMetadata",
                    "apiVersion": "meta.k8s.io/v1",
                    "metadata": it.get("metadata", {}),
                },
            })
        return {
            "kind": "Table",
            "apiVersion": "meta.k8s.io/v1",
            "metadata": meta,
            "columnDefinitions": defs,
            "rows": rows,
        }

    # -- watch --------------------------------------------------------------

    @staticmethod
    def _ev_rv(ev: dict) -> int:
        try:
            return int(((ev.get("object") or {}).get("metadata") or {})
                       .get("resourceVersion", 0))
        except (TypeError, ValueError):
            return 0

    async def _serve_watch(self, writer: asyncio.StreamWriter, path: str,
                           query: dict) -> None:
        """Watch with real resourceVersion semantics: a resume rv replays the
        retained event history after that rv (410 Gone when it predates the
        window — the client must re-list); no rv replays current state as
        ADDED. The live subscription is opened before the replay snapshot is
        taken and overlap is deduplicated by rv, so no event can fall between
        replay and stream. Framing: Connection: close (read-until-EOF)."""
        split = self._split(path)
        err = None
        if split is None:
            err = ApiError(f"the server could not find {path}")
        else:
            api_version, tail = split
            parts = [p for p in tail.split("/") if p]
            namespace = None
            if len(parts) >= 2 and parts[0] == "namespaces":
                namespace = parts[1]
                parts = parts[2:]
            try:
                info = self.server.registry.by_plural(api_version, parts[0] if parts else "")
            except KeyError:
                err = ApiError(f"unknown resource {tail}")
        if err is not None:
            payload = json.dumps(_status_body(err), separators=(",", ":")).encode()
            writer.write(
                (f"HTTP/1.1 404 Not Found\r\nContent-Type: application/json\r\n"
                 f"Content-Length: {len(payload)}\r\n\r\n").encode() + payload
            )
            await writer.drain()
            return
        kind = info.kind

        rv_param = query.get("resourceVersion")
        selector = {}
        fields = []
        if query.get("labelSelector"):
            from .memory import parse_label_selector

            try:
                selector = parse_label_selector(query["labelSelector"])
            except ApiError:
                selector = {}
        if query.get("fieldSelector"):
            from .memory import parse_field_selector

            try:
                fields = parse_field_selector(query["fieldSelector"])
            except ApiError:
                fields = []
        sub = self.server.watch(api_version, kind, namespace)
        self._live_subs.append(sub)
        try:
            replay = []
            last_rv = 0
            if rv_param:
                try:
                    replay = self.server.events_since(api_version, kind, namespace, rv_param)
                except ExpiredError as e:
                    payload = json.dumps(_status_body(e), separators=(",", ":")).encode()
                    writer.write(
                        (f"HTTP/1.1 410 Gone\r\nContent-Type: application/json\r\n"
                         f"Content-Length: {len(payload)}\r\n\r\n").encode() + payload
                    )
                    await writer.drain()
                    return
                try:
                    last_rv = int(rv_param)
                except ValueError:
                    last_rv = 0
            else:
                last_rv = int(self.server.resource_version())
                replay = [
                    {"type": "ADDED", "object": obj}
                    for obj in self.server.list(api_version, kind, namespace)
                ]
            writer.write(
                b"HTTP/1.1 200 OK\r\n"
                b"Content-Type: application/json;stream=watch\r\n"
                b"Connection: close\r\n\r\n"
            )
            from .memory import _fields_match, _labels_match

            def _matches(obj: dict) -> bool:
                if selector and not _labels_match(obj, selector):
                    return False
                if fields and not _fields_match(obj, fields):
                    return False
                return True

            for ev in replay:
                last_rv = max(last_rv, self._ev_rv(ev))
                if not _matches(ev.get("object") or {}):
                    continue
                writer.write((json.dumps(ev, separators=(",", ":")) + "\n").encode())
            await writer.drain()
            # watch budget: like a real apiserver, an expiring watch ends
            # with a clean stream close and the client resumes from its rv
            # (kubectl sends timeoutSeconds by default; reflectors resume).
            # Implemented as one timer that closes the subscription — the
            # event loop stays free of per-event wait_for wrappers.
            budget_handle = None
            if query.get("timeoutSeconds"):
                try:
                    budget_handle = asyncio.get_running_loop().call_later(
                        float(query["timeoutSeconds"]), sub.close
                    )
                except ValueError:
                    pass
            try:
                async for ev in sub:
                    if self._ev_rv(ev) <= last_rv:
                        continue  # already covered by the replay snapshot
                    if not _matches(ev.get("object") or {}):
                        continue
                    writer.write((json.dumps(ev, separators=(",", ":")) + "\n").encode())
                    await writer.drain()
            finally:
                if budget_handle is not None:
                    budget_handle.cancel()
        except (ConnectionError, asyncio.CancelledError):
            pass
        finally:
            sub.close()
            if sub in self._live_subs:
                self._live_subs.remove(sub)
