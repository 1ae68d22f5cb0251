"""HTTP apiserver frontend: serves the in-memory store over the Kubernetes
REST API.

This is the framework's envtest-style local apiserver: the HttpClient (and
kubectl-shaped tooling) can talk to a MemoryApiServer over real HTTP —
typed paths, status subresource, label selectors, and streaming watches
(JSON-lines, resourceVersion resume semantics are best-effort: a reconnect
replays current state as ADDED events, which the informer path tolerates).
"""
from __future__ import annotations

import asyncio
import json
import logging
from typing import Optional

from aiohttp import web

from .errors import ApiError, ExpiredError, InvalidError
from .memory import MemoryApiServer

log = logging.getLogger("active_monitor_amd.kube.server")


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
        self._runner: Optional[web.AppRunner] = None
        self._live_subs: list = []

    @property
    def url(self) -> str:
        return f"http://{self.host}:{self.port}"

    async def start(self) -> None:
        app = web.Application()
        # discovery endpoints (enough for kubectl --server=<url>)
        app.router.add_get("/api", self._discovery_api)
        app.router.add_get("/apis", self._discovery_apis)
        app.router.add_get("/version", self._discovery_version)
        app.router.add_get("/api/{version}", self._discovery_core_resources)
        app.router.add_get("/apis/{group}/{version}", self._discovery_group_resources)
        app.router.add_route("*", "/api/{version}/{tail:.*}", self._handle_core)
        app.router.add_route("*", "/apis/{group}/{version}/{tail:.*}", self._handle_group)
        self._runner = web.AppRunner(app, access_log=None, shutdown_timeout=1.0)
        await self._runner.setup()
        site = web.TCPSite(self._runner, self.host, self.port)
        await site.start()
        if self.port == 0:
            self.port = site._server.sockets[0].getsockname()[1]

    async def stop(self) -> None:
        self.kick_watches()
        if self._runner is not None:
            await self._runner.cleanup()

    def kick_watches(self) -> None:
        """Terminate every open watch stream (server stays up). Clients see a
        clean end-of-stream and reconnect with their resourceVersion — the
        fault-injection hook for RV-expiry/resume conformance tests."""
        for sub in list(self._live_subs):
            sub.close()  # end open watch streams so cleanup() is immediate

    # -- discovery ----------------------------------------------------------

    def _groups(self):
        groups = {}
        for (av, _), info in self.server.registry._by_kind.items():
            if "/" in av:
                g, v = av.split("/", 1)
                groups.setdefault(g, set()).add(v)
        return groups

    async def _discovery_api(self, request: web.Request) -> web.Response:
        return web.json_response({"kind": "APIVersions", "versions": ["v1"]})

    async def _discovery_apis(self, request: web.Request) -> web.Response:
        groups = []
        for g, versions in sorted(self._groups().items()):
            gv = sorted({f"{g}/{v}" for v in versions})
            groups.append({
                "name": g,
                "versions": [{"groupVersion": x, "version": x.split("/", 1)[1]} for x in gv],
                "preferredVersion": {"groupVersion": gv[0], "version": gv[0].split("/", 1)[1]},
            })
        return web.json_response({"kind": "APIGroupList", "apiVersion": "v1", "groups": groups})

    async def _discovery_version(self, request: web.Request) -> web.Response:
        from .. import __version__

        return web.json_response({
            "major": "1", "minor": "33",
            "gitVersion": f"v1.33.0-active-monitor-amd+{__version__}",
        })

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

    async def _discovery_core_resources(self, request: web.Request) -> web.Response:
        return web.json_response(self._resource_list(request.match_info["version"]))

    async def _discovery_group_resources(self, request: web.Request) -> web.Response:
        gv = f'{request.match_info["group"]}/{request.match_info["version"]}'
        return web.json_response(self._resource_list(gv))

    # -- routing -----------------------------------------------------------

    async def _handle_core(self, request: web.Request) -> web.StreamResponse:
        version = request.match_info["version"]
        return await self._dispatch(request, version, request.match_info["tail"])

    async def _handle_group(self, request: web.Request) -> web.StreamResponse:
        api_version = f'{request.match_info["group"]}/{request.match_info["version"]}'
        return await self._dispatch(request, api_version, request.match_info["tail"])

    async def _dispatch(self, request: web.Request, api_version: str, tail: str) -> web.StreamResponse:
        # tail forms:
        #   {plural}
        #   {plural}/{name}
        #   {plural}/{name}/status
        #   namespaces/{ns}/{plural}
        #   namespaces/{ns}/{plural}/{name}
        #   namespaces/{ns}/{plural}/{name}/status
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
            return web.json_response(
                _status_body(ApiError(f"unknown resource {plural}")), status=404
            )
        kind = info.kind

        try:
            if request.method == "GET" and not name:
                if request.query.get("watch") in ("true", "1"):
                    return await self._watch(request, api_version, kind, namespace or None)
                items = self.server.list(
                    api_version, kind, namespace or None,
                    request.query.get("labelSelector"),
                )
                return web.json_response({
                    "apiVersion": api_version, "kind": kind + "List",
                    "metadata": {"resourceVersion": self.server.resource_version()},
                    "items": items,
                })
            if request.method == "GET":
                return web.json_response(self.server.get(api_version, kind, namespace, name))
            if request.method == "POST":
                try:
                    obj = await request.json()
                except Exception:
                    return web.json_response(
                        _status_body(InvalidError("request body is not valid JSON")),
                        status=400,
                    )
                meta = obj.setdefault("metadata", {})
                if namespace and not meta.get("namespace"):
                    meta["namespace"] = namespace
                return web.json_response(self.server.create(obj), status=201)
            if request.method == "PUT":
                try:
                    body = await request.json()
                except Exception:
                    return web.json_response(
                        _status_body(InvalidError("request body is not valid JSON")),
                        status=400,
                    )
                if subresource == "status":
                    return web.json_response(self.server.update_status(body))
                return web.json_response(self.server.update(body))
            if request.method == "DELETE":
                self.server.delete(api_version, kind, namespace, name)
                return web.json_response({"kind": "Status", "status": "Success"})
        except ApiError as e:
            return web.json_response(_status_body(e), status=e.code)
        return web.json_response(
            _status_body(ApiError(f"unsupported method {request.method}")), status=405
        )

    @staticmethod
    def _ev_rv(ev: dict) -> int:
        try:
            return int(((ev.get("object") or {}).get("metadata") or {})
                       .get("resourceVersion", 0))
        except (TypeError, ValueError):
            return 0

    async def _watch(self, request: web.Request, api_version: str, kind: str,
                     namespace: Optional[str]) -> web.StreamResponse:
        """Watch with real resourceVersion semantics: a resume rv replays the
        retained event history after that rv (410 Gone when it predates the
        window — the client must re-list, like against a real apiserver); no
        rv replays current state as ADDED. The live subscription is opened
        before the replay snapshot is taken and overlap is deduplicated by
        rv, so no event can fall between replay and stream."""
        rv_param = request.query.get("resourceVersion")
        sub = self.server.watch(api_version, kind, namespace)
        self._live_subs.append(sub)
        try:
            replay = []
            last_rv = 0
            if rv_param:
                try:
                    replay = self.server.events_since(api_version, kind, namespace, rv_param)
                except ExpiredError as e:
                    sub.close()
                    self._live_subs.remove(sub)
                    return web.json_response(_status_body(e), status=410)
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
            resp = web.StreamResponse(
                status=200, headers={"Content-Type": "application/json;stream=watch"}
            )
            await resp.prepare(request)
            for ev in replay:
                last_rv = max(last_rv, self._ev_rv(ev))
                await resp.write((json.dumps(ev) + "\n").encode())
            async for ev in sub:
                if self._ev_rv(ev) <= last_rv:
                    continue  # already covered by the replay snapshot
                await resp.write((json.dumps(ev) + "\n").encode())
        except (ConnectionResetError, asyncio.CancelledError):
            pass
        finally:
            sub.close()
            if sub in self._live_subs:
                self._live_subs.remove(sub)
        return resp
