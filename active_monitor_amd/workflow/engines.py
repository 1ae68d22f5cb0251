"""Workflow execution engines.

The reference delegates execution to the external Argo Workflows controller
(deploy/deploy-argo.yaml; the controller only creates Workflow CRs and polls
``status.phase``, healthcheck_controller.go:525,617-624). This module provides
that role in-process:

- :class:`ScriptedWorkflowEngine` — drives submitted Workflow CRs to a phase
  decided by a policy callable. This is the integration-test linchpin: with no
  engine at all, workflows never reach a terminal phase and the IEB timeout
  forces the Failed path (the reference's envtest trick); with a
  policy, the success path — which the reference's envtest never reaches — is
  exercised too.
- :class:`LocalWorkflowEngine` — actually EXECUTES Argo-shaped workflows as
  local subprocesses (container/script templates, ``steps`` groups,
  ``retryStrategy.limit``, ``activeDeadlineSeconds``, output parameters), so
  the framework can run health checks standalone without a cluster.
"""
from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Any, Callable, Dict, List, Optional, Tuple

from ..kube.client import KubeClient
from ..kube.errors import ConflictError, NotFoundError
from ..kube.registry import WF_API_VERSION, WF_KIND

log = logging.getLogger("active_monitor_amd.workflow")

Phase = str  # "Running" | "Succeeded" | "Failed"

PolicyResult = Optional[Tuple[Phase, str]]
Policy = Callable[[Dict[str, Any]], PolicyResult]


def always_succeed(wf: Dict[str, Any]) -> PolicyResult:
    return ("Succeeded", "")


def always_fail(wf: Dict[str, Any]) -> PolicyResult:
    return ("Failed", "workflow failed")


def never_complete(wf: Dict[str, Any]) -> PolicyResult:
    return None


class _EngineBase:
    """``ttl_seconds`` deletes completed workflows after the given delay —
    the role Argo's ``ttlStrategy.secondsAfterCompletion`` plays for the
    reference (its install configures 1800s, deploy-argo.yaml:1162-1173);
    None disables (tests that inspect completed workflows)."""

    DEFAULT_TTL = 1800.0

    def __init__(self, client: KubeClient, namespace: Optional[str] = None,
                 ttl_seconds: Optional[float] = DEFAULT_TTL):
        self.client = client
        self.namespace = namespace
        self.ttl_seconds = ttl_seconds
        self._task: Optional[asyncio.Task] = None
        self._sub = None
        self._inflight: Dict[str, asyncio.Task] = {}
        from collections import deque

        self._ttl_handles: "deque" = deque()

    async def start(self) -> None:
        self._sub = self.client.watch(WF_API_VERSION, WF_KIND, self.namespace)
        self._task = asyncio.ensure_future(self._loop())

    async def stop(self) -> None:
        if self._sub is not None:
            self._sub.close()
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
        for t in self._inflight.values():
            t.cancel()
        for h in self._ttl_handles:
            h.cancel()
        self._ttl_handles.clear()

    async def _loop(self) -> None:
        async for ev in self._sub:
            if ev["type"] != "ADDED":
                continue
            wf = ev["object"]
            if (wf.get("status") or {}).get("phase") in ("Succeeded", "Failed"):
                continue
            key = f'{(wf["metadata"].get("namespace", ""))}/{wf["metadata"]["name"]}'
            task = asyncio.ensure_future(self._run(wf))
            self._inflight[key] = task
            task.add_done_callback(lambda t, k=key: self._inflight.pop(k, None))

    async def _run(self, wf: Dict[str, Any]) -> None:  # pragma: no cover - abstract
        raise NotImplementedError

    def _schedule_ttl(self, wf: Dict[str, Any]) -> None:
        if self.ttl_seconds is None:
            return
        meta = wf["metadata"]
        ns, name = meta.get("namespace", ""), meta["name"]

        def _gc() -> None:
            async def delete() -> None:
                try:
                    await self.client.delete(WF_API_VERSION, WF_KIND, ns, name)
                except Exception:
                    pass

            task = asyncio.ensure_future(delete())
            self._inflight[f"ttl:{ns}/{name}"] = task
            task.add_done_callback(
                lambda t: self._inflight.pop(f"ttl:{ns}/{name}", None)
            )

        loop = asyncio.get_event_loop()
        self._ttl_handles.append(loop.call_later(self.ttl_seconds, _gc))
        # handles are appended in firing order (constant ttl): drop spent
        # ones from the front — O(1) amortized
        now = loop.time()
        while self._ttl_handles and (
            self._ttl_handles[0].cancelled() or self._ttl_handles[0].when() <= now
        ):
            self._ttl_handles.popleft()

    async def _set_status(self, wf: Dict[str, Any], status: Dict[str, Any]) -> None:
        """Write the Workflow status the way the Argo controller does (the
        Workflow CRD has no status subresource — a plain update)."""
        meta = wf["metadata"]
        for _ in range(5):
            try:
                fresh = await self.client.get(
                    WF_API_VERSION, WF_KIND, meta.get("namespace", ""), meta["name"]
                )
            except NotFoundError:
                return
            fresh["status"] = status
            try:
                await self.client.update(fresh)
                return
            except ConflictError:
                await asyncio.sleep(0.005)


class ScriptedWorkflowEngine(_EngineBase):
    def __init__(
        self,
        client: KubeClient,
        policy: Policy = always_succeed,
        delay: float = 0.0,
        namespace: Optional[str] = None,
        ttl_seconds: Optional[float] = _EngineBase.DEFAULT_TTL,
    ):
        super().__init__(client, namespace, ttl_seconds)
        self.policy = policy
        self.delay = delay
        self.completed = 0

    async def _run(self, wf: Dict[str, Any]) -> None:
        decision = self.policy(wf)
        if decision is None:
            return  # leave pending: the controller's IEB timeout takes over
        if self.delay > 0:
            # instant completions skip the intermediate Running write — one
            # status update (and one spurious watcher wakeup) less per run
            await self._set_status(wf, {"phase": "Running", "startedAt": _now_iso()})
            await asyncio.sleep(self.delay)
        phase, message = decision[0], decision[1]
        status: Dict[str, Any] = {"phase": phase, "finishedAt": _now_iso()}
        if message:
            status["message"] = message
        if len(decision) > 2 and decision[2]:  # type: ignore[misc]
            status["outputs"] = decision[2]  # type: ignore[misc]
        await self._set_status(wf, status)
        self._schedule_ttl(wf)
        self.completed += 1


def _now_iso() -> str:
    from ..api.types import k8s_now

    return k8s_now()


class LocalWorkflowEngine(_EngineBase):
    """Executes Argo-shaped workflows locally.

    Supported template surface (enough to run every example shipped with the
    framework): ``container`` (command+args as a local subprocess; ``image``
    is informational without a container runtime), ``script`` (source piped to
    the command interpreter), ``steps`` (sequential groups of parallel steps),
    ``retryStrategy.limit``, workflow-level ``activeDeadlineSeconds``, and
    ``outputs.parameters`` with ``globalName`` surfacing into
    ``status.outputs.parameters`` (the custom-metrics contract,
    README.md:275-285).
    """

    def __init__(self, client: KubeClient, namespace: Optional[str] = None,
                 ttl_seconds: Optional[float] = _EngineBase.DEFAULT_TTL):
        super().__init__(client, namespace, ttl_seconds)
        self.completed = 0

    async def _run(self, wf: Dict[str, Any]) -> None:
        spec = wf.get("spec") or {}
        await self._set_status(wf, {"phase": "Running", "startedAt": _now_iso()})
        deadline = spec.get("activeDeadlineSeconds")
        try:
            if deadline:
                phase, message, outputs = await asyncio.wait_for(
                    self._run_spec(spec), float(deadline)
                )
            else:
                phase, message, outputs = await self._run_spec(spec)
        except asyncio.TimeoutError:
            phase, message, outputs = "Failed", "deadline exceeded", []
        except asyncio.CancelledError:
            raise
        except Exception as e:
            phase, message, outputs = "Failed", str(e), []
        status: Dict[str, Any] = {"phase": phase, "finishedAt": _now_iso()}
        if message:
            status["message"] = message
        if outputs:
            status["outputs"] = {"parameters": outputs}
        await self._set_status(wf, status)
        self._schedule_ttl(wf)
        self.completed += 1

    async def _run_spec(self, spec: Dict[str, Any]) -> Tuple[Phase, str, List[Dict[str, Any]]]:
        templates = {t.get("name"): t for t in spec.get("templates", []) or []}
        entry = spec.get("entrypoint")
        if entry not in templates:
            return "Failed", f"entrypoint {entry!r} not found", []
        outputs: List[Dict[str, Any]] = []
        ok, message = await self._run_template(templates, templates[entry], outputs)
        return ("Succeeded" if ok else "Failed"), message, outputs

    async def _run_template(
        self,
        templates: Dict[str, Dict[str, Any]],
        tmpl: Dict[str, Any],
        outputs: List[Dict[str, Any]],
    ) -> Tuple[bool, str]:
        retry_limit = int(((tmpl.get("retryStrategy") or {}).get("limit")) or 0)
        attempts = retry_limit + 1
        last_msg = ""
        for attempt in range(attempts):
            ok, last_msg = await self._run_template_once(templates, tmpl, outputs)
            if ok:
                return True, ""
        return False, last_msg

    async def _run_template_once(
        self,
        templates: Dict[str, Dict[str, Any]],
        tmpl: Dict[str, Any],
        outputs: List[Dict[str, Any]],
    ) -> Tuple[bool, str]:
        if "steps" in tmpl:
            for group in tmpl["steps"] or []:
                steps = group if isinstance(group, list) else [group]
                results = await asyncio.gather(
                    *[
                        self._run_template(
                            templates, templates.get(s.get("template"), {}), outputs
                        )
                        for s in steps
                        if isinstance(s, dict)
                    ]
                )
                for ok, msg in results:
                    if not ok:
                        return False, msg
            return True, ""
        if "container" in tmpl:
            c = tmpl["container"] or {}
            cmd = list(c.get("command", []) or []) + list(c.get("args", []) or [])
            if not cmd:
                return False, "container template has no command"
            ok, msg = await self._exec(cmd)
            if ok:
                self._collect_outputs(tmpl, outputs)
            return ok, msg
        if "script" in tmpl:
            s = tmpl["script"] or {}
            cmd = list(s.get("command", []) or ["sh"])
            source = s.get("source", "")
            ok, msg = await self._exec(cmd, stdin=source.encode())
            if ok:
                self._collect_outputs(tmpl, outputs)
            return ok, msg
        if "suspend" in tmpl:
            dur = tmpl["suspend"] or {}
            await asyncio.sleep(float(dur.get("duration", 0) or 0))
            return True, ""
        return False, "unsupported template type"

    async def _exec(self, cmd: List[str], stdin: Optional[bytes] = None) -> Tuple[bool, str]:
        try:
            proc = await asyncio.create_subprocess_exec(
                *[str(x) for x in cmd],
                stdin=asyncio.subprocess.PIPE if stdin is not None else None,
                stdout=asyncio.subprocess.PIPE,
                stderr=asyncio.subprocess.PIPE,
            )
        except FileNotFoundError as e:
            return False, str(e)
        out, err = await proc.communicate(stdin)
        if proc.returncode != 0:
            tail = (err or out or b"").decode(errors="replace")[-500:]
            return False, f"exit code {proc.returncode}: {tail}"
        return True, ""

    def _collect_outputs(self, tmpl: Dict[str, Any], outputs: List[Dict[str, Any]]) -> None:
        for p in ((tmpl.get("outputs") or {}).get("parameters")) or []:
            if not isinstance(p, dict):
                continue
            value = p.get("value")
            if value is None and isinstance(p.get("valueFrom"), dict):
                value = p["valueFrom"].get("default")
            if value is None:
                continue
            outputs.append(
                {
                    "name": p.get("globalName") or p.get("name"),
                    "value": value if isinstance(value, str) else json.dumps(value),
                }
            )
