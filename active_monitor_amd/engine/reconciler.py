"""The HealthCheck reconciler — the heart of the framework.

Re-implements every behavior of the reference's 1,485-line reconciler
(internal/controllers/healthcheck_controller.go) with one deliberate
architectural change (SURVEY.md §7 "the single biggest architectural
decision"): the workflow watch never blocks a reconcile worker — it runs as a
detached asyncio task — and repeat timers feed back through the workqueue, so
``MaxConcurrentReconciles`` genuinely bounds reconcile work and repeats are
reconcile-driven (picking up fresh spec, recomputing cron intervals every
cycle — the reference's bare ``time.AfterFunc`` repeats re-use stale values,
:479-500,:746-755).

Behavior contracts kept bit-for-bit:

- schedule branches: pause → ``Stopped`` + explanatory ErrorMessage (:238-250),
  cron → ``RepeatAfterSec = int(next−now)+1`` (:251-263), already-scheduled
  dedup (:264-267),
- RBAC provision → parse → submit (ownerRef, generateName, instance-id label,
  podGC, activeDeadlineSeconds) → watch with inverse-exponential backoff →
  status mutation + metrics → remedy state machine → re-arm timer,
- IEB timeout synthesizes ``{phase: Failed, message: Failed}`` (:627-632),
- remedy RunsLimit/ResetInterval state machine with the exact RemedyStatus
  strings and counter-zeroing sets (:649-660, :677-721),
- remedy RBAC create → use → delete cycle (:759-786),
- timers stopped on CR deletion (:180-184).
"""
from __future__ import annotations

import asyncio
import logging
import time
from dataclasses import dataclass
from typing import Any, Dict, Optional, Set, Tuple

from .. import API_VERSION
from ..api.types import HealthCheck, k8s_now, parse_k8s_time
from ..kube.client import EventRecorder, KubeClient
from ..kube.errors import ConflictError, NotFoundError, is_storage_error
from ..kube.registry import WF_API_VERSION, WF_KIND
from ..metrics import (
    MonitorError,
    MonitorFinishedTime,
    MonitorRuntime,
    MonitorStartedTime,
    MonitorSuccess,
    create_dynamic_prometheus_metric,
)
from .backoff import IEBTimeoutError, InverseExponentialBackoff, compute_backoff_params
from .cronx import CronParseError, seconds_until_next
from .parse import (
    parse_remedy_workflow_from_healthcheck_async,
    parse_workflow_from_healthcheck_async,
)
from .rbac import RBACProvisioner
from .workqueue import WorkQueue

log = logging.getLogger("active_monitor_amd.reconciler")

HC_KIND = "HealthCheck"
SUCC_STR = "Succeeded"
FAIL_STR = "Failed"
REMEDY = "remedy"
HEALTHCHECK = "healthCheck"
TIMER_FLAG = "timer"

#: sentinel returned by an informer-cache lookup that cannot answer (cache
#: not yet synced); distinct from None, which means "synced and not found"
CACHE_MISS = object()


@dataclass
class ReconcileResult:
    requeue_after: float = 0.0
    error: Optional[BaseException] = None


@dataclass
class RepeatTimer:
    """Entry in RepeatTimersByName. Like the reference's expired ``time.Timer``
    values (:746-755), a fired entry stays in the map so the already-scheduled
    dedup check keeps holding during the run it triggered."""

    handle: Optional[asyncio.TimerHandle] = None
    fired: bool = False

    def stop(self) -> None:
        if self.handle is not None:
            self.handle.cancel()


class HealthCheckReconciler:
    def __init__(
        self,
        client: KubeClient,
        recorder: EventRecorder,
        max_parallel: int = 10,
        queue: Optional[WorkQueue] = None,
    ):
        self.client = client
        self.recorder = recorder
        self.max_parallel = max_parallel
        # NB: not `queue or WorkQueue()` — an empty WorkQueue is falsy (__len__)
        self.queue = queue if queue is not None else WorkQueue()
        self.rbac = RBACProvisioner(client, recorder)
        # optional WorkflowWatchHub (set by the Manager): event-driven wakeups
        # for workflow completion; None falls back to pure IEB polling
        self.wf_hub = None
        # optional informer-cache lookup (set by the Manager): serves
        # HealthCheck reads from the watch-fed cache like controller-runtime's
        # cached client (healthcheck_controller.go:133 reads through the
        # informer cache, not the wire). Returns a dict, None (synced and
        # absent ⇒ NotFound) or CACHE_MISS (not synced / not covered).
        self.hc_lookup = None
        # read-your-writes floor per key: the rv our last status write
        # produced. A cache entry older than this is our own write's event
        # still in flight — reading it would lose the update (a stale
        # success/failed count rewritten over the real one). Cache reads
        # below the floor fall through to a direct GET. This closes a
        # lost-update race controller-runtime's cached client actually has.
        self._written_rv: Dict[Tuple[str, str], int] = {}
        self.repeat_timers_by_name: Dict[str, RepeatTimer] = {}
        self._watch_tasks: Dict[str, Set[asyncio.Task]] = {}
        # observability for benchmarks/tests
        self.reconcile_count = 0
        self.completed_runs = 0

    # ------------------------------------------------------------------
    # timers
    # ------------------------------------------------------------------

    def get_timer_by_name(self, name: str, namespace: str = "") -> Optional[RepeatTimer]:
        """Timers are keyed (namespace, name) — the reference keys by name
        alone (:139), making same-name CRs in different namespaces fight over
        one timer slot; fixed here with no other visible change. The
        name-only form falls back to any-namespace lookup for API parity."""
        if namespace:
            return self.repeat_timers_by_name.get((namespace, name))
        for (ns, n), entry in self.repeat_timers_by_name.items():
            if n == name:
                return entry
        return None

    def _arm_repeat_timer(self, name: str, namespace: str, delay: float) -> None:
        key = (namespace, name)
        old = self.repeat_timers_by_name.get(key)
        if old is not None:
            old.stop()
        entry = RepeatTimer()
        loop = asyncio.get_running_loop()
        entry.handle = loop.call_later(
            max(0.0, delay), self._on_timer_fired, entry, name, namespace
        )
        self.repeat_timers_by_name[key] = entry

    def _on_timer_fired(self, entry: RepeatTimer, name: str, namespace: str) -> None:
        entry.fired = True
        # repeats flow through the workqueue (bounded by max_parallel); the
        # TIMER flag bypasses only the already-scheduled dedup branch.
        # Synchronous enqueue — no fire-and-forget task to lose.
        self.queue.add_nowait((namespace, name), {TIMER_FLAG})

    def _stop_timer(self, name: str, namespace: str = "") -> bool:
        entry = self.repeat_timers_by_name.pop((namespace, name), None)
        if entry is not None:
            entry.stop()
            return True
        return False

    # ------------------------------------------------------------------
    # watch-task registry
    # ------------------------------------------------------------------

    def _spawn_watch(self, key, coro) -> asyncio.Task:
        task = asyncio.get_running_loop().create_task(coro)
        self._watch_tasks.setdefault(key, set()).add(task)

        def _done(t: asyncio.Task, key=key) -> None:
            tasks = self._watch_tasks.get(key)
            if tasks is not None:
                tasks.discard(t)
                if not tasks:
                    self._watch_tasks.pop(key, None)
            if not t.cancelled() and t.exception() is not None:
                log.error("watch task for %s failed: %s", key, t.exception())

        task.add_done_callback(_done)
        return task

    def _cancel_watches(self, key) -> None:
        for t in list(self._watch_tasks.get(key, ())):
            t.cancel()

    def active_watches(self) -> int:
        return sum(len(v) for v in self._watch_tasks.values())

    async def drain(self, timeout: float = 30.0) -> None:
        """Await all in-flight watch tasks (test/shutdown helper)."""
        deadline = time.monotonic() + timeout
        while self._watch_tasks and time.monotonic() < deadline:
            tasks = [t for ts in self._watch_tasks.values() for t in ts]
            if not tasks:
                break
            await asyncio.wait(tasks, timeout=min(1.0, deadline - time.monotonic()))

    # ------------------------------------------------------------------
    # events
    # ------------------------------------------------------------------

    async def _event(self, hc: HealthCheck, ev_type: str, message: str) -> None:
        # the recorder only reads object identity — don't serialize the spec
        involved = {
            "apiVersion": API_VERSION,
            "kind": HC_KIND,
            "metadata": {
                "name": hc.metadata.name,
                "namespace": hc.metadata.namespace,
                "uid": hc.metadata.uid,
            },
        }
        await self.recorder.event(involved, ev_type, ev_type, message)

    # ------------------------------------------------------------------
    # Reconcile entry (reference :170-223)
    # ------------------------------------------------------------------

    @staticmethod
    def _rv_of(obj: Optional[Dict[str, Any]]) -> int:
        try:
            return int((obj.get("metadata") or {}).get("resourceVersion", 0))
        except (TypeError, ValueError, AttributeError):
            return -1  # opaque rv: cannot order, treat as unusable floor

    def note_written(self, namespace: str, name: str, obj: Optional[Dict[str, Any]]) -> None:
        """Record the rv a write of ours produced (read-your-writes floor)."""
        rv = self._rv_of(obj)
        if rv > 0:
            key = (namespace, name)
            if rv > self._written_rv.get(key, 0):
                self._written_rv[key] = rv

    async def _get_hc(self, namespace: str, name: str) -> Dict[str, Any]:
        """Read a HealthCheck through the informer cache when one is synced,
        else over the wire. Cache hits return a shallow top-level copy so the
        caller may replace top-level keys (the status-update path does)
        without corrupting the shared cached object. Entries older than our
        own last write (the event is still in flight) are not served — the
        read goes direct instead, so a timer-fired repeat can never observe
        pre-write counters and lose the update."""
        lk = self.hc_lookup
        if lk is not None:
            hit = lk(namespace, name)
            if hit is None:
                self._written_rv.pop((namespace, name), None)
                raise NotFoundError(
                    f'healthchecks.activemonitor.keikoproj.io "{name}" not found'
                )
            if hit is not CACHE_MISS:
                floor = self._written_rv.get((namespace, name), 0)
                if floor == 0 or 0 < floor <= self._rv_of(hit):
                    return dict(hit)
        return await self.client.get(API_VERSION, HC_KIND, namespace, name,
                                     snapshot_read=True)

    async def reconcile(
        self, namespace: str, name: str, flags: Optional[Set[str]] = None
    ) -> ReconcileResult:
        flags = flags or set()
        self.reconcile_count += 1
        try:
            obj = await self._get_hc(namespace, name)
        except NotFoundError:
            # CR deleted: stop the repeat timer so self-scheduling halts
            # (:180-184); in-flight watches are cancelled proactively (the
            # reference lets them die on workflow NotFound after GC).
            if self._stop_timer(name, namespace):
                log.info("cancelled rescheduled workflow for deleted healthcheck %s", name)
                await self.recorder.event(
                    {"apiVersion": API_VERSION, "kind": HC_KIND,
                     "metadata": {"name": name, "namespace": namespace}},
                    "Normal", "Normal",
                    "Cancelling workflow for this healthcheck due to deletion",
                )
            self._cancel_watches((namespace, name))
            return ReconcileResult()
        hc = HealthCheck.from_dict(obj)
        return await self._process_or_recover(hc, from_timer=TIMER_FLAG in flags)

    async def _process_or_recover(self, hc: HealthCheck, from_timer: bool) -> ReconcileResult:
        """Panic guard + error policy (reference :190-223). The reference's
        trailing full-object Update (:208-215) is omitted: with the status
        subresource enabled it can persist neither status nor the in-memory
        spec mutations of the fresh object it writes — a pure no-op write per
        reconcile. Status persistence happens where it matters, in the watch
        completion paths (:734,:858) and the pause branch (:246)."""
        try:
            return await self._process_healthcheck(hc, from_timer)
        except asyncio.CancelledError:
            raise
        except Exception as e:  # recover() equivalent — never crash a worker
            if is_storage_error(e):
                return ReconcileResult()
            log.warning("healthcheck %s process error: %s", hc.name, e)
            return ReconcileResult(requeue_after=1.0, error=e)

    # ------------------------------------------------------------------
    # Scheduler (reference :225-291)
    # ------------------------------------------------------------------

    async def _process_healthcheck(self, hc: HealthCheck, from_timer: bool) -> ReconcileResult:
        spec = hc.spec
        if spec.workflow.resource is None:
            return ReconcileResult()
        wf_namespace = spec.workflow.resource.namespace

        finished_unix = 0.0
        t = parse_k8s_time(hc.status.finished_at)
        if t is not None:
            finished_unix = t.timestamp()

        if spec.repeat_after_sec <= 0 and spec.schedule.cron == "":
            # pause branch (:238-250) — exact status strings
            hc.status.status = "Stopped"
            hc.status.error_message = (
                "workflow execution is stopped; either spec.RepeatAfterSec or "
                "spec.Schedule must be provided. spec.RepeatAfterSec set to "
                f"{spec.repeat_after_sec}. spec.Schedule set to {spec.schedule.go_string()}"
            )
            hc.status.finished_at = k8s_now()
            await self._event(
                hc,
                "Warning",
                "Workflow execution is stopped; either spec.RepeatAfterSec or "
                "spec.Schedule must be provided",
            )
            await self.update_healthcheck_status(hc)
            return ReconcileResult()
        elif not from_timer and self._watch_tasks.get((hc.namespace, hc.name)):
            # a run for this CR is already in flight: spurious reconciles
            # (informer list/watch overlap at startup, spec edits mid-run)
            # must not submit a duplicate workflow. The reference double-
            # submits here and parks a second blocked goroutine per duplicate
            # (SURVEY.md §2.3.1); the clean rebuild closes that hole.
            return ReconcileResult()
        elif spec.repeat_after_sec <= 0 and spec.schedule.cron != "":
            # cron branch (:251-263): +1s compensates integer truncation
            try:
                hc.spec.repeat_after_sec = seconds_until_next(spec.schedule.cron)
            except CronParseError as e:
                await self._event(hc, "Warning", "Fail to parse cron")
                raise e
            # the reference's elif chain exempts cron CRs from the
            # already-scheduled dedup, so every status write re-submits them
            # in a tight loop until the blocking watch throttles it; apply the
            # dedup here with the freshly computed interval instead
            if (
                not from_timer
                and int(time.time() - finished_unix) < hc.spec.repeat_after_sec
                and self.get_timer_by_name(hc.name, hc.namespace) is not None
            ):
                return ReconcileResult()
        elif (
            not from_timer
            and int(time.time() - finished_unix) < spec.repeat_after_sec
            and self.get_timer_by_name(hc.name, hc.namespace) is not None
        ):
            # already executed recently and a repeat is scheduled (:264-267)
            return ReconcileResult()

        try:
            await self.rbac.create_rbac_for_workflow(hc, HEALTHCHECK)
        except Exception as e:
            await self._event(hc, "Warning", "Error creating RBAC for HealthCheckWorkflow")
            raise e

        generated_name = await self.create_submit_workflow(hc)
        # non-blocking watch: the reconcile worker is freed immediately
        self._spawn_watch(
            (hc.namespace, hc.name),
            self.watch_workflow_reschedule(wf_namespace, generated_name, hc),
        )
        return ReconcileResult()

    # ------------------------------------------------------------------
    # Submit (reference :502-571)
    # ------------------------------------------------------------------

    def _owner_reference(self, hc: HealthCheck) -> Dict[str, Any]:
        return {
            "kind": HC_KIND,
            "apiVersion": API_VERSION,
            "name": hc.name,
            "uid": hc.metadata.uid,
            "controller": True,
        }

    async def create_submit_workflow(self, hc: HealthCheck) -> str:
        try:
            spec, labels = await parse_workflow_from_healthcheck_async(hc)
        except Exception as e:
            await self._event(hc, "Warning", "Error creating or submitting workflow")
            raise e
        await self._event(hc, "Normal", "workflow is parsed from healthcheck")
        wf = {
            "apiVersion": WF_API_VERSION,
            "kind": WF_KIND,
            "metadata": {
                "generateName": hc.spec.workflow.generate_name,
                "namespace": hc.spec.workflow.resource.namespace,
                "labels": labels,
                "ownerReferences": [self._owner_reference(hc)],
            },
            "spec": spec,
        }
        created = await self.client.create(wf, transfer=True)
        await self._event(hc, "Normal", "Successfully created workflow")
        return created["metadata"]["name"]

    async def create_submit_remedy_workflow(self, hc: HealthCheck) -> str:
        if hc.spec.remedy_workflow.resource is None:
            raise ValueError("RemedyWorkflow Resource is nil")
        try:
            spec, labels = await parse_remedy_workflow_from_healthcheck_async(hc)
        except Exception as e:
            await self._event(hc, "Warning", "Error creating or submitting remedyworkflow")
            raise e
        await self._event(hc, "Normal", "Remedy workflow is parsed from healthcheck")
        wf = {
            "apiVersion": WF_API_VERSION,
            "kind": WF_KIND,
            "metadata": {
                "generateName": hc.spec.remedy_workflow.generate_name,
                "namespace": hc.spec.remedy_workflow.resource.namespace,
                "labels": labels,
                "ownerReferences": [self._owner_reference(hc)],
            },
            "spec": spec,
        }
        created = await self.client.create(wf, transfer=True)
        await self._event(hc, "Normal", "Successfully created remedyWorkflow")
        return created["metadata"]["name"]

    # ------------------------------------------------------------------
    # Watch: health-check workflow (reference :607-757)
    # ------------------------------------------------------------------

    _POLL_RETRIES = 3

    async def _poll_workflow(
        self, namespace: str, name: str, via_cache: bool = False
    ) -> Optional[Dict[str, Any]]:
        """One status poll. NotFound propagates (ends the watch, :618-622);
        transient apiserver errors are retried briefly and then treated as a
        missed poll — the reference aborts the whole watch on any error,
        stalling the CR until an external reconcile.

        ``via_cache=True`` reads from the watch hub's last-event cache (the
        watch payload already carried the full Workflow) instead of a GET
        round-trip — the informer-cache read discipline of controller-runtime.
        A cache miss means no event reached the hub yet, i.e. the freshly
        submitted workflow has no status — reported as such without touching
        the wire. The IEB fallback polls (hub wait timed out) always go
        direct, so a lost watch event costs one poll interval, never
        correctness."""
        if via_cache and self.wf_hub is not None:
            hit = self.wf_hub.cached(namespace, name)
            if hit is not None:
                ev_type, obj = hit
                if ev_type == "DELETED":
                    raise NotFoundError(f'workflow "{name}" not found (deleted)')
                status = obj.get("status")
                return status if isinstance(status, dict) else None
            return None
        last: Optional[BaseException] = None
        for attempt in range(self._POLL_RETRIES):
            try:
                wf = await self.client.get(WF_API_VERSION, WF_KIND, namespace, name,
                                           snapshot_read=True)
                status = wf.get("status")
                return status if isinstance(status, dict) else None
            except NotFoundError:
                raise
            except asyncio.CancelledError:
                raise
            except Exception as e:
                last = e
                await asyncio.sleep(0.05 * (attempt + 1))
        log.warning("workflow poll %s/%s failed transiently: %s", namespace, name, last)
        return None  # missed poll; the IEB deadline still bounds the watch

    async def _wait_next_poll(
        self, ieb: InverseExponentialBackoff, namespace: str, name: str, since: Optional[int]
    ) -> bool:
        """Wait until the next poll is due. With a watch hub the wait ends the
        moment the workflow changes (ms-scale completion detection); without
        one this is exactly the reference's IEB sleep. Raises IEBTimeoutError
        past the deadline either way — the synthesized-failure semantics are
        identical. Returns True when a watch event (not a timeout) ended the
        wait, i.e. the hub cache holds the fresh object and the next poll may
        read it without a wire round-trip."""
        if self.wf_hub is None:
            await ieb.next()
            return False
        ieb.check_deadline()
        interval = ieb.peek_interval()
        ieb.decay()
        changed = await self.wf_hub.wait_change(namespace, name, interval, since=since)
        ieb.check_deadline()
        return changed is not None

    async def watch_workflow_reschedule(
        self, wf_namespace: str, wf_name: str, hc: HealthCheck
    ) -> None:
        then = k8s_now()
        then_unix = time.time()
        repeat_after_sec = hc.spec.repeat_after_sec
        max_t, min_t, factor, timeout = compute_backoff_params(
            hc.spec.backoff_max, hc.spec.backoff_min, hc.spec.backoff_factor,
            hc.spec.workflow.timeout,
        )
        ieb: Optional[InverseExponentialBackoff] = None
        timed_out = False
        try:
            ieb = InverseExponentialBackoff(max_t, min_t, timeout, factor)
        except ValueError:
            # invalid params (e.g. timeout 0): the reference's loop sees a
            # constructor error on its first iteration and synthesizes Failed
            timed_out = True

        # the first poll may read the hub cache (a just-submitted workflow has
        # no status until an event lands); timed-out waits force direct polls
        via_cache = self.wf_hub is not None
        while True:
            # pre-poll change counter: any event landing between this poll and
            # the wait below is detected immediately (no lost wakeups)
            seq = self.wf_hub.seq(wf_namespace, wf_name) if self.wf_hub else None
            try:
                status = await self._poll_workflow(wf_namespace, wf_name,
                                                   via_cache=via_cache)
            except NotFoundError:
                # parent healthcheck likely deleted; don't reschedule (:618-622)
                await self._event(
                    hc,
                    "Warning",
                    "Error attempting to find workflow for healthcheck. This may "
                    "indicate that either the healthcheck was removed or the Workflow "
                    "was GC'd before active-monitor could obtain the status",
                )
                if self.wf_hub is not None:
                    self.wf_hub.forget(wf_namespace, wf_name)
                return
            if timed_out:
                status = {"phase": FAIL_STR, "message": FAIL_STR}
                await self._event(hc, "Warning", "Workflow timed out")
            if status is not None:
                phase = status.get("phase")
                now = k8s_now()
                now_unix = time.time()
                if phase == SUCC_STR:
                    await self._event(hc, "Normal", "Workflow status is Succeeded")
                    hc.status.status = SUCC_STR
                    hc.status.started_at = then
                    hc.status.finished_at = now
                    hc.status.success_count += 1
                    hc.status.total_healthcheck_runs = (
                        hc.status.success_count + hc.status.failed_count
                    )
                    hc.status.last_successful_workflow = wf_name
                    MonitorSuccess.labels(hc.name, HEALTHCHECK).inc()
                    MonitorRuntime.labels(hc.name, HEALTHCHECK).set(now_unix - then_unix)
                    MonitorStartedTime.labels(hc.name, HEALTHCHECK).set(int(then_unix))
                    MonitorFinishedTime.labels(hc.name, HEALTHCHECK).set(int(now_unix))
                    # custom metrics from workflow output parameters — the
                    # reference documents this (README.md:275-285) but never
                    # wires it; here the documented feature is real.
                    create_dynamic_prometheus_metric(hc.name, status)
                    if (
                        not hc.spec.remedy_workflow.is_empty()
                        and hc.status.remedy_total_runs >= 1
                    ):
                        hc.status.reset_remedy()
                        hc.status.remedy_status = "HealthCheck Passed so Remedy is reset"
                        await self._event(hc, "Normal", "HealthCheck passed so Remedy is reset")
                    break
                elif phase == FAIL_STR:
                    await self._event(hc, "Warning", "Workflow status is Failed")
                    hc.status.status = FAIL_STR
                    hc.status.started_at = then
                    hc.status.finished_at = now
                    hc.status.last_failed_at = now
                    msg = status.get("message")
                    hc.status.error_message = msg if isinstance(msg, str) else ""
                    hc.status.failed_count += 1
                    hc.status.total_healthcheck_runs = (
                        hc.status.success_count + hc.status.failed_count
                    )
                    hc.status.last_failed_workflow = wf_name
                    MonitorError.labels(hc.name, HEALTHCHECK).inc()
                    MonitorStartedTime.labels(hc.name, HEALTHCHECK).set(int(then_unix))
                    MonitorFinishedTime.labels(hc.name, HEALTHCHECK).set(int(now_unix))
                    try:
                        await self._maybe_run_remedy(hc, now_unix)
                    except asyncio.CancelledError:
                        raise
                    except Exception as e:
                        # a remedy failure must not lose the health-check
                        # status or the repeat schedule (the reference aborts
                        # the whole watch here, stalling the CR — :686-688)
                        log.warning("remedy for %s failed: %s", hc.name, e)
                        await self._event(hc, "Warning", "Error executing RemedyWorkflow")
                    break
            # not terminal yet: wait for the next poll (hub-accelerated)
            try:
                via_cache = await self._wait_next_poll(ieb, wf_namespace, wf_name, seq)
            except IEBTimeoutError:
                timed_out = True

        try:
            await self._finish_and_reschedule(hc, wf_namespace, wf_name, repeat_after_sec)
        finally:
            # this watch is done with the workflow forever (names are unique):
            # release its hub cache/seq entry now rather than waiting for the
            # server-side TTL delete (RSS leak at fleet rates otherwise)
            if self.wf_hub is not None:
                self.wf_hub.forget(wf_namespace, wf_name)
            self.completed_runs += 1

    async def _maybe_run_remedy(self, hc: HealthCheck, now_unix: float) -> None:
        """The remedy trigger state machine (reference :677-721)."""
        if hc.spec.remedy_workflow.is_empty():
            return
        limit = hc.spec.remedy_runs_limit
        reset_interval = hc.spec.remedy_reset_interval
        if limit != 0 and reset_interval != 0:
            if limit > hc.status.remedy_total_runs:
                await self.process_remedy_workflow(hc)
            else:
                rf = parse_k8s_time(hc.status.remedy_finished_at)
                since_last = int(now_unix - rf.timestamp()) if rf is not None else reset_interval + 1
                if reset_interval >= since_last:
                    log.info(
                        "skipping remedy for %s: remedy limit met, will run after reset interval",
                        hc.name,
                    )
                else:
                    hc.status.reset_remedy()
                    hc.status.remedy_status = "RemedyResetInterval elapsed so Remedy is reset"
                    await self._event(
                        hc, "Normal", "RemedyResetInterval elapsed so Remedy is reset"
                    )
                    await self.process_remedy_workflow(hc)
        else:
            await self.process_remedy_workflow(hc)

    async def _finish_and_reschedule(
        self, hc: HealthCheck, wf_namespace: str, wf_name: str, repeat_after_sec: float
    ) -> None:
        """Persist status and re-arm the repeat timer (reference :729-757).

        The timer is armed BEFORE the status write so the MODIFIED event the
        write triggers always observes an armed timer in the dedup branch —
        closing a duplicate-submission race the reference leaves open by
        updating first (:734) and arming after (:746)."""
        try:
            fresh = await self._get_hc(hc.namespace, hc.name)
        except NotFoundError:
            return
        if (fresh.get("metadata") or {}).get("deletionTimestamp"):
            return
        # cron CRs: recompute the delay at completion time so the next run
        # lands on the cron tick, not submit-time + interval (the reference
        # re-arms with the stale submit-time value, drifting by the run's
        # duration — :746-752)
        fresh_spec = fresh.get("spec") or {}
        cron = ((fresh_spec.get("schedule") or {}).get("cron")) or ""
        if int(fresh_spec.get("repeatAfterSec", 0) or 0) <= 0 and cron:
            try:
                repeat_after_sec = seconds_until_next(cron)
            except CronParseError:
                pass  # keep the submit-time value
        self._arm_repeat_timer(hc.name, hc.namespace, repeat_after_sec)
        try:
            await self.update_healthcheck_status(hc, fresh=fresh)
        except NotFoundError:
            self._stop_timer(hc.name, hc.namespace)
            return
        except Exception as e:
            await self._event(hc, "Warning", "Error updating healthcheck resource")
            self._stop_timer(hc.name, hc.namespace)
            raise e
        await self._event(hc, "Normal", "Rescheduled workflow for next run")

    # ------------------------------------------------------------------
    # Remedy (reference :759-874)
    # ------------------------------------------------------------------

    async def process_remedy_workflow(self, hc: HealthCheck) -> None:
        """create remedy RBAC → submit → watch → delete remedy RBAC
        (reference :759-786)."""
        await self.rbac.create_rbac_for_workflow(hc, REMEDY)
        generated_name = await self.create_submit_remedy_workflow(hc)
        # watch in the remedy's own namespace — the reference watches in the
        # health-check workflow's namespace (:773), which mis-targets when the
        # two differ; fixed here with no CR-visible change for the common case
        remedy_ns = hc.spec.remedy_workflow.resource.namespace
        await self.watch_remedy_workflow(remedy_ns, generated_name, hc)
        await self.rbac.delete_rbac_for_workflow(hc)

    async def watch_remedy_workflow(
        self, wf_namespace: str, wf_name: str, hc: HealthCheck
    ) -> None:
        """Remedy watch loop (reference :788-874): backoff params derive from
        the *health-check* workflow's timeout with factor fixed at 0.5
        (:791-801), a reference quirk kept for behavioral parity."""
        then = k8s_now()
        then_unix = time.time()
        max_t, min_t, _factor, timeout = compute_backoff_params(
            0, 0, "", hc.spec.workflow.timeout
        )
        ieb: Optional[InverseExponentialBackoff] = None
        timed_out = False
        try:
            ieb = InverseExponentialBackoff(max_t, min_t, timeout, 0.5)
        except ValueError:
            timed_out = True

        via_cache = self.wf_hub is not None
        while True:
            seq = self.wf_hub.seq(wf_namespace, wf_name) if self.wf_hub else None
            try:
                status = await self._poll_workflow(wf_namespace, wf_name,
                                                   via_cache=via_cache)
            except NotFoundError:
                if self.wf_hub is not None:
                    self.wf_hub.forget(wf_namespace, wf_name)
                return
            if timed_out:
                status = {"phase": FAIL_STR, "message": FAIL_STR}
                await self._event(hc, "Warning", "remedy workflow is timedout")
            if status is not None:
                phase = status.get("phase")
                now = k8s_now()
                now_unix = time.time()
                if phase == SUCC_STR:
                    await self._event(hc, "Normal", "Remedy workflow status is Succeeded")
                    hc.status.remedy_status = SUCC_STR
                    hc.status.remedy_started_at = then
                    hc.status.remedy_finished_at = now
                    hc.status.remedy_success_count += 1
                    hc.status.remedy_total_runs = (
                        hc.status.remedy_success_count + hc.status.remedy_failed_count
                    )
                    # reference quirk kept: remedy watch writes the shared
                    # last*Workflow fields with the remedy name (:830)
                    hc.status.last_successful_workflow = wf_name
                    MonitorSuccess.labels(hc.name, REMEDY).inc()
                    MonitorRuntime.labels(hc.name, REMEDY).set(now_unix - then_unix)
                    MonitorStartedTime.labels(hc.name, REMEDY).set(int(then_unix))
                    # reference quirk kept: finished-time gauge carries the
                    # HEALTH-CHECK finish epoch, not the remedy's (:834)
                    hc_fin = parse_k8s_time(hc.status.finished_at)
                    MonitorFinishedTime.labels(hc.name, REMEDY).set(
                        int(hc_fin.timestamp()) if hc_fin else int(now_unix)
                    )
                    break
                elif phase == FAIL_STR:
                    await self._event(hc, "Warning", "remedy workflow status is failed")
                    hc.status.remedy_status = FAIL_STR
                    hc.status.remedy_started_at = then
                    hc.status.remedy_finished_at = now
                    hc.status.remedy_last_failed_at = now
                    msg = status.get("message")
                    hc.status.remedy_error_message = msg if isinstance(msg, str) else ""
                    hc.status.remedy_failed_count += 1
                    hc.status.remedy_total_runs = (
                        hc.status.remedy_success_count + hc.status.remedy_failed_count
                    )
                    hc.status.last_failed_workflow = wf_name
                    MonitorError.labels(hc.name, REMEDY).inc()
                    MonitorStartedTime.labels(hc.name, REMEDY).set(int(then_unix))
                    MonitorFinishedTime.labels(hc.name, REMEDY).set(int(now_unix))
                    break
            try:
                via_cache = await self._wait_next_poll(ieb, wf_namespace, wf_name, seq)
            except IEBTimeoutError:
                timed_out = True

        if self.wf_hub is not None:  # done with this workflow forever
            self.wf_hub.forget(wf_namespace, wf_name)
        # persist remedy status promptly (reference :856-871)
        try:
            fresh = await self._get_hc(hc.namespace, hc.name)
        except NotFoundError:
            return
        if (fresh.get("metadata") or {}).get("deletionTimestamp"):
            return
        await self.update_healthcheck_status(hc, fresh=fresh)

    # ------------------------------------------------------------------
    # Status persistence (reference :1445-1462)
    # ------------------------------------------------------------------

    async def update_healthcheck_status(
        self, hc: HealthCheck, retries: int = 5,
        fresh: Optional[Dict[str, Any]] = None,
    ) -> None:
        """Fresh Get + status-subresource update with conflict retry.
        ``fresh`` lets callers that just read the object donate that read for
        the first attempt (one round-trip saved per completed run)."""
        last: Optional[BaseException] = None
        for attempt in range(retries):
            if fresh is None:
                fresh = await self.client.get(API_VERSION, HC_KIND, hc.namespace,
                                              hc.name, snapshot_read=True)
            fresh["status"] = hc.status.to_dict()
            try:
                out = await self.client.update_status(fresh)
                # read-your-writes: later cache reads must be ≥ this version
                self.note_written(hc.namespace, hc.name, out)
                return
            except ConflictError as e:
                last = e
                fresh = None  # stale: re-read on the next attempt
                await asyncio.sleep(0.01 * (attempt + 1))
        raise last if last else RuntimeError("status update failed")

    # ------------------------------------------------------------------
    # Shutdown
    # ------------------------------------------------------------------

    def stop_all(self) -> None:
        for (ns, name) in list(self.repeat_timers_by_name):
            self._stop_timer(name, ns)
        for key in list(self._watch_tasks):
            self._cancel_watches(key)
