"""Workflow YAML parsing + field injection.

Re-implements the reference's parseWorkflowFromHealthcheck /
parseRemedyWorkflowFromHealthcheck (healthcheck_controller.go:876-1125):

- read the definition through the artifact store,
- YAML-decode; non-map documents and missing/non-map ``spec`` are errors with
  the reference's messages,
- label handling: workflow-defined ``metadata.labels`` are used (values
  stringified), the Argo instance-id label
  ``workflows.argoproj.io/controller-instanceid: activemonitor-workflows``
  is guaranteed as a default. Unlike the reference — whose shared
  ``workflowLabels`` map leaks labels across HealthChecks —
  labels here are scoped per submission,
- inject ``spec.podGC = {strategy: OnPodCompletion}`` when unset (:953-977),
- inject ``spec.serviceAccountName`` from the resource (:988-991),
- ``spec.activeDeadlineSeconds`` defaulting: health-check workflows default to
  ``Workflow.Timeout``, itself defaulted to ``RepeatAfterSec`` (a spec mutation
  that later feeds the backoff computation, :980-995); remedy workflows default
  to ``RepeatAfterSec`` and round-trip an existing numeric deadline back into
  ``RemedyWorkflow.Timeout`` (:1108-1120).
"""
from __future__ import annotations

import asyncio
from typing import Any, Dict, Optional, Tuple

import yaml

from ..api.types import HealthCheck
from ..store import ArtifactReadError, get_artifact_reader

WF_INSTANCE_ID_LABEL_KEY = "workflows.argoproj.io/controller-instanceid"
WF_INSTANCE_ID = "activemonitor-workflows"
POD_GC_ON_POD_COMPLETION = "OnPodCompletion"


class WorkflowParseError(Exception):
    pass


def _extract_labels(data: Dict[str, Any]) -> Dict[str, str]:
    """Per-submission label derivation with the instance-id default."""
    labels: Dict[str, str] = {}
    metadata = data.get("metadata")
    if isinstance(metadata, dict):
        raw = metadata.get("labels")
        if isinstance(raw, dict):
            for k, v in raw.items():
                labels[str(k)] = str(v)
    labels.setdefault(WF_INSTANCE_ID_LABEL_KEY, WF_INSTANCE_ID)
    return labels


# libyaml loader when available (~10x faster than the pure-Python one)
_Loader = getattr(yaml, "CSafeLoader", yaml.SafeLoader)

# Workflow definitions are re-parsed every cycle (reference :894); fleets
# re-submit the same YAML thousands of times, so cache parses by content.
_PARSE_CACHE: Dict[bytes, Dict[str, Any]] = {}
_PARSE_CACHE_MAX = 2048


def _decode(content: bytes, remedy: bool) -> Dict[str, Any]:
    from ..utils.fastcopy import deep_copy

    cached = _PARSE_CACHE.get(content)
    if cached is not None:
        return deep_copy(cached)
    try:
        data = yaml.load(content, Loader=_Loader)
    except yaml.YAMLError as e:
        raise WorkflowParseError(f"Invalid spec file passed: {e}") from e
    if data is None:
        data = {}
    if not isinstance(data, dict):
        raise WorkflowParseError("Invalid spec file passed: not a mapping")
    if len(_PARSE_CACHE) >= _PARSE_CACHE_MAX:
        _PARSE_CACHE.clear()  # simple bound; refill costs one parse per doc
    _PARSE_CACHE[content] = deep_copy(data)
    return data


def _validated_spec(data: Dict[str, Any], remedy: bool) -> Dict[str, Any]:
    spec_raw = data.get("spec")
    if spec_raw is None:
        raise WorkflowParseError(
            "Invalid remedy workflow, missing spec" if remedy else "invalid workflow, missing spec"
        )
    if not isinstance(spec_raw, dict):
        raise WorkflowParseError(
            "invalid remedy workflow, spec is not a map"
            if remedy
            else "invalid workflow, spec is not a map"
        )
    return spec_raw


#: ceiling on one offloaded artifact read before the reconcile errors out
ARTIFACT_READ_TIMEOUT = 35.0


def _source_blocks(resource) -> bool:
    """URL and file sources do real I/O; inline is a memory read."""
    src = getattr(resource, "source", None)
    return src is not None and (src.url is not None or src.file is not None)


async def _read_source_async(resource) -> bytes:
    """Read an artifact source without blocking the event loop.

    The reference's URL fetch blocks only its own goroutine (url.go:20-57);
    a single-loop design must offload it. Inline sources stay on-loop (pure
    memory); URL/file reads run on a worker thread under a hard timeout so one
    hung source cannot stall other CRs' reconciles, watches or timers."""
    if resource is None:
        return b""
    reader = get_artifact_reader(resource.source)
    if not _source_blocks(resource):
        return reader.read()
    try:
        return await asyncio.wait_for(asyncio.to_thread(reader.read), ARTIFACT_READ_TIMEOUT)
    except asyncio.TimeoutError:
        raise ArtifactReadError(
            f"artifact read timed out after {ARTIFACT_READ_TIMEOUT:.0f}s"
        ) from None


def parse_workflow_from_healthcheck(
    hc: HealthCheck, content: Optional[bytes] = None
) -> Tuple[Dict[str, Any], Dict[str, str]]:
    """Returns ``(spec_dict, labels)`` for the health-check workflow, mutating
    ``hc.spec.workflow.timeout`` when it defaults from RepeatAfterSec
    (reference :980-995). ``content`` short-circuits the artifact read (used
    by the async variant, which offloads blocking sources to a thread)."""
    if content is None:
        content = b""
        if hc.spec.workflow.resource is not None:
            reader = get_artifact_reader(hc.spec.workflow.resource.source)
            content = reader.read()
    data = _decode(content, remedy=False)
    labels = _extract_labels(data)
    spec = _validated_spec(data, remedy=False)

    if spec.get("podGC") is None:
        spec["podGC"] = {"strategy": POD_GC_ON_POD_COMPLETION}

    if hc.spec.workflow.timeout == 0:
        hc.spec.workflow.timeout = hc.spec.repeat_after_sec
    timeout = hc.spec.workflow.timeout

    if hc.spec.workflow.resource is not None and hc.spec.workflow.resource.service_account:
        spec["serviceAccountName"] = hc.spec.workflow.resource.service_account

    if spec.get("activeDeadlineSeconds") is None:
        spec["activeDeadlineSeconds"] = timeout
    return spec, labels


async def parse_workflow_from_healthcheck_async(
    hc: HealthCheck,
) -> Tuple[Dict[str, Any], Dict[str, str]]:
    """Event-loop-safe variant: blocking sources (URL/file) are read on a
    worker thread with a timeout before the pure-CPU parse runs on-loop."""
    content = await _read_source_async(hc.spec.workflow.resource)
    return parse_workflow_from_healthcheck(hc, content=content)


def parse_remedy_workflow_from_healthcheck(
    hc: HealthCheck, content: Optional[bytes] = None
) -> Tuple[Dict[str, Any], Dict[str, str]]:
    """Remedy variant (reference :1002-1125): ``activeDeadlineSeconds``
    defaults from RepeatAfterSec; an existing numeric deadline round-trips
    into ``hc.spec.remedy_workflow.timeout``."""
    if content is None:
        content = b""
        if hc.spec.remedy_workflow.resource is not None:
            reader = get_artifact_reader(hc.spec.remedy_workflow.resource.source)
            content = reader.read()
    data = _decode(content, remedy=True)
    labels = _extract_labels(data)
    spec = _validated_spec(data, remedy=True)

    if spec.get("podGC") is None:
        spec["podGC"] = {"strategy": POD_GC_ON_POD_COMPLETION}

    if (
        hc.spec.remedy_workflow.resource is not None
        and hc.spec.remedy_workflow.resource.service_account
    ):
        spec["serviceAccountName"] = hc.spec.remedy_workflow.resource.service_account

    timeout = hc.spec.repeat_after_sec
    deadline = spec.get("activeDeadlineSeconds")
    if deadline is None:
        spec["activeDeadlineSeconds"] = timeout
        hc.spec.remedy_workflow.timeout = timeout
    elif isinstance(deadline, (int, float)) and not isinstance(deadline, bool):
        hc.spec.remedy_workflow.timeout = int(deadline)
    else:
        hc.spec.remedy_workflow.timeout = timeout
    return spec, labels


async def parse_remedy_workflow_from_healthcheck_async(
    hc: HealthCheck,
) -> Tuple[Dict[str, Any], Dict[str, str]]:
    """Event-loop-safe remedy variant (see
    :func:`parse_workflow_from_healthcheck_async`)."""
    content = await _read_source_async(hc.spec.remedy_workflow.resource)
    return parse_remedy_workflow_from_healthcheck(hc, content=content)
