"""CRD manifest generation for HealthCheck.

Produces the CustomResourceDefinition the reference ships generated
(config/crd/bases/activemonitor.keikoproj.io_healthchecks.yaml): identical
group/version/kind, shortnames ``hc``/``hcs``, status subresource, the six
kubectl printcolumns (healthcheck_types.go:68-76), and an openAPI v3 schema
derived from the same field set — including the wire-name quirks
(``workflowtimeout``, ``remedyTriggeredAt``).
"""
from __future__ import annotations

from typing import Any, Dict

import yaml

from .. import GROUP, VERSION


def _policy_rule_schema() -> Dict[str, Any]:
    arr = {"type": "array", "items": {"type": "string"}, "x-kubernetes-list-type": "atomic"}
    return {
        "type": "object",
        "required": ["verbs"],
        "properties": {
            "apiGroups": arr,
            "resources": arr,
            "resourceNames": arr,
            "nonResourceURLs": arr,
            "verbs": arr,
        },
    }


def _artifact_location_schema() -> Dict[str, Any]:
    return {
        "type": "object",
        "properties": {
            "inline": {"type": "string"},
            "file": {
                "type": "object",
                "properties": {"path": {"type": "string"}},
            },
            "url": {
                "type": "object",
                "properties": {
                    "path": {"type": "string"},
                    "verifyCert": {
                        "type": "boolean",
                        "description": (
                            "TLS verification when fetching the workflow; "
                            "verified unless explicitly false (secure default)"
                        ),
                    },
                },
            },
        },
    }


def _workflow_schema(remedy: bool) -> Dict[str, Any]:
    return {
        "type": "object",
        "properties": {
            "generateName": {"type": "string"},
            "workflowtimeout": {"type": "integer"},
            "rbacRules": {
                "type": "array",
                "items": _policy_rule_schema(),
                "x-kubernetes-list-type": "atomic",
            },
            "resource": {
                "type": "object",
                "required": ["namespace", "source"],
                "properties": {
                    "namespace": {"type": "string"},
                    "serviceAccount": {"type": "string"},
                    "source": _artifact_location_schema(),
                },
            },
        },
    }


def _spec_schema() -> Dict[str, Any]:
    return {
        "type": "object",
        "required": ["workflow"],
        "description": (
            "Either repeatAfterSec or schedule must be defined for the "
            "health check to run"
        ),
        "properties": {
            "repeatAfterSec": {"type": "integer"},
            "description": {"type": "string"},
            "level": {
                "type": "string",
                "description": "whether the workflow runs at namespace or cluster level",
            },
            "schedule": {
                "type": "object",
                "properties": {"cron": {"type": "string"}},
            },
            "workflow": _workflow_schema(False),
            "remedyworkflow": _workflow_schema(True),
            "backoffFactor": {"type": "string"},
            "backoffMax": {"type": "integer"},
            "backoffMin": {"type": "integer"},
            "remedyRunsLimit": {"type": "integer"},
            "remedyResetInterval": {"type": "integer"},
        },
    }


def _status_schema() -> Dict[str, Any]:
    t_str = {"type": "string"}
    t_int = {"type": "integer"}
    t_time = {"type": "string", "format": "date-time"}
    return {
        "type": "object",
        "properties": {
            "errorMessage": t_str,
            "remedyErrorMessage": t_str,
            "startedAt": t_time,
            "finishedAt": t_time,
            "lastFailedAt": t_time,
            "remedyTriggeredAt": t_time,
            "remedyFinishedAt": t_time,
            "remedyLastFailedAt": t_time,
            "lastFailedWorkflow": t_str,
            "lastSuccessfulWorkflow": t_str,
            "successCount": t_int,
            "failedCount": t_int,
            "remedySuccessCount": t_int,
            "remedyFailedCount": t_int,
            "remedyTotalRuns": t_int,
            "totalHealthCheckRuns": t_int,
            "status": t_str,
            "remedyStatus": t_str,
        },
    }


def healthcheck_crd() -> Dict[str, Any]:
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"healthchecks.{GROUP}"},
        "spec": {
            "group": GROUP,
            "names": {
                "kind": "HealthCheck",
                "listKind": "HealthCheckList",
                "plural": "healthchecks",
                "singular": "healthcheck",
                "shortNames": ["hc", "hcs"],
            },
            "scope": "Namespaced",
            "versions": [
                {
                    "name": VERSION,
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                    "additionalPrinterColumns": [
                        {"name": "LATEST STATUS", "type": "string",
                         "jsonPath": ".status.status"},
                        {"name": "SUCCESS CNT  ", "type": "string",
                         "jsonPath": ".status.successCount"},
                        {"name": "FAIL CNT", "type": "string",
                         "jsonPath": ".status.failedCount"},
                        {"name": "REMEDY SUCCESS CNT  ", "type": "string",
                         "jsonPath": ".status.remedySuccessCount"},
                        {"name": "REMEDY FAIL CNT", "type": "string",
                         "jsonPath": ".status.remedyFailedCount"},
                        {"name": "Age", "type": "date",
                         "jsonPath": ".metadata.creationTimestamp"},
                    ],
                    "schema": {
                        "openAPIV3Schema": {
                            "type": "object",
                            "properties": {
                                "apiVersion": {"type": "string"},
                                "kind": {"type": "string"},
                                "metadata": {"type": "object"},
                                "spec": _spec_schema(),
                                "status": _status_schema(),
                            },
                        }
                    },
                }
            ],
        },
    }


def healthcheck_crd_yaml() -> str:
    return yaml.safe_dump(healthcheck_crd(), sort_keys=False, default_flow_style=False)


if __name__ == "__main__":  # regenerate: python -m active_monitor_amd.api.crd
    print(healthcheck_crd_yaml())
