"""Workflow execution backends (Argo-CR driven, scripted fake, local executor)."""
from .engines import (
    LocalWorkflowEngine,
    ScriptedWorkflowEngine,
    always_fail,
    always_succeed,
    never_complete,
)

__all__ = [
    "LocalWorkflowEngine",
    "ScriptedWorkflowEngine",
    "always_fail",
    "always_succeed",
    "never_complete",
]
