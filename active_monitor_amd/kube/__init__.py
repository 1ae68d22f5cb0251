"""Kubernetes client stack: in-memory apiserver, async client facade, errors."""
from .client import EventRecorder, FakeRecorder, KubeClient, MemoryClient
from .errors import (
    AlreadyExistsError,
    ApiError,
    ConflictError,
    InvalidError,
    NotFoundError,
    contains_equal_fold_substring,
    ignore_not_found,
    is_conflict,
    is_not_found,
    is_storage_error,
)
from .memory import MemoryApiServer, Subscription
from .registry import DEFAULT_REGISTRY, WF_API_VERSION, WF_KIND, WF_PLURAL, Registry, ResourceInfo

__all__ = [
    "AlreadyExistsError",
    "ApiError",
    "ConflictError",
    "DEFAULT_REGISTRY",
    "EventRecorder",
    "FakeRecorder",
    "InvalidError",
    "KubeClient",
    "MemoryApiServer",
    "MemoryClient",
    "NotFoundError",
    "Registry",
    "ResourceInfo",
    "Subscription",
    "WF_API_VERSION",
    "WF_KIND",
    "WF_PLURAL",
    "contains_equal_fold_substring",
    "ignore_not_found",
    "is_conflict",
    "is_not_found",
    "is_storage_error",
]
