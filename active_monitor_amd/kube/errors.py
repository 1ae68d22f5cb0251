"""Kubernetes-shaped API errors.

Mirrors the apimachinery error taxonomy the reference relies on:
``IsNotFound`` swallowing (healthcheck_controller.go:144-149), conflict-retried
updates (:208-215, :1447-1454), and the ``StorageError: invalid object``
substring check (:1473-1478).
"""
from __future__ import annotations

from typing import Optional


class ApiError(Exception):
    """Base error with an HTTP-like status code."""

    code = 500
    reason = "InternalError"

    def __init__(self, message: str = ""):
        super().__init__(message or self.reason)
        self.message = message or self.reason


class NotFoundError(ApiError):
    code = 404
    reason = "NotFound"


class AlreadyExistsError(ApiError):
    code = 409
    reason = "AlreadyExists"


class ConflictError(ApiError):
    """Stale resourceVersion on update (optimistic-concurrency conflict)."""

    code = 409
    reason = "Conflict"


class InvalidError(ApiError):
    code = 422
    reason = "Invalid"


class ExpiredError(ApiError):
    """410 Gone: a watch resourceVersion older than the server's retained
    event window (the apiserver's 'too old resource version')."""

    code = 410
    reason = "Expired"


def is_not_found(err: Optional[BaseException]) -> bool:
    return isinstance(err, NotFoundError)


def is_conflict(err: Optional[BaseException]) -> bool:
    return isinstance(err, ConflictError)


def ignore_not_found(err: Optional[BaseException]) -> Optional[BaseException]:
    """Return None for NotFound errors, the error otherwise
    (reference ignoreNotFound, healthcheck_controller.go:144-149)."""
    if err is None or is_not_found(err):
        return None
    return err


def is_storage_error(err: Optional[BaseException]) -> bool:
    """Substring match for the apiserver's transient 'StorageError: invalid
    object' seen when racing a delete (healthcheck_controller.go:1473-1478)."""
    if err is None:
        return False
    return contains_equal_fold_substring(str(err), "StorageError: invalid object")


def contains_equal_fold_substring(s: str, substr: str) -> bool:
    """Case-insensitive substring test (reference ContainsEqualFoldSubstring,
    healthcheck_controller.go:1464-1471)."""
    return substr.lower() in s.lower()
