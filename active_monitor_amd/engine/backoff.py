"""Inverse-exponential backoff for workflow polling.

Native reimplementation of the small algorithm behind
keikoproj/inverse-exp-backoff v0.1.2 as used by the reference watch loops
(healthcheck_controller.go:613, :801): the first poll is immediate (the loop
body runs before the first ``Next()``), then intervals start at ``max_time``
and decay by ``factor`` toward ``min_time`` until ``timeout`` has elapsed
since ``start``; at that point ``next()`` raises and the caller synthesizes a
``Failed`` phase (healthcheck_controller.go:627-632).

Improvement over the reference library: a sleep never overshoots the
remaining timeout budget, so the synthesized failure lands promptly at the
deadline instead of up to one full interval late.
"""
from __future__ import annotations

import asyncio
import time
from typing import Awaitable, Callable, Optional


class IEBTimeoutError(Exception):
    """Raised by :meth:`InverseExponentialBackoff.next` once the total timeout
    has elapsed."""


class InverseExponentialBackoff:
    def __init__(
        self,
        max_interval: float,
        min_interval: float,
        timeout: float,
        factor: float,
        start: Optional[float] = None,
        sleep: Callable[[float], Awaitable[None]] = asyncio.sleep,
        clock: Callable[[], float] = time.monotonic,
    ):
        if max_interval <= 0 or min_interval <= 0:
            raise ValueError("intervals must be positive")
        if min_interval > max_interval:
            raise ValueError("min_interval must be <= max_interval")
        if not (0 < factor < 1):
            raise ValueError("factor must be in (0, 1)")
        if timeout <= 0:
            raise ValueError("timeout must be positive")
        self.min_interval = min_interval
        self.timeout = timeout
        self.factor = factor
        self._interval = max_interval
        self._sleep = sleep
        self._clock = clock
        self._start = start if start is not None else clock()

    @property
    def current_interval(self) -> float:
        return self._interval

    def remaining(self) -> float:
        return self.timeout - (self._clock() - self._start)

    def peek_interval(self) -> float:
        """The next wait duration, capped at the remaining budget (no decay)."""
        return min(self._interval, max(0.0, self.remaining()))

    def decay(self) -> None:
        self._interval = max(self.min_interval, self._interval * self.factor)

    def check_deadline(self) -> None:
        if self.remaining() <= 0:
            raise IEBTimeoutError(f"timeout of {self.timeout}s exceeded")

    async def next(self) -> None:
        """Sleep the current interval (capped at the remaining budget), decay
        it toward ``min_interval``, and raise :class:`IEBTimeoutError` once the
        deadline has passed."""
        self.check_deadline()
        interval = self.peek_interval()
        self.decay()
        await self._sleep(interval)
        self.check_deadline()


def compute_backoff_params(
    backoff_max: int,
    backoff_min: int,
    backoff_factor: str,
    workflow_timeout: int,
) -> tuple:
    """The reference's backoff-parameter defaulting chain
    (healthcheck_controller.go:575-605):

    - ``max = BackoffMax`` if set, else ``Timeout/2`` floored at 1s,
    - ``min = BackoffMin`` if set, else ``Timeout/60`` floored at 1s,
    - ``factor = float(BackoffFactor)`` if parseable, else 0.5,
    - ``timeout = Workflow.Timeout``.

    All values in seconds. Floors apply only to the derived defaults, matching
    the reference exactly.
    """
    if backoff_max == 0:
        max_time = workflow_timeout // 2
        if max_time <= 0:
            max_time = 1
    else:
        max_time = backoff_max
    if backoff_min == 0:
        min_time = workflow_timeout // 60
        if min_time <= 0:
            min_time = 1
    else:
        min_time = backoff_min
    factor = 0.5
    if backoff_factor != "":
        try:
            factor = float(backoff_factor)
        except ValueError:
            pass  # reference logs and keeps 0.5
    return float(max_time), float(min_time), factor, float(workflow_timeout)
