"""Rate-limited, deduplicating work queue (client-go workqueue semantics).

The reference relies on controller-runtime's workqueue: per-key dedup (a key
queued while being processed is re-queued once done, never processed twice
concurrently), delayed adds, and per-item exponential failure backoff. This is
the asyncio equivalent, with one extension: each item carries a ``flags`` set
that merges on dedup — used to mark timer-fired repeats so the reconciler can
distinguish them from informer-driven reconciles (repeats flow
through the queue here, unlike the reference's bare time.AfterFunc goroutines,
so ``MaxConcurrentReconciles`` genuinely bounds all reconcile work).

Implementation note: every mutation is a plain synchronous method — the queue
is single-event-loop owned, so critical sections contain no await points and
are atomic by construction. No async locks, no lost-wakeup hazards, and no
lock overhead on the reconcile hot path. Waiting consumers park on futures.
"""
from __future__ import annotations

import asyncio
import heapq
import time
from collections import deque
from typing import Deque, Dict, Hashable, List, Optional, Set, Tuple


class RateLimiter:
    """Per-item exponential failure backoff (client-go
    ItemExponentialFailureRateLimiter defaults: 5ms base, 1000s cap)."""

    def __init__(self, base: float = 0.005, cap: float = 1000.0):
        self.base = base
        self.cap = cap
        self._failures: Dict[Hashable, int] = {}

    def when(self, item: Hashable) -> float:
        n = self._failures.get(item, 0)
        self._failures[item] = n + 1
        return min(self.cap, self.base * (2 ** n))

    def forget(self, item: Hashable) -> None:
        self._failures.pop(item, None)

    def retries(self, item: Hashable) -> int:
        return self._failures.get(item, 0)


class WorkQueue:
    """Async dedup queue. Items are hashable keys; ``flags`` merge on dedup."""

    def __init__(self, rate_limiter: Optional[RateLimiter] = None):
        self._queue: Deque[Hashable] = deque()
        self._queued: Dict[Hashable, Set[str]] = {}
        self._processing: Dict[Hashable, Set[str]] = {}
        self._dirty: Set[Hashable] = set()
        self._waiters: Deque[asyncio.Future] = deque()
        self._shutting_down = False
        self.rate_limiter = rate_limiter or RateLimiter()
        self._delayed: List[Tuple[float, int, Hashable, Set[str]]] = []
        self._seq = 0
        self._delay_waker: Optional[asyncio.TimerHandle] = None

    # -- wakeups -----------------------------------------------------------

    def _wake_one(self) -> None:
        while self._waiters:
            fut = self._waiters.popleft()
            if not fut.done():
                fut.set_result(None)
                return

    def _wake_all(self) -> None:
        while self._waiters:
            fut = self._waiters.popleft()
            if not fut.done():
                fut.set_result(None)

    # -- core (synchronous mutations; safe to call from timer callbacks) ---

    def add_nowait(self, key: Hashable, flags: Optional[Set[str]] = None) -> None:
        flags = set(flags or ())
        if self._shutting_down:
            return
        if key in self._processing:
            self._dirty.add(key)
            self._processing[key] |= flags
            return
        if key in self._queued:
            self._queued[key] |= flags
            return
        self._queued[key] = flags
        self._queue.append(key)
        self._wake_one()

    async def add(self, key: Hashable, flags: Optional[Set[str]] = None) -> None:
        self.add_nowait(key, flags)

    async def get(self) -> Optional[Tuple[Hashable, Set[str]]]:
        """Pop the next key, or None once shut down and drained."""
        while True:
            if self._queue:
                key = self._queue.popleft()
                flags = self._queued.pop(key)
                self._processing[key] = set()
                return key, flags
            if self._shutting_down:
                return None
            fut = asyncio.get_running_loop().create_future()
            self._waiters.append(fut)
            try:
                await fut
            except asyncio.CancelledError:
                if fut in self._waiters:
                    self._waiters.remove(fut)
                # pass a pending wakeup to another waiter instead of losing it
                if fut.done() and not fut.cancelled():
                    self._wake_one()
                raise

    def done_nowait(self, key: Hashable) -> None:
        flags = self._processing.pop(key, set())
        if key in self._dirty:
            self._dirty.discard(key)
            if not self._shutting_down:
                self._queued[key] = flags
                self._queue.append(key)
                self._wake_one()

    async def done(self, key: Hashable) -> None:
        self.done_nowait(key)

    # -- delayed / rate-limited adds ----------------------------------------

    def add_after_nowait(self, key: Hashable, delay: float, flags: Optional[Set[str]] = None) -> None:
        if self._shutting_down:
            return
        if delay <= 0:
            self.add_nowait(key, flags)
            return
        self._seq += 1
        heapq.heappush(
            self._delayed, (time.monotonic() + delay, self._seq, key, set(flags or ()))
        )
        self._reschedule_delay_waker()

    async def add_after(self, key: Hashable, delay: float, flags: Optional[Set[str]] = None) -> None:
        self.add_after_nowait(key, delay, flags)

    def _reschedule_delay_waker(self) -> None:
        if not self._delayed or self._shutting_down:
            return
        when = self._delayed[0][0]
        if self._delay_waker is not None:
            if self._delay_waker.when() <= when + 1e-9:
                return  # existing waker fires early enough
            self._delay_waker.cancel()
        loop = asyncio.get_event_loop()
        self._delay_waker = loop.call_at(
            loop.time() + max(0.0, when - time.monotonic()), self._drain_delayed
        )

    def _drain_delayed(self) -> None:
        self._delay_waker = None
        now = time.monotonic()
        while self._delayed and self._delayed[0][0] <= now:
            _, _, key, flags = heapq.heappop(self._delayed)
            self.add_nowait(key, flags)
        self._reschedule_delay_waker()

    async def add_rate_limited(self, key: Hashable, flags: Optional[Set[str]] = None) -> None:
        self.add_after_nowait(key, self.rate_limiter.when(key), flags)

    def forget(self, key: Hashable) -> None:
        self.rate_limiter.forget(key)

    # -- lifecycle ----------------------------------------------------------

    async def shutdown(self) -> None:
        self._shutting_down = True
        if self._delay_waker is not None:
            self._delay_waker.cancel()
            self._delay_waker = None
        self._wake_all()

    @property
    def is_shutting_down(self) -> bool:
        return self._shutting_down

    def __len__(self) -> int:
        return len(self._queue)

    def __bool__(self) -> bool:
        return True  # an empty queue is still a queue
