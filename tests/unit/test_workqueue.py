"""Workqueue unit tests (client-go workqueue semantics + the flags extension)."""
import asyncio

import pytest

from active_monitor_amd.engine.workqueue import RateLimiter, WorkQueue


def test_rate_limiter_exponential():
    rl = RateLimiter(base=0.01, cap=1.0)
    assert rl.when("a") == 0.01
    assert rl.when("a") == 0.02
    assert rl.when("a") == 0.04
    for _ in range(20):
        rl.when("a")
    assert rl.when("a") == 1.0  # capped
    rl.forget("a")
    assert rl.when("a") == 0.01


def test_dedup_while_queued(run):
    async def go():
        q = WorkQueue()
        q.add_nowait("k", {"x"})
        q.add_nowait("k", {"y"})  # merges
        key, flags = await q.get()
        assert key == "k" and flags == {"x", "y"}
        q.done_nowait("k")
        return len(q)

    assert run(go()) == 0


def test_readd_while_processing_defers(run):
    """A key added mid-processing re-queues once done, never concurrently."""

    async def go():
        q = WorkQueue()
        q.add_nowait("k")
        key, _ = await q.get()
        q.add_nowait("k", {"timer"})  # while processing
        assert len(q) == 0           # not queued yet
        q.done_nowait("k")
        key2, flags2 = await asyncio.wait_for(q.get(), 1)
        assert key2 == "k" and flags2 == {"timer"}
        q.done_nowait("k")

    run(go())


def test_concurrent_consumers_never_same_key(run):
    async def go():
        q = WorkQueue()
        active = set()
        overlaps = []
        done = 0

        async def worker():
            nonlocal done
            while True:
                item = await q.get()
                if item is None:
                    return
                key, _ = item
                if key in active:
                    overlaps.append(key)
                active.add(key)
                await asyncio.sleep(0.001)
                active.discard(key)
                q.done_nowait(key)
                done += 1

        workers = [asyncio.ensure_future(worker()) for _ in range(8)]
        for i in range(200):
            q.add_nowait(f"key-{i % 5}")
            await asyncio.sleep(0)
        await asyncio.sleep(0.3)
        await q.shutdown()
        await asyncio.gather(*workers)
        assert overlaps == []
        assert done > 0

    run(go())


def test_add_after_fires_and_orders(run):
    async def go():
        q = WorkQueue()
        q.add_after_nowait("late", 0.25)
        q.add_after_nowait("early", 0.05)
        first = await asyncio.wait_for(q.get(), 2)
        q.done_nowait(first[0])
        second = await asyncio.wait_for(q.get(), 2)
        q.done_nowait(second[0])
        return first[0], second[0]

    assert run(go()) == ("early", "late")


def test_shutdown_drains_and_returns_none(run):
    async def go():
        q = WorkQueue()
        q.add_nowait("a")
        await q.shutdown()
        item = await q.get()       # drains the queued item first
        assert item[0] == "a"
        q.done_nowait("a")
        assert await q.get() is None
        q.add_nowait("b")          # post-shutdown adds ignored
        assert await q.get() is None

    run(go())


def test_get_waits_for_add(run):
    async def go():
        q = WorkQueue()
        loop = asyncio.get_running_loop()
        loop.call_later(0.05, q.add_nowait, "k")
        item = await asyncio.wait_for(q.get(), 2)
        return item[0]

    assert run(go()) == "k"


def test_delayed_add_earlier_item_reschedules_waker(run):
    """An earlier-firing delayed item added after a later one must still fire
    first (the waker reschedules)."""

    async def go():
        q = WorkQueue()
        q.add_after_nowait("later", 0.5)
        await asyncio.sleep(0.05)
        q.add_after_nowait("earlier", 0.1)
        first = await asyncio.wait_for(q.get(), 2)
        assert first[0] == "earlier"
        q.done_nowait("earlier")
        second = await asyncio.wait_for(q.get(), 2)
        assert second[0] == "later"
        q.done_nowait("later")

    run(go())
