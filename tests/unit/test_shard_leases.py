"""ShardCoordinator lease-protocol edge cases (deterministic, store-level):
the preferredHolder priority window, reclaim handshake on renewal, home
shard keeping its lease despite a reclaim request, and immediate-expiry
release on clean stop."""
import asyncio

from active_monitor_amd.api.types import k8s_now
from active_monitor_amd.engine.shards import LEASE_API_VERSION, ShardCoordinator
from active_monitor_amd.kube import MemoryApiServer, MemoryClient


def _coord(server, idx, ident, count=2, lease=4.0):
    return ShardCoordinator(
        MemoryClient(server), namespace="health", shard_index=idx,
        shard_count=count, identity=ident,
        lease_duration=lease, renew_interval=0.1,
    )


def _lease(server, coord, shard):
    return server.get(LEASE_API_VERSION, "Lease", "health", coord._lease_name(shard))


def test_preferred_release_window_blocks_third_parties(run):
    async def go():
        server = MemoryApiServer()
        a = _coord(server, 0, "proc-a")
        # b and c are FOREIGN adopters of shard 1 (their home is shard 0) —
        # only a non-home holder honors a reclaim request
        b = _coord(server, 0, "proc-b")
        c = _coord(server, 0, "proc-c")

        # b adopts shard 1, then gracefully releases it FOR proc-a
        assert await b._try_acquire(1)
        lease = _lease(server, b, 1)
        lease["spec"]["preferredHolder"] = "proc-a"
        server.update(lease)
        b.owned.add(1)
        assert not await b._renew_owned(1)  # reclaim honored: released

        spec = _lease(server, b, 1)["spec"]
        assert spec["holderIdentity"] == ""
        assert spec["preferredHolder"] == "proc-a"

        # within the priority window: only the preferred owner may take it
        assert not await c._try_acquire(1), "third party jumped the window"
        assert await a._try_acquire(1)
        assert _lease(server, a, 1)["spec"]["holderIdentity"] == "proc-a"
        # reclaim satisfied → marker cleared
        assert not _lease(server, a, 1)["spec"].get("preferredHolder")

    run(go(), timeout=20)


def test_home_shard_ignores_reclaim_requests(run):
    """preferredHolder only moves ADOPTED shards; a home shard never
    releases its own keyspace to a request."""

    async def go():
        server = MemoryApiServer()
        a = _coord(server, 0, "proc-a")
        assert await a._try_acquire(0)
        a.owned.add(0)
        lease = _lease(server, a, 0)
        lease["spec"]["preferredHolder"] = "intruder"
        server.update(lease)
        assert await a._renew_owned(0)  # renewed, not released
        assert _lease(server, a, 0)["spec"]["holderIdentity"] == "proc-a"

    run(go(), timeout=20)


def test_clean_stop_releases_without_expiry_wait(run):
    async def go():
        server = MemoryApiServer()
        a = _coord(server, 0, "proc-a", lease=30.0)  # long TTL
        await a.start()
        await a.stop()
        spec = _lease(server, a, 0)["spec"]
        assert spec["holderIdentity"] == ""
        # and a successor acquires instantly despite the long TTL
        b = _coord(server, 0, "proc-b", lease=30.0)
        assert await b._try_acquire(0)

    run(go(), timeout=20)


def test_expired_foreign_lease_is_adoptable(run):
    async def go():
        server = MemoryApiServer()
        dead = _coord(server, 1, "dead-proc", lease=2.0)
        assert await dead._try_acquire(1)
        # backdate the renew far past the TTL (simulated crash + time)
        lease = _lease(server, dead, 1)
        lease["spec"]["renewTime"] = "2000-01-01T00:00:00Z"
        server.update(lease)

        a = _coord(server, 0, "proc-a", lease=2.0)
        assert await a._try_acquire(1)
        assert _lease(server, a, 1)["spec"]["holderIdentity"] == "proc-a"

    run(go(), timeout=20)
