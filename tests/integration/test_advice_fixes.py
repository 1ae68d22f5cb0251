"""Regression tests for the round-1 advisor findings:

1. medium — blocking URL artifact reads froze the event loop
   (store/artifacts.py): reads now run on a worker thread with a timeout.
2. medium — lost leadership was silently dropped (engine/manager.py): it is
   now fatal for the replica (manager.fatal, entrypoint exits 1).
3. low — watch 410/oversized events: covered in test_watch_conformance.py.
4. low — watchhub DELETED popped the seq counter (engine/watchhub.py): it now
   increments, so late waiters return immediately and discover the NotFound.
"""
import asyncio
import socket
import threading

import pytest

import active_monitor_amd.engine as engine_pkg
import active_monitor_amd.engine.manager as manager_mod
from active_monitor_amd import API_VERSION
from active_monitor_amd.api import HealthCheck
from active_monitor_amd.engine import Manager, parse_workflow_from_healthcheck_async
from active_monitor_amd.engine import parse as parse_mod
from active_monitor_amd.engine.leader import LeaderElector
from active_monitor_amd.engine.watchhub import WorkflowWatchHub
from active_monitor_amd.kube import MemoryApiServer, MemoryClient
from active_monitor_amd.store import ArtifactReadError
from active_monitor_amd.api.types import k8s_now

from .conftest import make_hc


# ---------------------------------------------------------------------------
# 1. non-blocking artifact reads
# ---------------------------------------------------------------------------


class _HangingHttpServer:
    """Accepts TCP connections and never answers — the pathological URL
    source from the advisor's finding."""

    def __enter__(self):
        self.sock = socket.socket()
        self.sock.bind(("127.0.0.1", 0))
        self.sock.listen(8)
        self.port = self.sock.getsockname()[1]
        self._conns = []
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._accept_loop, daemon=True)
        self._thread.start()
        return self

    def _accept_loop(self):
        self.sock.settimeout(0.2)
        while not self._stop.is_set():
            try:
                conn, _ = self.sock.accept()
                self._conns.append(conn)  # hold open, never respond
            except socket.timeout:
                continue
            except OSError:
                return

    def __exit__(self, *exc):
        self._stop.set()
        for c in self._conns:
            try:
                c.close()
            except OSError:
                pass
        self.sock.close()


def _url_hc(url):
    d = make_hc(name="url-check")
    d["spec"]["workflow"]["resource"]["source"] = {"url": {"path": url}}
    return HealthCheck.from_dict(d)


def test_hung_url_source_does_not_stall_event_loop(run, monkeypatch):
    """While one CR's URL source hangs, the loop keeps making progress, and
    the read errors out at the deadline instead of hanging forever."""
    monkeypatch.setattr(parse_mod, "ARTIFACT_READ_TIMEOUT", 0.8)

    async def go():
        with _HangingHttpServer() as srv:
            hc = _url_hc(f"http://127.0.0.1:{srv.port}/wf.yaml")

            ticks = 0

            async def ticker():
                nonlocal ticks
                while True:
                    ticks += 1
                    await asyncio.sleep(0.01)

            t = asyncio.ensure_future(ticker())
            try:
                with pytest.raises(ArtifactReadError):
                    await parse_workflow_from_healthcheck_async(hc)
            finally:
                t.cancel()
            # ~0.8s of hung read; a frozen loop would have ticked ~0 times
            assert ticks > 20, f"event loop stalled during URL read ({ticks} ticks)"

    run(go(), timeout=30)


def test_url_reader_socket_timeout(run):
    """The reader itself enforces a socket timeout (no infinite GET)."""
    from active_monitor_amd.api.types import URLArtifact
    from active_monitor_amd.store import URLReader

    def go_sync():
        with _HangingHttpServer() as srv:
            reader = URLReader(
                URLArtifact(path=f"http://127.0.0.1:{srv.port}/x"), timeout=0.5
            )
            with pytest.raises(ArtifactReadError):
                reader.read()

    go_sync()


# ---------------------------------------------------------------------------
# 2. lost leadership is fatal
# ---------------------------------------------------------------------------


class _FastElector(LeaderElector):
    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        # lease_duration must exceed k8s timestamp resolution (1s) with slack,
        # or a freshly-stolen lease immediately looks expired and gets
        # re-acquired; only the renew cadence needs to be fast here
        self.lease_duration = 5.0
        self.renew_interval = 0.1
        self.retry_interval = 0.05


def test_lost_lease_is_fatal_for_the_manager(run, monkeypatch):
    """Steal the lease out from under a running manager: its renew loop must
    fail and set manager.fatal (the reference exits on lost leadership,
    cmd/main.go:87-88) — not keep reconciling as a deposed split-brain."""
    monkeypatch.setattr(manager_mod, "LeaderElector", _FastElector)

    async def go():
        server = MemoryApiServer()
        mgr = Manager(
            MemoryClient(server), max_workers=1,
            leader_elect=True, leader_identity="replica-a", namespace="health",
        )
        await mgr.start()
        try:
            assert not mgr.fatal.is_set()
            # another replica takes the lease (fresh renewTime, new holder)
            lease = server.get("coordination.k8s.io/v1", "Lease", "health",
                               "689451f8.keikoproj.io")
            lease["spec"]["holderIdentity"] = "replica-b"
            lease["spec"]["renewTime"] = k8s_now()
            server.update(lease)

            await asyncio.wait_for(mgr.fatal.wait(), 5)
            assert "lost leadership" in (mgr.fatal_reason or "")
            assert mgr.ready is False  # readyz goes red on a deposed replica
        finally:
            await mgr.stop()

    run(go(), timeout=30)


def test_entrypoint_exits_nonzero_on_fatal(run, monkeypatch):
    """cmd/main run() returns 1 when the manager reports a fatal condition
    (previously the renew task's exception was silently dropped)."""
    from active_monitor_amd.cmd.main import build_parser, run as cmd_run

    class FakeManager:
        def __init__(self, *a, **kw):
            self.fatal = asyncio.Event()
            self.fatal_reason = "lost leadership lease (test)"
            self.ready = True

        async def start(self):
            asyncio.get_running_loop().call_later(0.05, self.fatal.set)

        async def stop(self):
            pass

    monkeypatch.setattr(engine_pkg, "Manager", FakeManager)
    args = build_parser().parse_args(
        ["--backend", "memory", "--workflow-engine", "none",
         "--metrics-bind-address", "0", "--health-probe-bind-address", "0"]
    )
    rc = run(cmd_run(args), timeout=20)
    assert rc == 1


# ---------------------------------------------------------------------------
# 4. watchhub DELETED keeps (and bumps) the seq counter
# ---------------------------------------------------------------------------


def test_watchhub_delete_bumps_seq_for_late_waiters(run):
    async def go():
        server = MemoryApiServer()
        client = MemoryClient(server)
        hub = WorkflowWatchHub(client)
        await hub.start()
        try:
            wf = {
                "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
                "metadata": {"name": "wf-1", "namespace": "health"},
                "spec": {},
            }
            await client.create(wf)
            # wait until the ADDED event landed
            deadline = asyncio.get_running_loop().time() + 5
            while hub.seq("health", "wf-1") == 0:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.01)
            before_delete = hub.seq("health", "wf-1")

            await client.delete("argoproj.io/v1alpha1", "Workflow", "health", "wf-1")
            # a watcher that polled before the delete and registers after it
            # must return immediately (seq advanced), not sleep out the IEB
            t0 = asyncio.get_running_loop().time()
            got = await hub.wait_change("health", "wf-1", timeout=5.0,
                                        since=before_delete)
            assert got is not None
            assert asyncio.get_running_loop().time() - t0 < 1.0

            # lazy pruning: after the TTL the seq entry is dropped
            hub._DELETED_TTL = 0.0
            await client.create({
                "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
                "metadata": {"name": "wf-2", "namespace": "health"}, "spec": {},
            })
            deadline = asyncio.get_running_loop().time() + 5
            while hub.seq("health", "wf-1") != 0:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.01)
        finally:
            await hub.stop()

    run(go(), timeout=30)
