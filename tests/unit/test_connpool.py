"""Unit tests for the plain-HTTP connection pool behind HttpClient's unary
fast path (kube/http.py _ConnPool): keep-alive reuse, concurrency cap with
FIFO waiters, chunked framing, stale-connection retry at the client layer,
and fail-fast on close."""
import asyncio

import pytest

from active_monitor_amd.kube.http import _ConnPool


class MiniServer:
    """Scriptable HTTP/1.1 server: counts connections, can close after N
    responses, can answer chunked."""

    def __init__(self, close_after=None, chunked=False, delay=0.0):
        self.close_after = close_after
        self.chunked = chunked
        self.delay = delay
        self.connections = 0
        self.requests = 0

    async def __aenter__(self):
        self._srv = await asyncio.start_server(self._handle, "127.0.0.1", 0)
        self.port = self._srv.sockets[0].getsockname()[1]
        return self

    async def __aexit__(self, *exc):
        self._srv.close()

    async def _handle(self, reader, writer):
        self.connections += 1
        served = 0
        try:
            while True:
                line = await reader.readline()
                if not line or line in (b"\r\n", b"\n"):
                    return
                length = 0
                while True:
                    h = await reader.readline()
                    if h in (b"\r\n", b"\n", b""):
                        break
                    if h.lower().startswith(b"content-length:"):
                        length = int(h.split(b":", 1)[1])
                if length:
                    await reader.readexactly(length)
                self.requests += 1
                if self.delay:
                    await asyncio.sleep(self.delay)
                body = b'{"ok":true}'
                if self.chunked:
                    writer.write(
                        b"HTTP/1.1 200 OK\r\nTransfer-Encoding: chunked\r\n\r\n"
                        + hex(len(body))[2:].encode() + b"\r\n" + body + b"\r\n0\r\n\r\n"
                    )
                else:
                    writer.write(
                        b"HTTP/1.1 200 OK\r\nContent-Length: "
                        + str(len(body)).encode() + b"\r\n\r\n" + body
                    )
                await writer.drain()
                served += 1
                if self.close_after is not None and served >= self.close_after:
                    return
        except (ConnectionError, asyncio.IncompleteReadError):
            pass
        finally:
            writer.close()


def test_keepalive_reuses_one_connection(run):
    async def go():
        async with MiniServer() as srv:
            pool = _ConnPool("127.0.0.1", srv.port)
            for _ in range(20):
                status, body = await pool.request("GET", "/x", "", None)
                assert status == 200 and body == b'{"ok":true}'
            pool.close()
            assert srv.connections == 1, srv.connections
            assert srv.requests == 20

    run(go(), timeout=20)


def test_chunked_framing(run):
    async def go():
        async with MiniServer(chunked=True) as srv:
            pool = _ConnPool("127.0.0.1", srv.port)
            status, body = await pool.request("GET", "/x", "", None)
            assert status == 200 and body == b'{"ok":true}'
            pool.close()

    run(go(), timeout=20)


def test_concurrency_cap_with_fifo_waiters(run):
    async def go():
        async with MiniServer(delay=0.05) as srv:
            pool = _ConnPool("127.0.0.1", srv.port, max_conns=2)
            results = await asyncio.gather(
                *(pool.request("GET", f"/{i}", "", None) for i in range(6))
            )
            assert all(s == 200 for s, _ in results)
            assert srv.connections <= 2, srv.connections
            pool.close()

    run(go(), timeout=20)


def test_server_close_surfaces_as_connection_error(run):
    """A connection the server closed mid-keep-alive raises; HttpClient's
    layer above retries once on a fresh connection (covered end-to-end by
    the apiserver-outage test)."""

    async def go():
        async with MiniServer(close_after=1) as srv:
            pool = _ConnPool("127.0.0.1", srv.port)
            status, _ = await pool.request("GET", "/a", "", None)
            assert status == 200
            with pytest.raises((ConnectionError, asyncio.IncompleteReadError)):
                await pool.request("GET", "/b", "", None)
            # and a fresh request after the failure works (new connection)
            status, _ = await pool.request("GET", "/c", "", None)
            assert status == 200
            pool.close()

    run(go(), timeout=20)


def test_close_fails_queued_waiters(run):
    async def go():
        async with MiniServer(delay=0.2) as srv:
            pool = _ConnPool("127.0.0.1", srv.port, max_conns=1)
            t1 = asyncio.ensure_future(pool.request("GET", "/1", "", None))
            await asyncio.sleep(0.05)  # t1 holds the only connection
            t2 = asyncio.ensure_future(pool.request("GET", "/2", "", None))
            await asyncio.sleep(0.05)  # t2 queued
            pool.close()
            with pytest.raises(ConnectionError):
                await t2
            await t1  # in-flight request completes normally

    run(go(), timeout=20)
