"""httpx ecosystem adapter: stock httpx client over cueball pooling."""

import asyncio

import pytest

httpx = pytest.importorskip("httpx")

from cueball_amd.httpx_transport import CueballTransport  # noqa: E402
from cueball_amd.testing import MockHttpServer  # noqa: E402

RECOVERY = {"default": {"timeout": 2000, "retries": 3, "delay": 100,
                        "maxDelay": 2000}}


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_get_and_keepalive_reuse():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        transport = CueballTransport(recovery=RECOVERY, spares=1,
                                     maximum=2)
        async with httpx.AsyncClient(transport=transport) as client:
            for i in range(10):
                r = await client.get(
                    "http://127.0.0.1:%d/item/%d" % (srv.port, i))
                assert r.status_code == 200
                assert r.json()["path"] == "/item/%d" % i
            assert srv.request_count == 10
            assert srv.conn_count <= 2
            assert max(srv.requests_per_conn) > 1
            stats = transport.pool_stats()
            assert any(v["counters"]["claim"] == 10
                       for v in stats.values())
        srv.stop()

    run(body())


def test_post_body_and_query():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        transport = CueballTransport(recovery=RECOVERY)
        async with httpx.AsyncClient(transport=transport) as client:
            r = await client.post(
                "http://127.0.0.1:%d/echo" % srv.port,
                content=b"hello-body")
            assert r.status_code == 200
            assert r.content == b"hello-body"
            # query strings pass through in the request target
            r2 = await client.get(
                "http://127.0.0.1:%d/path?q=1&r=2" % srv.port)
            assert r2.status_code == 200
            assert r2.json()["path"] == "/path?q=1&r=2"
        srv.stop()

    run(body())


def test_connect_error_maps():
    async def body():
        import socket as mod_socket
        s = mod_socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        transport = CueballTransport(
            recovery={"default": {"timeout": 200, "retries": 1,
                                  "delay": 10, "maxDelay": 50}},
            agent_options={"errorOnEmpty": False})
        async with httpx.AsyncClient(transport=transport,
                                     timeout=10) as client:
            with pytest.raises(httpx.TransportError):
                await client.get("http://127.0.0.1:%d/x" % port)

    run(body())
