"""Ecosystem adapter: stock aiohttp client through cueball pooling.

The analog of the reference Agent duck-typing node's http.Agent
(lib/agent.js:30-44): unchanged aiohttp code gets cueball-pooled
sockets with keep-alive reuse.
"""

import asyncio

import pytest

aiohttp = pytest.importorskip("aiohttp")

from cueball_amd.aiohttp_connector import CueballConnector  # noqa: E402
from cueball_amd.testing import MockHttpServer  # noqa: E402

RECOVERY = {"default": {"timeout": 2000, "retries": 3, "delay": 100,
                        "maxDelay": 2000}}


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_get_through_cueball_pool():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        connector = CueballConnector(recovery=RECOVERY, spares=2,
                                     maximum=4)
        async with aiohttp.ClientSession(connector=connector) as sess:
            async with sess.get(
                    "http://127.0.0.1:%d/hello" % srv.port) as resp:
                assert resp.status == 200
                body_ = await resp.json(content_type=None)
                assert body_["path"] == "/hello"
            stats = connector.pool_stats()
            key = "127.0.0.1:%d" % srv.port
            assert stats[key]["counters"]["claim"] == 1
        srv.stop()

    run(body())


def test_keepalive_reuse_and_concurrency():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        connector = CueballConnector(recovery=RECOVERY, spares=2,
                                     maximum=4)
        async with aiohttp.ClientSession(connector=connector) as sess:
            url = "http://127.0.0.1:%d/x" % srv.port

            async def one(i):
                async with sess.get(url) as resp:
                    assert resp.status == 200
                    await resp.read()

            # 20 sequential requests: sockets must be reused
            for i in range(20):
                await one(i)
            assert srv.request_count == 20
            # far fewer connections than requests => keep-alive reuse
            assert srv.conn_count <= 4
            assert max(srv.requests_per_conn) > 1

            # concurrent burst within pool maximum
            await asyncio.gather(*[one(i) for i in range(16)])
            assert srv.request_count == 36
            assert srv.conn_count <= 8  # bounded by pool maximum=4 (+
            # reconnect slack)
        srv.stop()

    run(body())


def test_claims_shed_on_server_down():
    async def body():
        # connect to a port nobody listens on: claims must fail with a
        # ClientConnectionError (mapped from the cueball error), not
        # hang
        connector = CueballConnector(
            recovery={"default": {"timeout": 200, "retries": 1,
                                  "delay": 10, "maxDelay": 50}},
            spares=1, maximum=2, claim_timeout=2000)
        # an unused port: bind and close to find a free one
        import socket as mod_socket
        s = mod_socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        async with aiohttp.ClientSession(connector=connector) as sess:
            with pytest.raises(aiohttp.ClientConnectionError):
                async with sess.get("http://127.0.0.1:%d/x" % port,
                                    timeout=aiohttp.ClientTimeout(
                                        total=10)):
                    pass

    run(body())


def test_server_close_header_disposes_socket():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        connector = CueballConnector(recovery=RECOVERY, spares=1,
                                     maximum=2)
        async with aiohttp.ClientSession(connector=connector) as sess:
            # /close responds with Connection: close; the handle must
            # be closed (socket disposed), then a new request works on
            # a fresh socket
            async with sess.get(
                    "http://127.0.0.1:%d/close" % srv.port) as resp:
                assert resp.status == 200
                await resp.read()
            await asyncio.sleep(0.1)
            async with sess.get(
                    "http://127.0.0.1:%d/x" % srv.port) as resp:
                assert resp.status == 200
                await resp.read()
            assert srv.conn_count >= 2
        srv.stop()

    run(body())
