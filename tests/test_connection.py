"""TcpConnection tests over real sockets."""

import asyncio

import pytest

from cueball_amd.connection import TcpConnection, tcp_constructor


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


async def echo_server():
    async def handle(reader, writer):
        while True:
            data = await reader.read(4096)
            if not data:
                break
            writer.write(data)
            await writer.drain()
        writer.close()

    srv = await asyncio.start_server(handle, "127.0.0.1", 0)
    return srv, srv.sockets[0].getsockname()[1]


def test_connect_write_read_destroy():
    async def body():
        srv, port = await echo_server()
        loop = asyncio.get_running_loop()
        conn = TcpConnection({"address": "127.0.0.1", "port": port},
                             loop=loop)
        connected = loop.create_future()
        conn.on("connect", lambda: connected.set_result(None))
        await asyncio.wait_for(connected, 5)
        assert conn.connected
        assert conn.local_port is not None

        got = loop.create_future()
        conn.on("data", lambda d: got.done() or got.set_result(d))
        conn.write(b"abc")
        assert await asyncio.wait_for(got, 5) == b"abc"

        closed = loop.create_future()
        conn.on("close", lambda: closed.set_result(None))
        conn.destroy()
        await asyncio.wait_for(closed, 5)
        assert conn.dead
        srv.close()

    run(body())


def test_connect_refused_emits_connect_error():
    async def body():
        srv, port = await echo_server()
        srv.close()
        await asyncio.sleep(0.05)
        loop = asyncio.get_running_loop()
        conn = TcpConnection({"address": "127.0.0.1", "port": port},
                             loop=loop)
        errs = []
        closed = loop.create_future()
        conn.on("connectError", errs.append)
        conn.on("close", lambda: closed.set_result(None))
        await asyncio.wait_for(closed, 5)
        assert len(errs) == 1
        assert isinstance(errs[0], OSError)

    run(body())


def test_connect_timeout():
    async def body():
        loop = asyncio.get_running_loop()

        # deterministic "never completes" connect
        async def never(*a, **k):
            await asyncio.sleep(3600)

        loop.create_connection = never  # type: ignore[method-assign]
        conn = TcpConnection({"address": "10.255.255.1", "port": 80},
                             loop=loop, connect_timeout=200)
        timeouts = []
        closed = loop.create_future()
        conn.on("connectTimeout", lambda: timeouts.append(1))
        conn.on("close", lambda: closed.set_result(None))
        await asyncio.wait_for(closed, 10)
        assert timeouts == [1]

    run(body())


def test_peer_close_emits_close_once():
    async def body():
        async def handle(reader, writer):
            writer.close()

        srv = await asyncio.start_server(handle, "127.0.0.1", 0)
        port = srv.sockets[0].getsockname()[1]
        loop = asyncio.get_running_loop()
        conn = TcpConnection({"address": "127.0.0.1", "port": port},
                             loop=loop)
        closes = []
        done = loop.create_future()

        def on_close():
            closes.append(1)
            if not done.done():
                done.set_result(None)

        conn.on("close", on_close)
        await asyncio.wait_for(done, 5)
        await asyncio.sleep(0.05)
        assert closes == [1]
        conn.destroy()
        await asyncio.sleep(0.05)
        assert closes == [1]  # destroy after close: no second emit
        srv.close()

    run(body())


def test_constructor_factory():
    async def body():
        srv, port = await echo_server()
        loop = asyncio.get_running_loop()
        ctor = tcp_constructor(loop=loop)
        conn = ctor({"address": "127.0.0.1", "port": port, "key": "b1"})
        connected = loop.create_future()
        conn.on("connect", lambda: connected.set_result(None))
        await asyncio.wait_for(connected, 5)
        conn.set_unwanted()
        assert conn.unwanted
        conn.ref()
        conn.unref()
        conn.destroy()
        await asyncio.sleep(0.05)  # let the server handler see EOF
        srv.close()
        await srv.wait_closed()

    run(body())


def test_pool_claim_over_ipv6_loopback():
    """End-to-end claim -> echo -> release over a real ::1 TCP socket
    (the reference supports v6 backends throughout; srvKey normalizes
    v6 text forms, lib/resolver.js:1157-1171)."""
    import asyncio

    from cueball_amd.connection import tcp_constructor
    from cueball_amd.pool import ConnectionPool
    from cueball_amd.resolver import StaticIpResolver

    async def body():
        async def on_conn(reader, writer):
            data = await reader.read(64)
            writer.write(b"echo:" + data)
            await writer.drain()
            writer.close()

        server = await asyncio.start_server(on_conn, "::1", 0)
        port = server.sockets[0].getsockname()[1]

        res = StaticIpResolver({"backends": [
            {"address": "::1", "port": port}]})
        pool = ConnectionPool({
            "domain": "v6.test", "resolver": res,
            "constructor": tcp_constructor(),
            "spares": 1, "maximum": 2,
            "recovery": {"default": {"timeout": 2000, "retries": 2,
                                     "delay": 100, "maxDelay": 500}},
        })
        res.start()
        handle, conn = await asyncio.wait_for(pool.claim_async(), 10)
        got = asyncio.get_running_loop().create_future()
        conn.on("data", lambda d: (not got.done()) and got.set_result(d))
        conn.write(b"ping6")
        data = await asyncio.wait_for(got, 5)
        assert data == b"echo:ping6"
        handle.close()

        fut = asyncio.get_running_loop().create_future()
        pool.on("stateChanged", lambda st: st == "stopped"
                and not fut.done() and fut.set_result(None))
        pool.stop()
        await asyncio.wait_for(fut, 10)
        server.close()
        await server.wait_closed()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(body())
    finally:
        loop.close()
