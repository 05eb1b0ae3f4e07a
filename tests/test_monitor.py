"""Kang monitor + metrics endpoint tests (port of test/monitor.test.js).

The snapshot is scraped over real HTTP — using the framework's own
HttpAgent as the client.
"""

import asyncio
import json

import pytest

from cueball_amd.agent import HttpAgent
from cueball_amd.connection_set import ConnectionSet
from cueball_amd.kang import KangServer
from cueball_amd.metrics import create_collector
from cueball_amd.pool import ConnectionPool
from cueball_amd.pool_monitor import PoolMonitor
from cueball_amd.resolver import ResolverFSM
from cueball_amd.testing import DummyConnection, DummyResolver

RECOVERY = {"default": {"timeout": 1000, "retries": 2, "delay": 50}}


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


async def scrape(agent, port, path="/kang/snapshot"):
    resp = await agent.request_async(
        {"host": "127.0.0.1", "port": port}, "GET", path)
    assert resp.status_code == 200, resp.status_code
    return resp


def test_kang_snapshot_lifecycle():
    async def body():
        monitor = PoolMonitor()
        kang = KangServer(monitor=monitor)
        await kang.start()

        agent = HttpAgent({
            "defaultPort": kang.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })

        # empty snapshot
        resp = await scrape(agent, kang.port)
        obj = json.loads(resp.body)
        assert obj["service"]["name"] == "cueball"
        assert obj["types"] == ["pool", "set", "dns_res"]
        assert obj["pool"] == {}
        assert obj["set"] == {}
        assert obj["dns_res"] == {}

        # bring up a pool (fake resolver + scripted conns)
        import cueball_amd.pool as mod_pool
        orig_monitor = mod_pool.global_monitor
        mod_pool.global_monitor = monitor
        try:
            connections = []
            resolver = DummyResolver()
            rfsm = ResolverFSM(resolver, {})

            def ctor(backend):
                c = DummyConnection(backend)
                c.backend = backend.get("key")
                connections.append(c)
                return c

            pool = ConnectionPool({
                "domain": "foobar",
                "spares": 2,
                "maximum": 2,
                "constructor": ctor,
                "recovery": RECOVERY,
                "resolver": rfsm,
            })
            rfsm.start()
            resolver.add("b1", {})
            await asyncio.sleep(0.05)
            assert len(connections) == 2

            resp = await scrape(agent, kang.port)
            obj = json.loads(resp.body)
            assert list(obj["pool"].keys()) == [pool.p_uuid]
            pinf = obj["pool"][pool.p_uuid]
            assert list(pinf["backends"].keys()) == ["b1"]
            assert pinf["connections"] == {"b1": {"connecting": 2}}
            assert pinf["state"] == "starting"
            assert pinf["options"]["spares"] == 2

            connections[0].connect()
            await asyncio.sleep(0.05)
            resp = await scrape(agent, kang.port)
            pinf = json.loads(resp.body)["pool"][pool.p_uuid]
            assert pinf["connections"] == {"b1": {"idle": 1,
                                                  "connecting": 1}}
            assert pinf["state"] == "running"

            pool.stop()
            await asyncio.sleep(0.5)
        finally:
            mod_pool.global_monitor = orig_monitor

        fut = asyncio.get_running_loop().create_future()
        agent.stop(lambda e: fut.set_result(None))
        await fut
        kang.stop()

    run(body())


def test_kang_snapshot_cset():
    async def body():
        monitor = PoolMonitor()
        kang = KangServer(monitor=monitor)
        await kang.start()
        agent = HttpAgent({
            "defaultPort": kang.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })

        import cueball_amd.connection_set as mod_cset
        orig_monitor = mod_cset.global_monitor
        mod_cset.global_monitor = monitor
        try:
            connections = []
            resolver = DummyResolver()

            def ctor(backend):
                c = DummyConnection(backend)
                c.backend = backend.get("key")
                connections.append(c)
                return c

            cset = ConnectionSet({
                "domain": "foobar",
                "constructor": ctor,
                "recovery": RECOVERY,
                "target": 2,
                "maximum": 4,
                "resolver": resolver,
            })
            cset.on("added", lambda k, c, h: None)
            cset.on("removed", lambda k, c, h: h.release())
            resolver.start()
            resolver.add("b1", {})
            await asyncio.sleep(0.05)
            for c in connections:
                c.connect()
            await asyncio.sleep(0.1)

            resp = await scrape(agent, kang.port)
            obj = json.loads(resp.body)
            assert list(obj["set"].keys()) == [cset.cs_uuid]
            sinf = obj["set"][cset.cs_uuid]
            assert list(sinf["backends"].keys()) == ["b1"]
            assert sinf["fsms"] == {"b1": {"busy": 1}}
            assert sinf["state"] == "running"
            assert sinf["target"] == 2

            cset.stop()
            await asyncio.sleep(0.5)
        finally:
            mod_cset.global_monitor = orig_monitor

        fut = asyncio.get_running_loop().create_future()
        agent.stop(lambda e: fut.set_result(None))
        await fut
        kang.stop()

    run(body())


def test_metrics_endpoint():
    async def body():
        collector = create_collector(labels={"component": "cueball"})
        c = collector.counter(name="cueball_events",
                              help="Total number of cueball error events")
        c.increment({"evt": "claim-timeout", "type": "error"})
        c.increment({"evt": "claim-timeout", "type": "error"})

        kang = KangServer(monitor=PoolMonitor(), collector=collector)
        await kang.start()
        agent = HttpAgent({
            "defaultPort": kang.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })
        resp = await agent.request_async(
            {"host": "127.0.0.1", "port": kang.port}, "GET", "/metrics")
        text = resp.body.decode()
        assert "# TYPE cueball_events counter" in text
        assert 'evt="claim-timeout"' in text
        assert "} 2" in text

        fut = asyncio.get_running_loop().create_future()
        agent.stop(lambda e: fut.set_result(None))
        await fut
        kang.stop()

    run(body())


def test_pool_live_gauges():
    """Pools publish live connection gauges through the shared metrics
    collector (piggybacked on the 5 Hz LPF tick; beyond the reference's
    kang-only state exposure)."""
    async def body():
        from cueball_amd.metrics import create_collector
        from cueball_amd.testing import MockHttpServer

        srv = MockHttpServer()
        await srv.start()
        collector = create_collector(labels={"component": "cueball"})
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 2,
            "maximum": 4,
            "collector": collector,
        })
        await agent.request_async("127.0.0.1", "GET", "/warm")
        await asyncio.sleep(0.5)  # a couple of LPF ticks
        text = collector.collect()
        assert "cueball_pool_connections" in text
        assert 'state="idle"' in text

        fut = asyncio.get_running_loop().create_future()
        agent.stop(lambda e: fut.set_result(None))
        await fut
        srv.stop()

    run(body())
