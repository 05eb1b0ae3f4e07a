"""Kang snapshot schema parity, field by field against the reference
serializer (lib/pool-monitor.js:60-216) and the expectations of the
reference's monitor test (test/monitor.test.js:120-260)."""

import time

from cueball_amd.connection_set import ConnectionSet
from cueball_amd.pool import ConnectionPool
from cueball_amd.pool_monitor import monitor
from cueball_amd.resolver import DNSResolver, ResolverFSM
from cueball_amd.testing import (DummyConnection, DummyResolver,
                                 MockDnsServer, settle)
from conftest import run_vt

RECOVERY = {"default": {"timeout": 500, "retries": 1, "delay": 0}}

POOL_KEYS = {"backends", "connections", "dead_backends",
             "last_rebalance", "resolvers", "state", "counters",
             "options"}
POOL_OPT_KEYS = {"domain", "service", "defaultPort", "spares", "maximum"}
SET_KEYS = {"backends", "fsms", "connections", "dead_backends",
            "last_rebalance", "resolvers", "state", "counters",
            "target", "maximum", "options"}
SET_OPT_KEYS = {"domain", "service", "defaultPort"}
DNS_KEYS = {"domain", "service", "resolvers", "defaultPort", "state",
            "next", "backends", "counters"}


def test_kang_options_shape():
    opts = monitor.to_kang_options()
    assert opts["uri_base"] == "/kang"
    assert opts["service_name"] == "cueball"
    assert opts["version"] == "1.0.0"
    assert isinstance(opts["ident"], str)
    assert opts["list_types"]() == ["pool", "set", "dns_res"]
    assert opts["stats"]() == {}


def test_pool_snapshot_schema():
    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []

        def constructor(backend):
            c = DummyConnection(backend)
            conns.append(c)
            return c

        pool = ConnectionPool({
            "domain": "foobar",
            "constructor": constructor,
            "recovery": RECOVERY,
            "spares": 2,
            "maximum": 2,
            "resolver": rfsm,
            "loop": loop,
        })
        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        assert len(conns) == 2

        pid = pool.p_uuid
        assert pid in monitor.list_objects("pool")
        pinf = monitor.get("pool", pid)
        # exact top-level keys (last_rebalance appears once a
        # rebalance has run, which settle() guarantees here)
        assert set(pinf.keys()) == POOL_KEYS
        assert set(pinf["options"].keys()) == POOL_OPT_KEYS
        # test/monitor.test.js:183-186 expectations
        assert list(pinf["backends"].keys()) == ["b1"]
        assert pinf["connections"] == {"b1": {"connecting": 2}}
        assert pinf["state"] == "starting"
        assert pinf["options"]["spares"] == 2
        assert pinf["options"]["maximum"] == 2
        assert pinf["dead_backends"] == []
        assert isinstance(pinf["counters"], dict)
        assert isinstance(pinf["last_rebalance"], int)
        assert abs(pinf["last_rebalance"] - time.time()) < 60

        conns[0].connect()
        await settle(loop)
        pinf = monitor.get("pool", pid)
        assert pinf["connections"] == {"b1": {"idle": 1,
                                              "connecting": 1}}
        assert pinf["state"] == "running"

        conns[1].connect()
        await settle(loop)
        pool.stop()
        await settle(loop)
        assert pid not in monitor.list_objects("pool")

    run_vt(lambda loop: body(loop))


def test_set_snapshot_schema():
    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []

        def constructor(backend):
            c = DummyConnection(backend)
            conns.append(c)
            return c

        cset = ConnectionSet({
            "constructor": constructor,
            "recovery": RECOVERY,
            "target": 1,
            "maximum": 2,
            "resolver": rfsm,
            "loop": loop,
        })
        held = {}
        cset.on("added", lambda ck, conn, hdl: held.setdefault(ck, hdl))
        cset.on("removed", lambda ck, conn, hdl:
                held.pop(ck).release() if ck in held else None)
        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        for c in conns:
            c.connect()
        await settle(loop)

        sid = cset.cs_uuid
        assert sid in monitor.list_objects("set")
        sinf = monitor.get("set", sid)
        assert set(sinf.keys()) == SET_KEYS
        assert set(sinf["options"].keys()) == SET_OPT_KEYS
        assert sinf["target"] == 1
        assert sinf["maximum"] == 2
        assert list(sinf["backends"].keys()) == ["b1"]
        # one fsm, in busy state (its logical connection holds it)
        assert list(sinf["fsms"].keys()) == ["b1"]
        assert sum(sinf["fsms"]["b1"].values()) == 1
        assert isinstance(sinf["connections"], list)
        assert sinf["dead_backends"] == []

        cset.stop()
        await settle(loop)
        assert sid not in monitor.list_objects("set")

    run_vt(lambda loop: body(loop))


def test_dns_resolver_snapshot_schema():
    # real sockets => real event loop (like test_dns_client.py)
    async def body():
        import asyncio
        loop = asyncio.get_running_loop()
        dns = MockDnsServer()
        await dns.start()
        dns.add_srv("_svc._tcp.kang.test", "b1.kang.test", 1234, ttl=30)
        dns.add_a("b1.kang.test", "127.0.0.1", ttl=30)

        res = DNSResolver({
            "domain": "kang.test",
            "service": "_svc._tcp",
            "resolvers": [dns.resolver_address],
            "recovery": RECOVERY,
            "loop": loop,
        })
        res.start()
        # wait for first resolution
        t0 = loop.time()
        while not res.is_in_state("running") and loop.time() - t0 < 10:
            await asyncio.sleep(0.05)
        assert res.is_in_state("running")

        ids = monitor.list_objects("dns_res")
        assert len(ids) >= 1
        rid = res.r_fsm.r_uuid
        dinf = monitor.get("dns_res", rid)
        assert set(dinf.keys()) == DNS_KEYS
        assert dinf["domain"] == "kang.test"
        assert dinf["service"] == "_svc._tcp"
        assert dinf["state"] in ("sleep", "process", "srv", "a",
                                 "a_try", "a_next", "aaaa", "aaaa_try",
                                 "aaaa_next", "srv_try")
        assert len(dinf["backends"]) == 1
        # next-expiry ISO times must be real wall-clock (close to now
        # + TTL), not monotonic-epoch artifacts from 1970
        assert "srv" in dinf["next"]
        t = time.strptime(dinf["next"]["srv"], "%Y-%m-%dT%H:%M:%SZ")
        import calendar
        exp = calendar.timegm(t)
        assert abs(exp - (time.time() + 30)) < 120

        import asyncio
        res.stop()
        dns.stop()
        await asyncio.sleep(0.05)

    import asyncio
    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(body())
    finally:
        loop.close()
