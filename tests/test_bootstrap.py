"""Dynamic Resolver (bootstrap) mode + stack-trace capture tests.

Bootstrap mode (lib/resolver.js:172-180, :476-491): when `resolvers`
names a DNS domain instead of IPs, the resolver bootstraps by looking
that name up as `_dns._udp` via the system resolvers, then keeps its
nameserver list in sync with DNS — shared, refcounted, one bootstrap
instance per name.
"""

import pytest

import cueball_amd
from cueball_amd.dns_client import NoNameError
from cueball_amd.dns_wire import DnsMessage
from cueball_amd.resolver import DNSResolver, DNSResolverFSM
from cueball_amd.testing import advance, settle
from conftest import run_vt

RECOVERY = {"default": {"timeout": 1000, "retries": 3, "delay": 100}}

INT_NO_V6 = {"foo0": [{"address": "1.2.3.4", "family": "IPv4"}]}


class ZoneDnsClient:
    """Fake DNS client serving a scripted zone of (name, type) -> list
    of record dicts."""

    def __init__(self, zone):
        self.zone = zone
        self.history = []

    def lookup(self, opts, cb, loop=None):
        self.history.append(dict(opts))
        key = (opts["domain"], opts["type"])
        msg = DnsMessage()
        recs = self.zone.get(key)
        if recs is None:
            loop.call_soon(lambda: cb(NoNameError(opts["domain"]), None))
            return
        msg.answers.extend(recs)
        loop.call_soon(lambda: cb(None, msg))


def _reset_bootstrap_state():
    DNSResolverFSM.bootstrap_resolvers.clear()
    DNSResolverFSM.global_ns_clients.clear()


def test_bootstrap_mode():
    async def body(loop):
        _reset_bootstrap_state()
        DNSResolverFSM._nic_cache = INT_NO_V6
        DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0

        zone = {
            # the bootstrap resolver finds the nameservers themselves
            ("_dns._udp.binder.test", "SRV"): [
                {"type": "SRV", "name": "_dns._udp.binder.test", "ttl": 60,
                 "priority": 0, "weight": 1, "port": 53,
                 "target": "ns1.binder.test"}],
            ("ns1.binder.test", "A"): [
                {"type": "A", "name": "ns1.binder.test", "ttl": 60,
                 "target": "10.0.0.53"}],
            # ...and the real service is then resolved through them
            ("_svc._tcp.app.test", "SRV"): [
                {"type": "SRV", "name": "_svc._tcp.app.test", "ttl": 60,
                 "priority": 0, "weight": 1, "port": 9000,
                 "target": "app1.test"}],
            ("app1.test", "A"): [
                {"type": "A", "name": "app1.test", "ttl": 60,
                 "target": "10.0.1.1"}],
        }
        nsc = ZoneDnsClient(zone)
        # both the bootstrap (concurrency 10) and the main resolver use
        # the shared client cache
        DNSResolverFSM.global_ns_clients[10] = nsc

        res = DNSResolver({
            "domain": "app.test",
            "service": "_svc._tcp",
            "resolvers": ["binder.test"],  # a DNS name => bootstrap mode
            "recovery": RECOVERY,
            "_nsclient": nsc,
            "loop": loop,
        })
        added = {}
        res.on("added", lambda k, b: added.__setitem__(k, b))
        res.start()
        await advance(loop, 2.0)

        assert res.is_in_state("running")
        assert len(added) == 1
        b = list(added.values())[0]
        assert b == {"name": "app1.test", "port": 9000,
                     "address": "10.0.1.1"}

        # the bootstrap resolver exists, is shared and refcounted
        boot = DNSResolverFSM.bootstrap_resolvers.get("binder.test")
        assert boot is not None
        assert boot.r_ref_count == 1
        assert boot.r_service == "_dns._udp"
        # the app query went to the bootstrap-discovered nameserver
        svc_queries = [h for h in nsc.history
                       if h["domain"] == "_svc._tcp.app.test"]
        assert svc_queries
        assert svc_queries[0]["resolvers"] == ["10.0.0.53"]

        # a second resolver for the same bootstrap name shares it
        res2 = DNSResolver({
            "domain": "app.test",
            "service": "_svc._tcp",
            "resolvers": ["binder.test"],
            "recovery": RECOVERY,
            "_nsclient": nsc,
            "loop": loop,
        })
        res2.on("added", lambda k, b: None)
        res2.start()
        await advance(loop, 2.0)
        assert boot.r_ref_count == 2

        # stopping both releases the bootstrap
        res.stop()
        res2.stop()
        await advance(loop, 1.0)
        assert boot.r_ref_count == 0

        _reset_bootstrap_state()

    run_vt(lambda loop: body(loop))


def test_stack_trace_capture_toggle():
    """enable_stack_traces() makes double-release errors name the real
    releasing function (lib/utils.js:52-58, lib/connection-fsm.js:598)."""
    from cueball_amd.pool import ConnectionPool
    from cueball_amd.resolver import ResolverFSM
    from cueball_amd.testing import DummyConnection, DummyResolver

    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []

        def ctor(b):
            c = DummyConnection(b)
            conns.append(c)
            return c

        pool = ConnectionPool({
            "domain": "st.test",
            "constructor": ctor,
            "recovery": {"default": {"timeout": 500, "retries": 1,
                                     "delay": 0}},
            "spares": 1,
            "maximum": 1,
            "resolver": rfsm,
            "loop": loop,
        })
        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        conns[0].connect()
        await settle(loop)

        got = {}

        def cb(err, hdl=None, conn=None):
            got["hdl"] = hdl

        pool.claim({}, cb)
        await settle(loop)

        cueball_amd.enable_stack_traces()
        try:
            assert cueball_amd.stack_traces_enabled()

            def my_release_site():
                got["hdl"].release()

            my_release_site()
            with pytest.raises(Exception) as ei:
                got["hdl"].release()
            # with traces enabled the error names the real call site
            assert "my_release_site" in str(ei.value) or \
                "test_bootstrap" in str(ei.value)
        finally:
            cueball_amd.disable_stack_traces()
        assert not cueball_amd.stack_traces_enabled()
        pool.stop()
        await advance(loop, 1.0)

    run_vt(lambda loop: body(loop))
