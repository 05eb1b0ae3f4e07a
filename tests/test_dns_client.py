"""DNS client + wire-format integration tests over real UDP sockets.

These exercise the mname-client-equivalent engine end-to-end: encoding,
the mock server's decoding, failover between resolvers, timeout errors
and MultiError aggregation, plus a full DNSResolver run against a real
local server.
"""

import asyncio

import pytest

from cueball_amd import dns_wire
from cueball_amd.dns_client import (DnsClient, DnsError, MultiError,
                                    NoNameError, TimeoutError_)
from cueball_amd.resolver import DNSResolver
from cueball_amd.testing import MockDnsServer


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_wire_roundtrip():
    q = dns_wire.encode_query(0x1234, "foo.example.com", "SRV")
    msg = dns_wire.decode_message(q)
    assert msg.id == 0x1234
    assert msg.question[0]["name"] == "foo.example.com"
    assert msg.question[0]["type"] == "SRV"

    resp = dns_wire.encode_response(
        0x1234, msg.question[0],
        answers=[{"type": "SRV", "name": "foo.example.com", "ttl": 30,
                  "priority": 0, "weight": 5, "port": 8080,
                  "target": "b1.example.com"}],
        additionals=[{"type": "A", "name": "b1.example.com", "ttl": 30,
                      "target": "10.0.0.1"}])
    m2 = dns_wire.decode_message(resp)
    assert m2.rcode_name == "NOERROR"
    assert m2.answers[0]["target"] == "b1.example.com"
    assert m2.answers[0]["port"] == 8080
    assert m2.additionals[0]["target"] == "10.0.0.1"


def test_wire_aaaa_and_soa():
    resp = dns_wire.encode_response(
        7, {"name": "x.y", "type": "AAAA"},
        answers=[{"type": "AAAA", "name": "x.y", "ttl": 60,
                  "target": "fe80::1"}],
        authority=[{"type": "SOA", "name": "y", "ttl": 300}])
    m = dns_wire.decode_message(resp)
    assert m.answers[0]["target"] == "fe80::1"
    assert m.authority[0]["type"] == "SOA"
    assert m.authority[0]["ttl"] == 300


def test_lookup_against_mock_server():
    async def body():
        srv = MockDnsServer()
        await srv.start()
        srv.add_a("b1.test", "127.0.0.2", ttl=12)
        client = DnsClient()
        msg = await client.lookup_async({
            "domain": "b1.test", "type": "A", "timeout": 2000,
            "resolvers": [srv.resolver_address]})
        assert msg.get_answers()[0]["target"] == "127.0.0.2"
        assert msg.get_answers()[0]["ttl"] == 12
        srv.stop()

    run(body())


def test_lookup_nxdomain():
    async def body():
        srv = MockDnsServer()
        await srv.start()
        client = DnsClient()
        with pytest.raises(DnsError) as ei:
            await client.lookup_async({
                "domain": "nope.test", "type": "A", "timeout": 2000,
                "resolvers": [srv.resolver_address]})
        assert ei.value.code == "NXDOMAIN"
        srv.stop()

    run(body())


def test_lookup_failover_to_second_resolver():
    async def body():
        dead = MockDnsServer()   # drops everything
        await dead.start()
        dead.drop_next = 10**6
        live = MockDnsServer()
        await live.start()
        live.add_a("b1.test", "127.0.0.3")
        client = DnsClient()
        msg = await client.lookup_async({
            "domain": "b1.test", "type": "A", "timeout": 300,
            "resolvers": [dead.resolver_address, live.resolver_address]})
        assert msg.get_answers()[0]["target"] == "127.0.0.3"
        dead.stop()
        live.stop()

    run(body())


def test_lookup_all_timeout_multierror():
    async def body():
        d1 = MockDnsServer()
        await d1.start()
        d1.drop_next = 10**6
        d2 = MockDnsServer()
        await d2.start()
        d2.drop_next = 10**6
        client = DnsClient()
        with pytest.raises(MultiError) as ei:
            await client.lookup_async({
                "domain": "b1.test", "type": "A", "timeout": 150,
                "resolvers": [d1.resolver_address, d2.resolver_address]})
        errs = ei.value.errors()
        assert len(errs) == 2
        assert all(isinstance(e, TimeoutError_) for e in errs)
        d1.stop()
        d2.stop()

    run(body())


def test_full_resolver_against_real_server():
    """End-to-end: DNSResolver over real UDP, SRV + A with additionals-
    free two-stage resolution."""
    async def body():
        srv = MockDnsServer()
        await srv.start()
        srv.add_srv("_http._tcp.svc.test", "b1.svc.test", 8081, ttl=60)
        srv.add_srv("_http._tcp.svc.test", "b2.svc.test", 8082, ttl=60)
        srv.add_a("b1.svc.test", "127.0.0.11", ttl=60)
        srv.add_a("b2.svc.test", "127.0.0.12", ttl=60)

        from cueball_amd.resolver import DNSResolverFSM
        DNSResolverFSM._nic_cache = {"lo": [
            {"family": "IPv4", "address": "127.0.0.1"}]}
        DNSResolverFSM._nic_cache_updated = \
            asyncio.get_running_loop().time() * 1000.0

        res = DNSResolver({
            "domain": "svc.test",
            "service": "_http._tcp",
            "resolvers": [srv.resolver_address],
            "recovery": {"default": {"timeout": 2000, "retries": 2,
                                     "delay": 50}},
        })
        added = {}
        res.on("added", lambda k, b: added.__setitem__(k, b))
        res.start()
        for _ in range(200):
            await asyncio.sleep(0.01)
            if len(added) == 2:
                break
        assert res.is_in_state("running")
        backs = sorted(added.values(), key=lambda b: b["port"])
        assert backs[0] == {"name": "b1.svc.test", "port": 8081,
                            "address": "127.0.0.11"}
        assert backs[1] == {"name": "b2.svc.test", "port": 8082,
                            "address": "127.0.0.12"}
        res.stop()
        await asyncio.sleep(0.05)
        srv.stop()

    run(body())


def test_tcp_fallback_on_truncation():
    """A TC=1 UDP response makes the client retry the same server over
    TCP (RFC 1035 §4.2.2)."""
    async def body():
        srv = MockDnsServer()
        await srv.start()
        srv.truncate_udp = True
        srv.add_a("big.test", "127.0.0.4")
        client = DnsClient()
        msg = await client.lookup_async({
            "domain": "big.test", "type": "A", "timeout": 3000,
            "resolvers": [srv.resolver_address]})
        assert msg.get_answers()[0]["target"] == "127.0.0.4"
        assert srv.queries, "UDP query must have been tried first"
        assert srv.tcp_queries == [("big.test", "A")]
        srv.stop()

    run(body())


def test_rcode_voting_on_multierror():
    """MultiError rcode voting: the resolver adopts the most common
    rcode across failing nameservers (lib/resolver.js:1230-1259)."""
    import asyncio

    from cueball_amd.resolver import DNSResolverFSM

    async def body():
        s1 = MockDnsServer()
        await s1.start()
        s1.rcode_override = "REFUSED"
        s2 = MockDnsServer()
        await s2.start()
        s2.rcode_override = "NXDOMAIN"
        s3 = MockDnsServer()
        await s3.start()
        s3.rcode_override = "NXDOMAIN"

        res_fsm = DNSResolverFSM({
            "domain": "vote.test",
            "resolvers": [s1.resolver_address, s2.resolver_address,
                          s3.resolver_address],
            "recovery": {"default": {"timeout": 2000, "retries": 1,
                                     "delay": 10}},
        })
        req = res_fsm.resolve("vote.test", "A", 2000)
        errs = []
        done = asyncio.get_running_loop().create_future()
        req.on("error", lambda e: (errs.append(e),
                                   done.done() or done.set_result(None)))
        req.on("answers", lambda *a: done.done() or done.set_result(None))
        req.send()
        await asyncio.wait_for(done, 10)
        assert len(errs) == 1
        # NXDOMAIN won the vote 2:1 => converted to NoNameError
        from cueball_amd.dns_client import NoNameError
        assert isinstance(errs[0], NoNameError)
        s1.stop(); s2.stop(); s3.stop()

    run(body())


def test_client_concurrency_cap():
    """The shared client caps in-flight lookups (mname-client's
    concurrency option, lib/resolver.js:385-392)."""
    import asyncio

    async def body():
        srv = MockDnsServer()
        await srv.start()
        for i in range(8):
            srv.add_a("c%d.test" % i, "127.0.0.%d" % (i + 1))
        client = DnsClient(concurrency=2)

        in_flight = {"now": 0, "max": 0}
        orig = client._query_one

        async def counted(*args, **kw):
            in_flight["now"] += 1
            in_flight["max"] = max(in_flight["max"], in_flight["now"])
            try:
                await asyncio.sleep(0.02)
                return await orig(*args, **kw)
            finally:
                in_flight["now"] -= 1

        client._query_one = counted
        results = await asyncio.gather(*[
            client.lookup_async({"domain": "c%d.test" % i, "type": "A",
                                 "timeout": 3000,
                                 "resolvers": [srv.resolver_address]})
            for i in range(8)])
        assert len(results) == 8
        assert in_flight["max"] <= 2
        srv.stop()

    run(body())
