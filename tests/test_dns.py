"""DNS resolver pipeline tests (port of reference test/dns.test.js).

A scripted FakeDnsClient picks its answer from the queried name's
suffix (`.ok`, `.notfound`, `.notimp`, `.short-ttl`, `.timeout`),
mirroring the reference's DummyDnsClient; NIC info is stubbed through
the class-level cache to exercise the v6 short-cut.
"""

import pytest

from cueball_amd.dns_client import (DnsError, NoNameError, NoRecordsError,
                                    TimeoutError_)
from cueball_amd.dns_wire import DnsMessage
from cueball_amd.resolver import DNSResolver, DNSResolverFSM
from cueball_amd.testing import advance, settle
from conftest import run_vt

RECOVERY = {"default": {"timeout": 1000, "retries": 3, "delay": 100}}

INT_NO_V6 = {
    "lo0": [{"address": "::1", "family": "IPv6"}],
    "foo0": [{"address": "1.2.3.4", "family": "IPv4"}],
}
INT_V6 = {
    "lo0": [{"address": "::1", "family": "IPv6"}],
    "foo0": [{"address": "1.2.3.4", "family": "IPv4"},
             {"address": "fe80::1:2:3:4", "family": "IPv6"}],
}


class FakeDnsClient:
    """Scripted DNS client: the domain suffix selects the response."""

    def __init__(self):
        self.history = []
        self.use_a2 = False
        self.srv_ttl = 3600

    def lookup(self, opts, cb, loop=None):
        self.history.append(dict(opts))
        domain = opts["domain"]
        rtype = opts["type"]
        parts = domain.split(".")[::-1] + ["", ""]
        msg = DnsMessage()

        def srv_rr(target, port=111, ttl=None):
            return {"type": "SRV", "name": domain,
                    "ttl": self.srv_ttl if ttl is None else ttl,
                    "priority": 0, "weight": 10, "port": port,
                    "target": target}

        def deliver(err, m):
            loop.call_soon(lambda: cb(err, m))

        if parts[0] == "ok":
            if parts[1] == "srv" and parts[2] == "_tcp" and rtype == "SRV":
                msg.answers.append(srv_rr("a.ok"))
                msg.answers.append(srv_rr("aaaa.ok"))
                if self.use_a2:
                    msg.answers.append(srv_rr("a2.ok"))
            elif parts[1] == "dupe" and parts[2] == "_tcp" and \
                    rtype == "SRV":
                msg.answers.append(srv_rr("dupe.ok", port=112))
                if self.use_a2:
                    msg.answers.append(srv_rr("dupe.ok", port=112))
            elif parts[1] == "a" and rtype == "A":
                msg.answers.append({"type": "A", "name": domain,
                                    "ttl": 3600, "target": "1.2.3.4"})
            elif parts[1] == "a2" and rtype == "A":
                msg.answers.append({"type": "A", "name": domain,
                                    "ttl": 3600, "target": "1.2.3.5"})
            elif parts[1] == "a2" and rtype == "AAAA":
                msg.answers.append({"type": "AAAA", "name": domain,
                                    "ttl": 1, "target": "1234:abcd::2"})
            elif parts[1] == "aaaa" and rtype == "AAAA":
                msg.answers.append({"type": "AAAA", "name": domain,
                                    "ttl": 3600, "target": "1234:abcd::1"})
            elif parts[1] == "dupe" and rtype == "A":
                for _ in range(3):
                    msg.answers.append({"type": "A", "name": domain,
                                        "ttl": 3600, "target": "1.2.3.1"})
            elif parts[1] in ("a", "aaaa", "a2", "dupe"):
                pass  # NODATA
            else:
                deliver(NoNameError(domain), None)
                return
            deliver(None, msg)
        elif parts[0] == "notfound":
            deliver(DnsError("NXDOMAIN for %s" % domain, code="NXDOMAIN"),
                    None)
        elif parts[0] == "notimp":
            if parts[1] == "srv" and parts[2] == "_tcp" and rtype == "SRV":
                msg.answers.append(srv_rr("a.notimp"))
                deliver(None, msg)
            else:
                deliver(DnsError("NOTIMP for %s" % domain, code="NOTIMP"),
                        None)
        elif parts[0] == "short-ttl":
            if parts[1] == "a" and rtype == "A":
                msg.answers.append({"type": "A", "name": domain,
                                    "ttl": 1, "target": "1.2.3.4"})
                deliver(None, msg)
            else:
                deliver(None, msg)  # NODATA
        elif parts[0] == "timeout":
            loop.call_later(opts["timeout"] / 1000.0,
                            lambda: cb(TimeoutError_(domain), None))
        else:
            raise AssertionError("unexpected domain %r" % domain)


def make_resolver(loop, domain, nsc=None, interfaces=INT_V6, **opts):
    DNSResolverFSM._nic_cache = interfaces
    DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0
    nsc = nsc or FakeDnsClient()
    ropts = {
        "domain": domain,
        "resolvers": ["1.1.1.1"],
        "recovery": RECOVERY,
        "_nsclient": nsc,
        "loop": loop,
    }
    ropts.update(opts)
    return DNSResolver(ropts), nsc


def collect(res):
    state = {"added": {}, "removed": []}
    res.on("added", lambda k, b: state["added"].__setitem__(k, b))
    res.on("removed", lambda k: state["removed"].append(k))
    return state


def test_srv_lookup():
    async def body(loop):
        res, nsc = make_resolver(loop, "srv.ok", service="_svc._tcp",
                                 defaultPort=80)
        st = collect(res)
        res.start()
        await advance(loop, 1.0)
        assert res.is_in_state("running")
        backends = sorted(st["added"].values(),
                          key=lambda b: b["name"])
        assert len(backends) == 2
        assert backends[0]["name"] == "a.ok"
        assert backends[0]["address"] == "1.2.3.4"
        assert backends[0]["port"] == 111
        assert backends[1]["name"] == "aaaa.ok"
        assert backends[1]["address"] == "1234:abcd::1"
        assert backends[1]["port"] == 111
        # SRV query went out for _svc._tcp.srv.ok
        srv_qs = [h for h in nsc.history if h["type"] == "SRV"]
        assert srv_qs[0]["domain"] == "_svc._tcp.srv.ok"
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_plain_a_lookup():
    async def body(loop):
        res, nsc = make_resolver(loop, "a.ok", service="_svc._tcp",
                                 defaultPort=123)
        st = collect(res)
        res.start()
        await advance(loop, 1.0)
        assert res.is_in_state("running")
        backends = list(st["added"].values())
        assert len(backends) == 1
        assert backends[0]["address"] == "1.2.3.4"
        assert backends[0]["port"] == 123  # defaultPort: no SRV records
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_not_found_fails():
    async def body(loop):
        res, nsc = make_resolver(loop, "x.notfound", service="_svc._tcp")
        st = collect(res)
        res.start()
        await advance(loop, 60.0)
        assert res.is_in_state("failed")
        assert res.get_last_error() is not None
        assert st["added"] == {}
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_notimp_fails():
    async def body(loop):
        res, nsc = make_resolver(loop, "a.notimp", service="_svc._tcp")
        res.start()
        await advance(loop, 60.0)
        assert res.is_in_state("failed")
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_srv_ok_notimp_on_a_fails():
    async def body(loop):
        res, nsc = make_resolver(loop, "srv.notimp", service="_svc._tcp")
        res.start()
        await advance(loop, 60.0)
        # SRV gave a.notimp, but its A/AAAA lookups NOTIMP out
        assert res.is_in_state("failed")
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_short_ttl_requeries():
    async def body(loop):
        res, nsc = make_resolver(loop, "a.short-ttl", service="_svc._tcp",
                                 defaultPort=80)
        st = collect(res)
        res.start()
        await advance(loop, 0.5)
        assert res.is_in_state("running")
        n0 = len([h for h in nsc.history if h["type"] == "A"])
        await advance(loop, 5.0)
        n1 = len([h for h in nsc.history if h["type"] == "A"])
        # ttl=1s: the A stage must have re-queried several times
        assert n1 - n0 >= 3
        # and the backend was never removed (same records each time)
        assert st["removed"] == []
        assert len(st["added"]) == 1
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_only_one_record_expires():
    async def body(loop):
        nsc = FakeDnsClient()
        nsc.use_a2 = True
        res, _ = make_resolver(loop, "srv.ok", nsc=nsc, service="_svc._tcp")
        st = collect(res)
        res.start()
        await advance(loop, 0.5)
        assert res.is_in_state("running")
        # a.ok(v4), aaaa.ok(v6), a2.ok(v6 ttl=1) + a2.ok(v4)
        assert len(st["added"]) == 4

        hist_before = len(nsc.history)
        await advance(loop, 3.0)
        new = nsc.history[hist_before:]
        # the expiring AAAA for a2.ok re-queries; SRV and the cached A
        # records don't.  (aaaa.ok's A is NODATA and carries no expiry,
        # so the fall-through A stage re-asks it — reference behavior.)
        assert any(h["domain"] == "a2.ok" and h["type"] == "AAAA"
                   for h in new)
        assert not any(h["type"] == "SRV" for h in new)
        assert not any(h["domain"] == "a.ok" for h in new)
        assert not any(h["domain"] == "a2.ok" and h["type"] == "A"
                       for h in new)
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_only_services_expire():
    async def body(loop):
        nsc = FakeDnsClient()
        nsc.srv_ttl = 1
        res, _ = make_resolver(loop, "srv.ok", nsc=nsc, service="_svc._tcp")
        st = collect(res)
        res.start()
        await advance(loop, 0.5)
        assert res.is_in_state("running")
        hist_before = len(nsc.history)
        await advance(loop, 3.0)
        new = nsc.history[hist_before:]
        # SRV re-queried; cached A/AAAA answers carried over across the
        # SRV refresh (srv_try's oldLookup carryover)
        assert any(h["type"] == "SRV" for h in new)
        assert not any(h["domain"] == "a.ok" and h["type"] == "A"
                       for h in new)
        assert not any(h["domain"] == "aaaa.ok" and h["type"] == "AAAA"
                       for h in new)
        assert st["removed"] == []
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_v6_shortcut_skips_aaaa():
    async def body(loop):
        res, nsc = make_resolver(loop, "srv.ok", service="_svc._tcp",
                                 interfaces=INT_NO_V6)
        st = collect(res)
        res.start()
        await advance(loop, 1.0)
        assert res.is_in_state("running")
        # no global v6 on any NIC: no AAAA queries at all
        assert not any(h["type"] == "AAAA" for h in nsc.history)
        # only the A-record backend is found
        assert len(st["added"]) == 1
        assert list(st["added"].values())[0]["name"] == "a.ok"
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_duped_records_deduplicated():
    async def body(loop):
        res, nsc = make_resolver(loop, "dupe.ok", service="_svc._tcp")
        st = collect(res)
        res.start()
        await advance(loop, 1.0)
        assert res.is_in_state("running")
        # 2 identical SRVs x 3 identical A answers -> exactly 1 backend
        assert len(st["added"]) == 1
        b = list(st["added"].values())[0]
        assert b["address"] == "1.2.3.1"
        assert b["port"] == 112
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_timeout_retries_then_fails():
    async def body(loop):
        recovery = {"default": {"timeout": 100, "retries": 2, "delay": 10}}
        res, nsc = make_resolver(loop, "x.timeout", service="_svc._tcp",
                                 recovery=recovery)
        res.start()
        await advance(loop, 120.0)
        assert res.is_in_state("failed")
        # several SRV attempts (initial + retries, then A-stage fallbacks)
        assert len(nsc.history) >= 3
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_resolver_stop_restart():
    """stop() then start() runs the pipeline again (stop drains to init
    via the sleep state; lib/resolver.js:1110-1118, :452-463)."""
    async def body(loop):
        res, nsc = make_resolver(loop, "a.ok", service="_svc._tcp",
                                 defaultPort=80)
        st = collect(res)
        res.start()
        await advance(loop, 1.0)
        assert res.is_in_state("running")
        assert len(st["added"]) == 1
        n_queries = len(nsc.history)

        res.stop()
        await advance(loop, 1.0)
        assert res.is_in_state("stopped")

        res.start()
        await advance(loop, 1.0)
        assert res.is_in_state("running")
        # pipeline re-ran (fresh SRV query at least)
        assert len(nsc.history) > n_queries
        res.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_static_resolver_stop_restart():
    from cueball_amd.resolver import StaticIpResolver

    async def body(loop):
        res = StaticIpResolver({
            "backends": [{"address": "10.0.0.1", "port": 80}],
            "loop": loop,
        })
        added = []
        res.on("added", lambda k, b: added.append(k))
        res.start()
        await settle(loop)
        assert res.is_in_state("running")
        assert len(added) == 1
        res.stop()
        await settle(loop)
        assert res.is_in_state("stopped")
        res.start()
        await settle(loop)
        assert res.is_in_state("running")
        assert len(added) == 2  # re-emitted on restart

    run_vt(lambda loop: body(loop))
