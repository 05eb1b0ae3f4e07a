"""Property-based fuzzing of the DNS resolver's staged-TTL pipeline.

Hypothesis mutates a scripted zone (SRV target churn, address changes,
transient server failures) under virtual time and checks the pipeline's
eventual consistency: once the zone stabilizes and TTLs expire, the
resolver's backend set equals the zone-derived truth, added/removed
events reconcile, and the FSM never wedges (survey hard-part #4: "DNS
TTL bookkeeping across partial expiries").
"""

from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from cueball_amd.dns_client import NoNameError, TimeoutError_
from cueball_amd.dns_wire import DnsMessage
from cueball_amd.resolver import DNSResolver, DNSResolverFSM, srv_key
from cueball_amd.testing import VirtualLoop, advance

RECOVERY = {"default": {"timeout": 1000, "retries": 3, "delay": 100,
                        "maxDelay": 1000}}

INT_NO_V6 = {"foo0": [{"address": "1.2.3.4", "family": "IPv4"}]}

TARGET_POOL = ["n1.z.test", "n2.z.test", "n3.z.test", "n4.z.test"]
ADDR_POOL = ["10.0.0.1", "10.0.0.2", "10.0.0.3", "10.0.0.4"]


class MutableZoneClient:
    """Scripted DNS client over a mutable zone with failure injection."""

    def __init__(self):
        # name -> (port) for SRV targets currently advertised
        self.srv_targets = {}           # target -> port
        self.addrs = {}                 # target -> v4 address
        self.srv_ttl = 2
        self.a_ttl = 2
        self.fail_next = 0              # timeout the next N lookups
        self.history = []

    def lookup(self, opts, cb, loop=None):
        self.history.append((opts["domain"], opts["type"]))
        if self.fail_next > 0:
            self.fail_next -= 1
            loop.call_later(opts["timeout"] / 1000.0,
                            lambda: cb(TimeoutError_(opts["domain"]), None))
            return
        domain, rtype = opts["domain"], opts["type"]
        msg = DnsMessage()
        if rtype == "SRV":
            if domain == "_z._tcp.zone.test" and self.srv_targets:
                for tgt, port in sorted(self.srv_targets.items()):
                    msg.answers.append({
                        "type": "SRV", "name": domain, "ttl": self.srv_ttl,
                        "priority": 0, "weight": 1, "port": port,
                        "target": tgt})
                loop.call_soon(lambda: cb(None, msg))
            else:
                loop.call_soon(lambda: cb(NoNameError(domain), None))
        elif rtype == "A":
            addr = self.addrs.get(domain)
            if addr is not None:
                msg.answers.append({"type": "A", "name": domain,
                                    "ttl": self.a_ttl, "target": addr})
                loop.call_soon(lambda: cb(None, msg))
            else:
                loop.call_soon(lambda: cb(NoNameError(domain), None))
        else:
            loop.call_soon(lambda: cb(NoNameError(domain), None))

    def expected_backends(self):
        out = {}
        for tgt, port in self.srv_targets.items():
            addr = self.addrs.get(tgt)
            if addr is not None:
                b = {"name": tgt, "port": port, "address": addr}
                out[srv_key(b)] = b
        return out


ACTIONS = st.lists(
    st.tuples(
        st.sampled_from(["add_target", "remove_target", "change_addr",
                         "fail_lookups", "advance_small",
                         "advance_large"]),
        st.integers(min_value=0, max_value=3),
    ),
    min_size=3, max_size=25,
)


@settings(max_examples=120, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(actions=ACTIONS)
def test_resolver_converges_to_zone(actions):
    loop = VirtualLoop()
    try:
        loop.run_until_complete(_scenario(loop, actions))
    finally:
        loop.close()


async def _scenario(loop, actions):
    DNSResolverFSM._nic_cache = INT_NO_V6
    DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0

    zone = MutableZoneClient()
    # start with one live target so the resolver can come up
    zone.srv_targets[TARGET_POOL[0]] = 7000
    zone.addrs[TARGET_POOL[0]] = ADDR_POOL[0]

    res = DNSResolver({
        "domain": "zone.test",
        "service": "_z._tcp",
        "resolvers": ["9.9.9.9"],
        "recovery": RECOVERY,
        "_nsclient": zone,
        "loop": loop,
    })
    live = {}
    res.on("added", lambda k, b: live.__setitem__(k, b))
    res.on("removed", lambda k: live.pop(k))
    res.start()
    await advance(loop, 5.0)

    for kind, seed in actions:
        if kind == "add_target":
            tgt = TARGET_POOL[seed]
            zone.srv_targets[tgt] = 7000 + seed
            zone.addrs.setdefault(tgt, ADDR_POOL[seed])
        elif kind == "remove_target":
            if len(zone.srv_targets) > 1:
                tgt = sorted(zone.srv_targets)[seed % len(zone.srv_targets)]
                del zone.srv_targets[tgt]
        elif kind == "change_addr":
            if zone.addrs:
                tgt = sorted(zone.addrs)[seed % len(zone.addrs)]
                zone.addrs[tgt] = ADDR_POOL[(seed + 1) % len(ADDR_POOL)]
        elif kind == "fail_lookups":
            zone.fail_next += 1 + seed
        elif kind == "advance_small":
            await advance(loop, 0.5)
        elif kind == "advance_large":
            await advance(loop, 4.0)

        # the resolver's backend set only ever contains zone-derived
        # keys from *some* point in time (never invents backends)
        # -- structural sanity: key == srv_key(backend)
        for k, b in res.list().items():
            assert k == srv_key(b)

    # let the dust settle: failures drain, every TTL expires, re-query
    zone.fail_next = 0
    await advance(loop, 60.0)

    expected = zone.expected_backends()
    assert res.list() == expected, (res.list(), expected)
    assert live == expected
    if expected:
        assert res.is_in_state("running")

    res.stop()
    await advance(loop, 2.0)
    assert res.is_in_state("stopped")
