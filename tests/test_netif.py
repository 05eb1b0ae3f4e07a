"""Interface enumeration (netif.py): the AAAA-stage gate's data source.

Mirrors the reference's os.networkInterfaces()-based NIC fixtures
(test/dns.test.js:34-52, INT_NO_V6 / INT_V6) plus the /proc parser and
a live getifaddrs sanity check.
"""

import socket

import pytest

from cueball_amd import netif

# node-style fixtures (test/dns.test.js:34-52)
INT_NO_V6 = {
    "lo0": [
        {"family": "IPv4", "address": "127.0.0.1", "internal": True},
        {"family": "IPv6", "address": "::1", "internal": True},
    ],
    "net0": [
        {"family": "IPv4", "address": "10.0.0.5", "internal": False},
    ],
}

INT_V6 = {
    "lo0": [
        {"family": "IPv4", "address": "127.0.0.1", "internal": True},
        {"family": "IPv6", "address": "::1", "internal": True},
    ],
    "net0": [
        {"family": "IPv4", "address": "10.0.0.5", "internal": False},
        {"family": "IPv6", "address": "2001:db8::5", "internal": False},
    ],
}

INT_LINK_LOCAL_ONLY = {
    "lo0": [
        {"family": "IPv6", "address": "::1", "internal": True},
    ],
    "net0": [
        {"family": "IPv4", "address": "10.0.0.5", "internal": False},
        {"family": "IPv6", "address": "fe80::1", "internal": False},
    ],
}


def test_have_non_loopback_v6_fixtures():
    # the reference counts ANY v6 address except ::1 — including
    # link-locals (lib/resolver.js:749-754)
    assert netif.have_non_loopback_v6(INT_NO_V6) is False
    assert netif.have_non_loopback_v6(INT_V6) is True
    assert netif.have_non_loopback_v6(INT_LINK_LOCAL_ONLY) is True
    assert netif.have_non_loopback_v6({}) is False


def test_have_global_v6_fixtures():
    assert netif.have_global_v6(INT_NO_V6) is False
    assert netif.have_global_v6(INT_V6) is True
    # link-local does NOT count for the strict variant
    assert netif.have_global_v6(INT_LINK_LOCAL_ONLY) is False


PROC_IF_INET6 = """\
00000000000000000000000000000001 01 80 10 80       lo
fe8000000000000002163efffe123456 02 40 20 80     eth0
20010db8000000000000000000000005 02 40 00 80     eth0
"""


def test_parse_proc_if_inet6():
    nics = netif.parse_proc_if_inet6(PROC_IF_INET6)
    assert set(nics) == {"lo", "eth0"}
    assert nics["lo"] == [{"family": "IPv6", "address": "::1",
                           "internal": True}]
    addrs = [a["address"] for a in nics["eth0"]]
    assert "fe80::216:3eff:fe12:3456" in addrs
    assert "2001:db8::5" in addrs
    assert netif.have_non_loopback_v6(nics) is True


def test_parse_proc_if_inet6_garbage():
    assert netif.parse_proc_if_inet6("") == {}
    assert netif.parse_proc_if_inet6("not an if_inet6 line\n") == {}
    assert netif.parse_proc_if_inet6("zzzz 01 80 10 80 lo\n") == {}


def test_getifaddrs_live():
    """On any Linux host the loopback interface must be visible with
    127.0.0.1 (the round-1 heuristic failed exactly this on hosts
    whose hostname does not resolve to an interface address)."""
    nics = netif._getifaddrs()
    all_addrs = [(a["family"], a["address"])
                 for addrs in nics.values() for a in addrs]
    assert ("IPv4", "127.0.0.1") in all_addrs
    # loopback flagged internal
    lo = [a for addrs in nics.values() for a in addrs
          if a["address"] == "127.0.0.1"]
    assert lo[0]["internal"] is True


def test_network_interfaces_live_matches_getifaddrs():
    nics = netif.network_interfaces()
    assert any(a["address"] == "127.0.0.1"
               for addrs in nics.values() for a in addrs)


def test_proc_v6_live():
    """If the host exposes /proc/net/if_inet6, the parser must accept
    the real file."""
    try:
        with open("/proc/net/if_inet6") as f:
            text = f.read()
    except OSError:
        pytest.skip("no /proc/net/if_inet6 on this host")
    nics = netif.parse_proc_if_inet6(text)
    for addrs in nics.values():
        for a in addrs:
            assert a["family"] == "IPv6"
            socket.inet_pton(socket.AF_INET6, a["address"])
