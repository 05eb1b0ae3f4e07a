"""resolver_for_ip_or_domain() factory + StaticIpResolver behaviors
(reference test/resolver_for.test.js, test/resolver_static.test.js)."""

import asyncio

import pytest

from cueball_amd.resolver import (DNSResolver, ResolverFSM, StaticIpResolver,
                                  config_for_ip_or_domain, parse_ip_or_domain,
                                  resolver_for_ip_or_domain)


def run(coro):
    loop = asyncio.new_event_loop()
    asyncio.set_event_loop(loop)
    try:
        return loop.run_until_complete(coro)
    finally:
        asyncio.set_event_loop(None)
        loop.close()


def test_resolver_for_bad_arguments():
    """Structurally invalid args raise (reference :16-39); invalid
    *input strings* return Error instances instead."""
    with pytest.raises(Exception):
        resolver_for_ip_or_domain({})
    with pytest.raises(Exception):
        resolver_for_ip_or_domain({"input": 1234})


def test_parse_ip_or_domain_ipv4():
    result = parse_ip_or_domain("127.0.0.1")
    assert not isinstance(result, Exception)
    assert result["kind"] == "static"
    assert result["cons"] is StaticIpResolver
    assert result["config"] == {
        "backends": [{"address": "127.0.0.1", "port": None}]}

    result = parse_ip_or_domain("127.0.0.1:1234")
    assert not isinstance(result, Exception)
    assert result["config"] == {
        "backends": [{"address": "127.0.0.1", "port": 1234}]}

    result = parse_ip_or_domain("127.0.0.1:-3")
    assert isinstance(result, Exception)
    assert "unsupported port in input:" in str(result)

    result = parse_ip_or_domain("127.0.0.1:ab123")
    assert isinstance(result, Exception)
    assert "unsupported port in input:" in str(result)


def test_parse_ip_or_domain_hostname():
    result = parse_ip_or_domain("1.moray.emy-10.joyent.us")
    assert not isinstance(result, Exception)
    assert result["kind"] == "dns"
    assert result["cons"] is DNSResolver
    assert result["config"] == {"domain": "1.moray.emy-10.joyent.us"}

    result = parse_ip_or_domain("1.moray.emy-10.joyent.us:2020")
    assert not isinstance(result, Exception)
    assert result["config"] == {"domain": "1.moray.emy-10.joyent.us",
                                "defaultPort": 2020}


def test_config_for_ip_or_domain_static():
    result = config_for_ip_or_domain({"input": "127.0.0.1:2020"})
    assert not isinstance(result, Exception)
    assert result["kind"] == "static"
    assert result["config"] == {
        "backends": [{"address": "127.0.0.1", "port": 2020}]}
    assert result["mergedConfig"] == result["config"]

    # input port overrides resolverConfig defaultPort; other fields merge
    result = config_for_ip_or_domain({
        "input": "1.moray:4567",
        "resolverConfig": {"defaultPort": 1234, "service": "_moray_.tcp"},
    })
    assert not isinstance(result, Exception)
    assert result["kind"] == "dns"
    assert result["mergedConfig"] == {
        "defaultPort": 4567, "service": "_moray_.tcp", "domain": "1.moray"}


def test_resolver_for_static_ip():
    async def body():
        result = resolver_for_ip_or_domain({"input": "127.0.0.1:2020"})
        assert not isinstance(result, Exception)
        assert isinstance(result, ResolverFSM)
        lst = result.list()
        assert len(lst) == 1
        (backend,) = lst.values()
        assert backend == {"name": "127.0.0.1:2020",
                           "address": "127.0.0.1", "port": 2020}

        bad = resolver_for_ip_or_domain({"input": "127.0.0.1:70000"})
        assert isinstance(bad, Exception)
        assert "unsupported port in input:" in str(bad)

    run(body())


def test_resolver_for_hostname():
    async def body():
        result = resolver_for_ip_or_domain({
            "input": "1.moray.emy-10.joyent.us",
            "resolverConfig": {
                "recovery": {"default": {"retries": 1, "timeout": 1000,
                                         "delay": 1000, "maxDelay": 1000}},
            },
        })
        assert not isinstance(result, Exception)
        assert isinstance(result, ResolverFSM)
        assert result.is_in_state("stopped")

    run(body())


def test_static_resolver_bad_arguments():
    with pytest.raises(Exception):
        StaticIpResolver({})
    with pytest.raises(Exception):
        StaticIpResolver({"backends": "nope"})


def test_static_resolver_no_backends():
    async def body():
        resolver = StaticIpResolver({"backends": []})
        added = []
        resolver.on("added", lambda *a: added.append(a))
        running = asyncio.get_running_loop().create_future()
        resolver.on("stateChanged", lambda st: st == "running"
                    and not running.done() and running.set_result(None))
        resolver.start()
        await asyncio.wait_for(running, 5)
        assert added == []
        assert resolver.list() == {}
        assert resolver.count() == 0
        resolver.stop()

    run(body())


def test_static_resolver_default_port():
    async def body():
        resolver = StaticIpResolver({
            "defaultPort": 2021,
            "backends": [
                {"address": "10.0.0.3", "port": 2022},
                {"address": "10.0.0.4"},
                {"address": "10.0.0.5"},
            ],
        })
        found = []
        resolver.on("added", lambda key, b: found.append(b))
        running = asyncio.get_running_loop().create_future()
        resolver.on("stateChanged", lambda st: st == "running"
                    and not running.done() and running.set_result(None))
        resolver.start()
        await asyncio.wait_for(running, 5)
        assert resolver.count() == 3
        assert found == [
            {"name": "10.0.0.3:2022", "address": "10.0.0.3", "port": 2022},
            {"name": "10.0.0.4:2021", "address": "10.0.0.4", "port": 2021},
            {"name": "10.0.0.5:2021", "address": "10.0.0.5", "port": 2021},
        ]
        resolver.stop()

    run(body())


def test_static_resolver_several_backends():
    async def body():
        resolver = StaticIpResolver({"backends": [
            {"address": "10.0.0.1", "port": 1111},
            {"address": "10.0.0.2", "port": 2222},
        ]})
        found = {}
        resolver.on("added", lambda key, b: found.__setitem__(key, b))
        running = asyncio.get_running_loop().create_future()
        resolver.on("stateChanged", lambda st: st == "running"
                    and not running.done() and running.set_result(None))
        resolver.start()
        await asyncio.wait_for(running, 5)
        assert resolver.count() == 2
        assert found == resolver.list()
        addrs = sorted(b["address"] for b in found.values())
        assert addrs == ["10.0.0.1", "10.0.0.2"]
        resolver.stop()

    run(body())
