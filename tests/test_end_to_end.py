"""Flagship end-to-end integration: everything real.

A DNS-SRV-resolved connection pool over real UDP DNS and real TCP
backends, on a real clock: SRV records with short TTLs drive backend
discovery, the pool spreads connections, claims echo traffic, a backend
disappears from DNS and its connections drain, a new backend appears
and gets picked up — the full production shape of the framework in one
test.
"""

import asyncio

import pytest

from cueball_amd.connection import tcp_constructor
from cueball_amd.pool import ConnectionPool
from cueball_amd.resolver import DNSResolver, DNSResolverFSM
from cueball_amd.testing import MockDnsServer


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


async def start_echo():
    async def echo(reader, writer):
        try:
            while data := await reader.read(4096):
                writer.write(data)
                await writer.drain()
        except (ConnectionResetError, BrokenPipeError):
            pass
        finally:
            writer.close()

    srv = await asyncio.start_server(echo, "127.0.0.1", 0)
    return srv, srv.sockets[0].getsockname()[1]


async def claim_echo(pool, payload=b"ping"):
    hdl, conn = await asyncio.wait_for(pool.claim_async(), timeout=10)
    loop = asyncio.get_running_loop()
    fut = loop.create_future()
    listener = conn.on("data", lambda d: fut.done() or fut.set_result(d))
    try:
        conn.write(payload)
        data = await asyncio.wait_for(fut, timeout=5)
        assert data == payload
        return conn.backend["port"]
    finally:
        conn.remove_listener("data", listener)
        hdl.release()


@pytest.mark.timeout(120)
def test_dns_srv_pool_end_to_end():
    async def body():
        loop = asyncio.get_running_loop()
        DNSResolverFSM._nic_cache = {"lo": [
            {"family": "IPv4", "address": "127.0.0.1"}]}
        DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0

        # two echo backends advertised over SRV with a 1s TTL
        s1, p1 = await start_echo()
        s2, p2 = await start_echo()
        dns = MockDnsServer()
        await dns.start()
        dns.add_srv("_echo._tcp.cluster.test", "n1.cluster.test", p1,
                    ttl=1)
        dns.add_srv("_echo._tcp.cluster.test", "n2.cluster.test", p2,
                    ttl=1)
        dns.add_a("n1.cluster.test", "127.0.0.1", ttl=1)
        dns.add_a("n2.cluster.test", "127.0.0.1", ttl=1)

        resolver = DNSResolver({
            "domain": "cluster.test",
            "service": "_echo._tcp",
            "resolvers": [dns.resolver_address],
            "recovery": {"default": {"timeout": 2000, "retries": 3,
                                     "delay": 100, "maxDelay": 1000}},
        })
        pool = ConnectionPool({
            "domain": "cluster.test",
            "constructor": tcp_constructor(loop=loop),
            "resolver": resolver,
            "recovery": {"default": {"timeout": 2000, "retries": 3,
                                     "delay": 100, "maxDelay": 1000}},
            "spares": 2,
            "maximum": 6,
        })
        resolver.start()

        # pool comes up and spreads over both backends
        t0 = loop.time()
        while pool.get_stats()["idleConnections"] < 2 and \
                loop.time() - t0 < 15:
            await asyncio.sleep(0.02)
        assert pool.is_in_state("running")

        ports = set()
        for _ in range(12):
            ports.add(await claim_echo(pool))
        assert ports == {p1, p2}

        # n2 vanishes from DNS; within a few TTLs the pool only uses n1
        del dns.zone[("_echo._tcp.cluster.test", "SRV")][1]
        t0 = loop.time()
        while loop.time() - t0 < 20:
            await asyncio.sleep(0.25)
            if resolver.count() == 1:
                break
        assert resolver.count() == 1

        t0 = loop.time()
        while loop.time() - t0 < 15:
            ports = {await claim_echo(pool) for _ in range(6)}
            if ports == {p1}:
                break
            await asyncio.sleep(0.25)
        assert ports == {p1}

        # a replacement backend appears in DNS and gets traffic
        s3, p3 = await start_echo()
        dns.add_srv("_echo._tcp.cluster.test", "n3.cluster.test", p3,
                    ttl=1)
        dns.add_a("n3.cluster.test", "127.0.0.1", ttl=1)
        t0 = loop.time()
        seen_p3 = False
        while loop.time() - t0 < 20 and not seen_p3:
            await asyncio.sleep(0.25)
            for _ in range(6):
                if await claim_echo(pool) == p3:
                    seen_p3 = True
                    break
        assert seen_p3, "new backend from DNS never served traffic"

        pool.stop()
        resolver.stop()
        t0 = loop.time()
        while not pool.is_in_state("stopped") and loop.time() - t0 < 15:
            await asyncio.sleep(0.05)
        assert pool.is_in_state("stopped")
        dns.stop()
        for s in (s1, s2, s3):
            s.close()
        await asyncio.sleep(0.1)

    run(body())


@pytest.mark.timeout(60)
def test_pool_owns_its_resolver():
    """No `resolver` option: the pool builds the DNS resolver from
    domain/resolvers/service, starts it, and stopping the pool stops
    the resolver too (p_started_resolver path, lib/pool.js:207-221,
    :433-448)."""
    async def body():
        loop = asyncio.get_running_loop()
        DNSResolverFSM._nic_cache = {"lo": [
            {"family": "IPv4", "address": "127.0.0.1"}]}
        DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0

        srv, port = await start_echo()
        dns = MockDnsServer()
        await dns.start()
        dns.add_srv("_own._tcp.own.test", "o1.own.test", port, ttl=30)
        dns.add_a("o1.own.test", "127.0.0.1", ttl=30)

        pool = ConnectionPool({
            "domain": "own.test",
            "service": "_own._tcp",
            "resolvers": [dns.resolver_address],
            "constructor": tcp_constructor(loop=loop),
            "recovery": {"default": {"timeout": 2000, "retries": 3,
                                     "delay": 100, "maxDelay": 1000}},
            "spares": 1,
            "maximum": 2,
        })
        # the pool started its own resolver
        assert pool.p_started_resolver or \
            pool.p_resolver.is_in_state("stopped")

        assert (await claim_echo(pool)) == port
        assert pool.p_started_resolver
        assert pool.p_resolver.is_in_state("running")

        pool.stop()
        t0 = loop.time()
        while not pool.is_in_state("stopped") and loop.time() - t0 < 20:
            await asyncio.sleep(0.05)
        assert pool.is_in_state("stopped")
        assert pool.p_resolver.is_in_state("stopped")
        dns.stop()
        srv.close()
        await asyncio.sleep(0.1)

    run(body())
