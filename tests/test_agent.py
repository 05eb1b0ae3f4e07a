"""HttpAgent / HttpsAgent tests (port of reference test/agent.test.js).

Real localhost servers (MockHttpServer), fixed-IP pools, keep-alive
reuse, initialDomains, create_pool args, the pinger, connection-refused
and broken-server error paths, and HTTPS with a self-signed cert.
"""

import asyncio
import json
import os
import ssl

import pytest

from cueball_amd.agent import HttpAgent, HttpsAgent
from cueball_amd.testing import MockHttpServer

RECOVERY = {"default": {"timeout": 2000, "retries": 2, "delay": 50,
                        "maxDelay": 200}}

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


async def stop_agent(agent):
    fut = asyncio.get_running_loop().create_future()
    agent.stop(lambda err: fut.set_result(err))
    await fut


def test_basic_agent_usage_fixed_ip():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 1,  # exactly one conn => reuse is observable
        })
        resp = await agent.request_async("127.0.0.1", "GET", "/hello")
        assert resp.status_code == 200
        data = json.loads(resp.body)
        assert data["path"] == "/hello"

        # second request reuses the kept-alive connection
        resp2 = await agent.request_async("127.0.0.1", "GET", "/again")
        assert resp2.status_code == 200
        assert json.loads(resp2.body)["count"] == 2
        assert srv.conn_count == 1

        await stop_agent(agent)
        srv.stop()

    run(body())


def test_agent_initial_domains():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
            "initialDomains": ["127.0.0.1"],
        })
        # the pool exists before any request
        assert agent.get_pool("127.0.0.1") is not None
        await asyncio.sleep(0.2)
        # spare connection pre-opened
        assert srv.conn_count >= 1
        resp = await agent.request_async("127.0.0.1", "GET", "/x")
        assert resp.status_code == 200
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_agent_create_pool_args():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": 1,  # wrong on purpose; create_pool overrides
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })
        agent.create_pool("127.0.0.1", {"port": srv.port})
        assert agent.get_pool("127.0.0.1") is not None
        with pytest.raises(Exception):
            agent.create_pool("127.0.0.1")
        resp = await agent.request_async("127.0.0.1", "GET", "/y")
        assert resp.status_code == 200
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_agent_pinger():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
            "ping": "/ping",
            "pingInterval": 150,
        })
        resp = await agent.request_async("127.0.0.1", "GET", "/warm")
        assert resp.status_code == 200
        await asyncio.sleep(0.8)
        assert srv.ping_count >= 2  # idle conn pinged repeatedly
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_agent_connection_refused():
    async def body():
        # a port with nothing listening
        probe = MockHttpServer()
        await probe.start()
        dead_port = probe.port
        probe.stop()
        await asyncio.sleep(0.05)

        agent = HttpAgent({
            "defaultPort": dead_port,
            "recovery": {"default": {"timeout": 500, "retries": 1,
                                     "delay": 10}},
            "spares": 1,
            "maximum": 1,
            "errorOnEmpty": False,
        })
        with pytest.raises(Exception):
            await asyncio.wait_for(
                agent.request_async("127.0.0.1", "GET", "/"), timeout=10)
        await stop_agent(agent)

    run(body())


def test_agent_on_broken_server():
    async def body():
        srv = MockHttpServer(broken=True)
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": {"default": {"timeout": 500, "retries": 1,
                                     "delay": 10}},
            "spares": 1,
            "maximum": 1,
        })
        with pytest.raises(Exception):
            await asyncio.wait_for(
                agent.request_async("127.0.0.1", "GET", "/"), timeout=10)
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_https_agent():
    async def body():
        server_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        server_ctx.load_cert_chain(
            os.path.join(FIXTURES, "test_cert.pem"),
            os.path.join(FIXTURES, "test_key.pem"))
        srv = MockHttpServer(tls_context=server_ctx)
        await srv.start()

        client_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
        client_ctx.load_verify_locations(
            os.path.join(FIXTURES, "test_cert.pem"))
        client_ctx.check_hostname = False

        agent = HttpsAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })
        agent.create_pool("127.0.0.1", {"port": srv.port,
                                        "ssl_context": client_ctx})
        resp = await agent.request_async("127.0.0.1", "GET", "/tls")
        assert resp.status_code == 200
        assert json.loads(resp.body)["path"] == "/tls"
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_agent_stopped_rejects_requests():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })
        await agent.request_async("127.0.0.1", "GET", "/")
        await stop_agent(agent)
        with pytest.raises(Exception):
            await agent.request_async("127.0.0.1", "GET", "/")
        srv.stop()

    run(body())


def test_connection_close_header_not_reused():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
        })
        resp = await agent.request_async("127.0.0.1", "GET", "/close")
        assert resp.status_code == 200
        assert resp.body == b"bye"
        await asyncio.sleep(0.3)
        resp2 = await agent.request_async("127.0.0.1", "GET", "/n")
        assert resp2.status_code == 200
        # server closed after /close: second request used a new conn
        assert srv.conn_count >= 2
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_pinger_5xx_closes_connection():
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
            "ping": "/err500",
            "pingInterval": 120,
        })
        await agent.request_async("127.0.0.1", "GET", "/warm")
        n0 = srv.conn_count
        await asyncio.sleep(0.8)
        # every ping 500s -> connection closed and re-made repeatedly
        assert srv.conn_count > n0
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_no_listener_accumulation_across_keepalive_requests():
    """Regression: each request's data/close/error listeners must be
    removed when it completes — bound-method listeners need ==
    comparison in remove_listener, not identity (native-core bug
    found by the leak detector under the agent benchmark)."""
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 1,
        })
        for _ in range(5):
            resp = await agent.request_async("127.0.0.1", "GET", "/x")
            assert resp.status_code == 200
        pool = agent.get_pool("127.0.0.1")
        socks = [f.get_socket_mgr().sm_socket
                 for fsms in pool.p_connections.values() for f in fsms]
        assert socks
        for sock in socks:
            # only the socket-manager's own listeners remain
            assert sock.listener_count("data") == 0
            assert sock.listener_count("close") <= 1
            assert sock.listener_count("error") <= 1
        await stop_agent(agent)
        srv.stop()

    run(body())


def test_remove_listener_bound_method_semantics():
    from cueball_amd.events import EventEmitter

    class Obj:
        def __init__(self):
            self.hits = 0

        def handler(self, *a):
            self.hits += 1

    em = EventEmitter()
    o = Obj()
    em.on("evt", o.handler)   # one bound-method object...
    em.emit("evt")
    em.remove_listener("evt", o.handler)  # ...a different, == one
    em.emit("evt")
    assert o.hits == 1
    assert em.listener_count("evt") == 0

    # once() removal by the inner bound method
    em.once("evt", o.handler)
    em.remove_listener("evt", o.handler)
    em.emit("evt")
    assert o.hits == 1


def test_agent_dns_hostname_pool():
    """A DNS-name host makes the agent build a DNS-SRV resolver pool
    (_http._tcp service) via resolver_for_ip_or_domain."""
    async def body():
        from cueball_amd.resolver import DNSResolverFSM
        from cueball_amd.testing import MockDnsServer

        loop = asyncio.get_running_loop()
        DNSResolverFSM._nic_cache = {"lo": [
            {"family": "IPv4", "address": "127.0.0.1"}]}
        DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0

        srv = MockHttpServer()
        await srv.start()
        dns = MockDnsServer()
        await dns.start()
        dns.add_srv("_http._tcp.web.test", "w1.web.test", srv.port,
                    ttl=60)
        dns.add_a("w1.web.test", "127.0.0.1", ttl=60)

        agent = HttpAgent({
            "defaultPort": 80,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
            "resolvers": [dns.resolver_address],
        })
        resp = await asyncio.wait_for(
            agent.request_async("web.test", "GET", "/dns-routed"),
            timeout=20)
        assert resp.status_code == 200
        assert json.loads(resp.body)["path"] == "/dns-routed"
        # the pool resolved through DNS SRV
        assert ("_http._tcp.web.test", "SRV") in dns.queries

        await stop_agent(agent)
        dns.stop()
        srv.stop()

    run(body())


def test_agent_stop_with_inflight_request():
    """Stopping the agent fails queued claims with PoolStoppingError;
    an in-flight request either completes or errors — never hangs."""
    async def body():
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 1,
        })
        # occupy the single connection
        resp = await agent.request_async("127.0.0.1", "GET", "/a")
        assert resp.status_code == 200

        # queue a request, then stop the agent in the same breath
        fut = asyncio.get_running_loop().create_future()

        def cb(err, resp=None):
            if not fut.done():
                fut.set_result((err, resp))

        agent.request("127.0.0.1", "GET", "/b", cb=cb)
        stop_fut = asyncio.get_running_loop().create_future()
        agent.stop(lambda e: stop_fut.set_result(None))
        err, resp2 = await asyncio.wait_for(fut, timeout=15)
        # either it squeaked through before the stop or it failed
        # cleanly — both acceptable; hanging is not
        assert err is not None or resp2.status_code == 200
        await asyncio.wait_for(stop_fut, timeout=15)
        srv.stop()

    run(body())


def test_streaming_request_body_chunked():
    """HttpRequest streaming uploads: write()/end() with chunked
    transfer-encoding (node ClientRequest parity)."""
    async def body():
        from cueball_amd.http_client import HttpRequest

        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 1,
        })
        req = HttpRequest("POST", "/echo", host="127.0.0.1",
                          streaming=True)
        fut = asyncio.get_running_loop().create_future()

        def on_resp(resp):
            if resp.complete:
                fut.set_result(resp)
            else:
                resp.on("end", lambda: fut.set_result(resp))

        req.on("response", on_resp)
        req.on("error", lambda e: fut.done() or fut.set_exception(e))

        # write a chunk BEFORE the socket is even claimed (buffered)
        req.write(b"hello ")
        agent.add_request(req, "127.0.0.1")
        await asyncio.sleep(0.05)
        # and more after the head went out
        req.write(b"streaming ")
        req.end(b"world")

        resp = await asyncio.wait_for(fut, timeout=15)
        assert resp.status_code == 200
        assert resp.body == b"hello streaming world"

        # the connection survives for keep-alive reuse
        resp2 = await agent.request_async("127.0.0.1", "GET", "/after")
        assert resp2.status_code == 200
        assert srv.conn_count == 1

        await stop_agent(agent)
        srv.stop()

    run(body())


def test_streaming_write_requires_flag():
    from cueball_amd.http_client import HttpRequest

    req = HttpRequest("POST", "/x", host="h")
    with pytest.raises(RuntimeError):
        req.write(b"nope")


def test_agent_with_custom_resolver():
    """create_pool(host, {resolver}) routes requests for an arbitrary
    hostname through a user-supplied resolver object
    (reference test/agent.test.js:214-265)."""
    async def body():
        from cueball_amd.resolver import StaticIpResolver
        srv1 = MockHttpServer()
        srv2 = MockHttpServer()
        await srv1.start()
        await srv2.start()
        res = StaticIpResolver({"backends": [
            {"address": "127.0.0.1", "port": srv1.port},
            {"address": "127.0.0.1", "port": srv2.port},
        ]})
        agent = HttpAgent({
            "recovery": RECOVERY,
            "spares": 2,
            "maximum": 4,
        })
        assert agent.get_pool("foobar") is None
        agent.create_pool("foobar", {"resolver": res})
        assert agent.get_pool("foobar") is not None
        assert res.is_in_state("stopped")
        res.start()
        resp = await asyncio.wait_for(
            agent.request_async("foobar", "GET", "/test4"), 10)
        assert resp.status_code == 200
        # the request landed on one of the two resolver-listed servers
        assert srv1.request_count + srv2.request_count >= 1
        await stop_agent(agent)
        srv1.stop()
        srv2.stop()

    run(body())
