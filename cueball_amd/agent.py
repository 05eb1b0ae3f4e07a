"""HTTP(S) agents: pooled keep-alive HTTP clients (reference lib/agent.js).

``HttpAgent``/``HttpsAgent`` auto-create one ConnectionPool per host
(static or DNS-resolved via resolver_for_ip_or_domain), hand claimed
connections to requests, and run the optional HTTP ping health check on
idle connections.  The reference duck-types node's ``http.Agent``; the
rebuild keeps the same architecture against its own HttpRequest
machinery (cueball_amd.http_client):

- ``add_request(req, options)`` — the low-level socket-event protocol:
  claim -> req.on_socket(conn); 'free' => release, 'close' => disable
  leak check + release (benefit of the doubt), req 'abort' => cancel or
  close, 'agentRemove' => hold until close (lib/agent.js:275-396).
- ``request(...)``/``request_async(...)`` — convenience wrappers.
- ping checker: claims the exact idle socket and runs GET <ping>; a 5xx
  or error closes the connection, anything else releases it
  (lib/agent.js:398-455, PingAgent :530-569).
"""

from __future__ import annotations

import ssl as mod_ssl
from typing import Any, Callable, Dict, Optional

from . import utils as mod_utils
from .connection import TcpConnection
from .errors import CueballError
from .events import EventEmitter
from .fsm import get_loop
from .http_client import HttpRequest, HttpResponse
from .logutil import CueballLogger, default_logger
from .pool import ConnectionPool
from .resolver import resolver_for_ip_or_domain

__all__ = ["Agent", "HttpAgent", "HttpsAgent", "FakeSocket", "PingAgent"]

#: TLS/connect fields passed through from request options to the socket
#: constructor (lib/agent.js:96-97)
PASS_FIELDS = ("ssl_context", "server_hostname", "cert", "key", "ca",
               "rejectUnauthorized")


class FakeSocket(EventEmitter):
    """Error-delivery stub: addRequest cannot return an async error any
    other way (lib/agent.js:306-317, :518-527)."""

    _is_fake_socket = True

    def read(self) -> None:
        return None

    def destroy(self) -> None:
        return None


class Agent(EventEmitter):
    def __init__(self, options: Dict[str, Any]) -> None:
        super().__init__()
        if not isinstance(options.get("defaultPort"), int):
            raise TypeError("options.defaultPort is required")
        if not isinstance(options.get("protocol"), str):
            raise TypeError("options.protocol is required")

        self.collector = mod_utils.create_error_metrics(options)
        self._loop = get_loop(options.get("loop"))

        self.default_port: int = options["defaultPort"]
        self.protocol: str = options["protocol"] + ":"
        self.service = "_" + options["protocol"] + "._tcp"

        self.keep_alive = True
        self.tcp_ka_delay = options.get("tcpKeepAliveInitialDelay")

        self.pools: Dict[str, ConnectionPool] = {}
        self.pool_resolvers: Dict[str, Any] = {}
        self.pool_external_resolvers: Dict[str, Any] = {}
        self.resolvers = options.get("resolvers")
        log: CueballLogger = options.get("log") or default_logger()
        self.log = log.child(component="CueBallAgent")
        self.cba_stopped = False

        spares = options.get("spares")
        maximum = options.get("maximum")
        if not isinstance(spares, int) or not isinstance(maximum, int):
            raise TypeError("options.spares and options.maximum required")
        self.spares = spares
        self.maximum = maximum

        self.cba_ping: Optional[str] = options.get("ping")
        self.cba_ping_interval: Optional[float] = options.get("pingInterval")

        recovery = options.get("recovery")
        if not isinstance(recovery, dict):
            raise TypeError("options.recovery is required")
        mod_utils.assert_recovery(recovery.get("default"), "recovery.default")
        self.cba_recovery = recovery

        self.cba_err_on_empty = bool(options.get("errorOnEmpty"))

        for host in options.get("initialDomains") or ():
            self._add_pool(host, {})

    # -- pool management --------------------------------------------------
    def _add_pool(self, host: str, options: Dict[str, Any]) -> None:
        if self.cba_stopped:
            raise CueballError("Cannot add a pool to a stopped agent")
        def_port = self.default_port
        port_opt = options.get("port")
        if isinstance(port_opt, str):
            port_opt = int(port_opt)
        if isinstance(port_opt, int):
            def_port = port_opt

        use_external = options.get("resolver") is not None
        if use_external:
            res = options["resolver"]
        else:
            res = resolver_for_ip_or_domain({
                "input": host,
                "resolverConfig": {
                    "resolvers": self.resolvers,
                    "service": self.service,
                    "defaultPort": def_port,
                    "recovery": self.cba_recovery,
                    "log": self.log,
                    "loop": self._loop,
                },
            })
            if isinstance(res, Exception):
                raise res

        tls = self.protocol == "https:"
        agent = self

        def construct_socket(backend: Dict[str, Any]) -> TcpConnection:
            sslctx = options.get("ssl_context")
            if tls and sslctx is None:
                sslctx = mod_ssl.create_default_context()
                if options.get("ca") is not None:
                    sslctx = mod_ssl.create_default_context(
                        cadata=options["ca"])
                if options.get("rejectUnauthorized") is False:
                    sslctx.check_hostname = False
                    sslctx.verify_mode = mod_ssl.CERT_NONE
            conn = TcpConnection(
                {
                    "key": backend.get("key"),
                    "name": backend.get("name") or host,
                    "address": backend.get("address") or backend.get("name"),
                    "port": backend.get("port") or def_port,
                },
                loop=agent._loop,
                tls=tls,
                ssl_context=sslctx,
                server_hostname=options.get("server_hostname")
                or backend.get("name") or host,
            )
            if agent.tcp_ka_delay is not None:
                def enable_ka() -> None:
                    sock = None
                    if conn._transport is not None:
                        sock = conn._transport.get_extra_info("socket")
                    if sock is not None:
                        import socket as mod_socket
                        sock.setsockopt(mod_socket.SOL_SOCKET,
                                        mod_socket.SO_KEEPALIVE, 1)
                        # node's setKeepAlive(initialDelay) also sets the
                        # idle time before the first probe; map the ms
                        # option onto TCP_KEEPIDLE (whole seconds, >=1).
                        idle_s = max(1, int(agent.tcp_ka_delay / 1000))
                        for opt in ("TCP_KEEPIDLE", "TCP_KEEPALIVE"):
                            if hasattr(mod_socket, opt):
                                try:
                                    sock.setsockopt(
                                        mod_socket.IPPROTO_TCP,
                                        getattr(mod_socket, opt), idle_s)
                                except OSError:
                                    pass
                                break
                conn.on("connect", enable_ka)
            return conn

        pool_opts: Dict[str, Any] = {
            "resolver": res,
            "domain": host,
            "constructor": construct_socket,
            "maximum": self.maximum,
            "spares": self.spares,
            "log": self.log,
            "recovery": self.cba_recovery,
            "collector": self.collector,
            "loop": self._loop,
        }
        if self.cba_ping is not None:
            pool_opts["checkTimeout"] = self.cba_ping_interval or 30000
            pool_opts["checker"] = \
                lambda hdl, sock: self._check_socket(host, hdl, sock)

        self.log.debug("CueBallAgent creating new pool", host=host)
        self.pools[host] = ConnectionPool(pool_opts)
        if use_external:
            self.pool_external_resolvers[host] = res
        else:
            res.start()
            self.pool_resolvers[host] = res

    def get_pool(self, host: str) -> Optional[ConnectionPool]:
        return self.pools.get(host)

    def create_pool(self, host: str,
                    options: Optional[Dict[str, Any]] = None) -> None:
        if host in self.pools:
            raise CueballError("Attempting to create a pool for a hostname "
                               "that already has one.")
        self._add_pool(host, dict(options or {}))

    def create_connection(self, options: Any = None,
                          connect_listener: Any = None) -> None:
        """UNIX-socket path: unsupported, as in the reference
        (lib/agent.js:492-495)."""
        raise CueballError("UNIX domain sockets not supported")

    def is_stopped(self) -> bool:
        return self.cba_stopped

    def stop(self, cb: Optional[Callable] = None) -> None:
        if self.cba_stopped:
            raise CueballError("Cannot stop a CueBallAgent that has "
                               "already stopped")
        self.cba_stopped = True
        self.log.debug("CueBallAgent stopping all pools")
        hosts = list(self.pools.keys())
        remaining = {"n": len(hosts)}

        def done_one() -> None:
            remaining["n"] -= 1
            if remaining["n"] == 0 and cb is not None:
                self._loop.call_soon(lambda: cb(None))

        if not hosts:
            if cb is not None:
                self._loop.call_soon(lambda: cb(None))
            return

        for host in hosts:
            pool = self.pools.pop(host)
            res = self.pool_resolvers.pop(host, None)
            ext = res is None
            if ext:
                res = self.pool_external_resolvers.pop(host, None)

            def make_on_stopped(p=pool, r=res, is_ext=ext):
                fired = {"done": False}

                def on_state(st: str) -> None:
                    if fired["done"] or st != "stopped":
                        return
                    fired["done"] = True
                    if not is_ext and r is not None and \
                            not r.is_in_state("stopped"):
                        r.stop()
                    done_one()
                return on_state

            if pool.is_in_state("stopped"):
                if not ext and res is not None and \
                        not res.is_in_state("stopped"):
                    res.stop()
                done_one()
            else:
                pool.on("stateChanged", make_on_stopped())
                pool.stop()

    # -- the request path ---------------------------------------------------
    def add_request(self, req: HttpRequest,
                    options_or_host: Any, port: Optional[int] = None) -> None:
        """Low-level: claim a connection and wire the socket-event
        protocol (lib/agent.js:275-396)."""
        if self.cba_stopped:
            raise CueballError("CueBallAgent is stopped and cannot handle "
                               "new requests")
        if isinstance(options_or_host, str):
            options: Dict[str, Any] = {"host": options_or_host}
            if port is not None:
                options["port"] = port
        else:
            options = dict(options_or_host or {})
        host = options.get("host") or options.get("hostname")
        if not isinstance(host, str):
            raise TypeError("hostname is required")
        if host not in self.pools:
            self._add_pool(host, options)
        pool = self.pools[host]
        _RequestTicket(self, pool, req)

    def request(self, options_or_host: Any, method: str = "GET",
                path: str = "/", headers: Optional[Dict[str, str]] = None,
                body: Optional[bytes] = None,
                cb: Optional[Callable] = None) -> HttpRequest:
        """Convenience: run one request; cb(err, response) after the
        body has fully arrived."""
        if isinstance(options_or_host, str):
            host = options_or_host
        else:
            host = options_or_host.get("host")
        req = HttpRequest(method, path, headers=headers, body=body,
                          host=host)

        if cb is not None:
            def on_resp(resp: HttpResponse) -> None:
                if resp.complete:
                    cb(None, resp)
                else:
                    resp.on("end", lambda: cb(None, resp))

            req.on("response", on_resp)
            req.on("error", lambda e: cb(e, None))
        self.add_request(req, options_or_host)
        return req

    async def request_async(self, options_or_host: Any, method: str = "GET",
                            path: str = "/",
                            headers: Optional[Dict[str, str]] = None,
                            body: Optional[bytes] = None) -> HttpResponse:
        fut = self._loop.create_future()

        def cb(err: Optional[BaseException],
               resp: Optional[HttpResponse]) -> None:
            if fut.done():
                return
            if err is not None:
                fut.set_exception(err)
            else:
                fut.set_result(resp)

        self.request(options_or_host, method, path, headers, body, cb)
        return await fut

    # -- health checking -----------------------------------------------------
    def _check_socket(self, host: str, handle: Any, socket: Any) -> None:
        t1 = self._loop.time()
        log = self.log.child(component="CueBallAgentPing", domain=host,
                             path=self.cba_ping)
        agent = PingAgent({"protocol": self.protocol, "socket": socket,
                           "log": log})
        req = HttpRequest("GET", self.cba_ping, host=host)

        def on_resp(resp: HttpResponse) -> None:
            def finish() -> None:
                if 500 <= resp.status_code < 600:
                    log.warn("got a 5xx code, closing",
                             statusCode=resp.status_code,
                             latency=round(
                                 (self._loop.time() - t1) * 1000, 2))
                    handle.close()
                else:
                    log.debug("health check ok, releasing",
                              statusCode=resp.status_code)
                    handle.release()

            if resp.complete:
                finish()
            else:
                resp.on("end", finish)

        req.on("response", on_resp)
        req.once("error", lambda e: (
            log.warn("check failed: %s", e,
                     latency=round((self._loop.time() - t1) * 1000, 2)),
            handle.close()))
        agent.add_request(req, {})


class _RequestTicket:
    """One request's claim + socket-event protocol (the closure-free
    form of lib/agent.js:296-396; this sits on the HTTP hot path)."""

    __slots__ = ("agent", "req", "waiter", "conn", "sock")

    def __init__(self, agent: "Agent", pool: Any, req: HttpRequest) -> None:
        self.agent = agent
        self.req = req
        self.conn: Any = None
        self.sock: Any = None
        req.once("abort", self.on_abort)
        self.waiter = pool.claim(
            {"errorOnEmpty": agent.cba_err_on_empty}, self.claimed)

    def claimed(self, err: Optional[BaseException], connh: Any = None,
                socket: Any = None) -> None:
        self.waiter = None
        if err is not None:
            fakesock = FakeSocket()
            self.req.on_socket(fakesock)
            self.agent._loop.call_soon(
                lambda: fakesock.emit("error", err))
            return
        self.conn = connh
        self.sock = socket
        socket.once("free", self.on_free)
        socket.once("close", self.on_close)
        socket.once("agentRemove", self.on_agent_remove)
        self.req.on_socket(socket)

    def on_abort(self) -> None:
        if self.waiter is not None:
            self.waiter.cancel()
            self.waiter = None
        if self.conn is not None:
            sock = self.sock
            sock.remove_listener("close", self.on_close)
            sock.remove_listener("free", self.on_free)
            sock.remove_listener("agentRemove", self.on_agent_remove)
            self.conn.close()
            self.conn = None
            self.sock = None

    def on_close(self) -> None:
        sock = self.sock
        sock.remove_listener("free", self.on_free)
        sock.remove_listener("agentRemove", self.on_agent_remove)
        self.req.remove_listener("abort", self.on_abort)
        # A 'close' straight after a normally-completed request is
        # indistinguishable from a premature close; give it the benefit
        # of the doubt: disable the leak check and release rather than
        # close (lib/agent.js:328-360).
        conn = self.conn
        conn.disable_release_leak_check()
        conn.release()
        self.conn = None
        self.sock = None

    def on_free(self) -> None:
        sock = self.sock
        sock.remove_listener("close", self.on_close)
        sock.remove_listener("agentRemove", self.on_agent_remove)
        self.req.remove_listener("abort", self.on_abort)
        self.conn.release()
        self.conn = None
        self.sock = None

    def on_agent_remove(self) -> None:
        # Upgrade etc.: the socket now belongs to someone else; hold
        # the lease until 'close'.
        self.sock.remove_listener("free", self.on_free)
        self.req.remove_listener("abort", self.on_abort)


class PingAgent(EventEmitter):
    """Runs a request on one specific, already-claimed socket
    (lib/agent.js:530-569)."""

    def __init__(self, options: Dict[str, Any]) -> None:
        super().__init__()
        self.protocol = options["protocol"]
        self.keep_alive = True
        self.pa_socket = options["socket"]
        self.log = options.get("log")

    def add_request(self, req: HttpRequest,
                    options: Optional[Dict[str, Any]] = None) -> None:
        sock = self.pa_socket

        def on_abort() -> None:
            sock.remove_listener("free", on_free)
            sock.remove_listener("agentRemove", on_agent_remove)

        def on_free() -> None:
            sock.remove_listener("agentRemove", on_agent_remove)
            req.remove_listener("abort", on_abort)

        def on_agent_remove() -> None:
            sock.remove_listener("free", on_free)
            req.remove_listener("abort", on_abort)

        sock.once("free", on_free)
        sock.once("agentRemove", on_agent_remove)
        req.once("abort", on_abort)
        req.on_socket(sock)


class HttpAgent(Agent):
    def __init__(self, options: Dict[str, Any]) -> None:
        options = dict(options)
        options["protocol"] = "http"
        options.setdefault("defaultPort", 80)
        super().__init__(options)


class HttpsAgent(Agent):
    def __init__(self, options: Dict[str, Any]) -> None:
        options = dict(options)
        options["protocol"] = "https"
        options.setdefault("defaultPort", 443)
        super().__init__(options)
