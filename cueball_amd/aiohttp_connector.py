"""aiohttp integration: cueball-pooled connections for stock clients.

The reference's killer integration property is that its Agent
duck-types node's ``http.Agent`` and slots into existing consumers
unchanged (lib/agent.js:30-44, :275-396).  The Python ecosystem analog
is an :class:`aiohttp.BaseConnector`: hand
``ClientSession(connector=CueballConnector(...))`` to any aiohttp code
and every request claims its socket from a cueball ConnectionPool —
DNS-SRV service discovery, spares/maximum sizing, recovery backoff,
CoDel shedding and Kang introspection included — with zero changes to
the calling code.

Usage::

    from cueball_amd.aiohttp_connector import CueballConnector
    connector = CueballConnector(
        recovery={"default": {"timeout": 2000, "retries": 3,
                              "delay": 100, "maxDelay": 2000}},
        spares=4, maximum=16)
    async with aiohttp.ClientSession(connector=connector) as sess:
        async with sess.get("http://svc.example:8080/x") as resp:
            ...

One pool is created per (host, port, ssl) connection key, each backed
by ``resolver_for_ip_or_domain`` (static resolver for IP literals, the
DNS SRV/A resolver for names), exactly like the reference Agent's
pool-per-host factory (lib/agent.js:105-211).
"""

from __future__ import annotations

import asyncio
from typing import Any, Dict, Optional

try:
    from aiohttp.client_proto import ResponseHandler
    from aiohttp.connector import BaseConnector, Connection
    HAVE_AIOHTTP = True
except ImportError:  # pragma: no cover - aiohttp is an optional extra
    HAVE_AIOHTTP = False
    BaseConnector = object  # type: ignore[assignment,misc]

from . import errors as mod_errors
from .events import EventEmitter
from .logutil import default_logger
from .pool import ConnectionPool
from .resolver import resolver_for_ip_or_domain

__all__ = ["CueballConnector", "AioConnection", "HAVE_AIOHTTP"]


class _Handler(ResponseHandler if HAVE_AIOHTTP else object):
    """aiohttp ResponseHandler that reports connection loss to the
    owning AioConnection so cueball's slot FSM sees socket death."""

    def __init__(self, owner: "AioConnection",
                 loop: asyncio.AbstractEventLoop) -> None:
        super().__init__(loop)
        self._cueball_owner = owner

    def connection_lost(self, exc: Optional[BaseException]) -> None:
        super().connection_lost(exc)
        owner = self._cueball_owner
        if owner is not None:
            self._cueball_owner = None
            owner._on_lost(exc)


class AioConnection(EventEmitter):
    """cueball Connection contract (docs/api.adoc:580-645) wrapping an
    aiohttp transport+ResponseHandler pair.

    Emits ``connect`` once the transport is up, ``error`` on connect
    failure or mid-life transport error, ``close`` when the transport
    goes away; ``destroy()`` tears the transport down.  The live
    ResponseHandler is exposed as ``aio_protocol`` for the connector.
    """

    def __init__(self, backend: Dict[str, Any], *,
                 ssl: Any = None, server_hostname: Optional[str] = None,
                 loop: Optional[asyncio.AbstractEventLoop] = None) -> None:
        super().__init__()
        self.backend = backend
        self._loop = loop or asyncio.get_event_loop()
        self._ssl = ssl
        self._server_hostname = server_hostname
        self.aio_protocol: Optional[ResponseHandler] = None
        self.aio_transport = None
        self._destroyed = False
        self._connected = False
        self._task = self._loop.create_task(self._connect())

    async def _connect(self) -> None:
        try:
            kwargs: Dict[str, Any] = {}
            if self._ssl is not None:
                kwargs["ssl"] = self._ssl
                if self._server_hostname:
                    kwargs["server_hostname"] = self._server_hostname
            tr, proto = await self._loop.create_connection(
                lambda: _Handler(self, self._loop),
                self.backend.get("address") or self.backend.get("name"),
                self.backend["port"], **kwargs)
        except Exception as e:  # noqa: BLE001 - delivered as an event
            if not self._destroyed:
                self.emit("error", e)
            return
        if self._destroyed:
            tr.close()
            return
        self.aio_transport = tr
        self.aio_protocol = proto
        self._connected = True
        self.emit("connect")

    def _on_lost(self, exc: Optional[BaseException]) -> None:
        if self._destroyed:
            return
        was_connected = self._connected
        self._connected = False
        if exc is not None and was_connected:
            self.emit("error", exc)
        else:
            self.emit("close")

    # -- cueball Connection contract --------------------------------
    def destroy(self) -> None:
        self._destroyed = True
        if self.aio_transport is not None:
            self.aio_transport.close()
            self.aio_transport = None
        elif not self._task.done():
            self._task.cancel()

    def ref(self) -> None:  # optional contract methods
        pass

    def unref(self) -> None:
        pass


class CueballConnector(BaseConnector):
    """aiohttp connector backed by cueball ConnectionPools.

    Overrides ``connect``/``_release`` so acquisition and keep-alive
    go through a cueball pool per connection key instead of aiohttp's
    built-in per-host deques; everything else (request writing,
    response parsing, session lifecycle) is stock aiohttp.
    """

    def __init__(self, *, recovery: Optional[Dict[str, Any]] = None,
                 spares: int = 2, maximum: int = 16,
                 resolvers: Optional[list] = None,
                 service: str = "_http._tcp",
                 target_claim_delay: Optional[float] = None,
                 claim_timeout: Optional[float] = None,
                 log: Any = None,
                 loop: Optional[asyncio.AbstractEventLoop] = None,
                 **kwargs: Any) -> None:
        if not HAVE_AIOHTTP:
            raise RuntimeError("aiohttp is not installed")
        super().__init__(loop=loop, **kwargs)
        self._cb_recovery = recovery or {
            "default": {"timeout": 5000, "retries": 3, "delay": 250,
                        "maxDelay": 5000}}
        self._cb_spares = spares
        self._cb_maximum = maximum
        self._cb_resolvers = resolvers
        self._cb_service = service
        self._cb_target_claim_delay = target_claim_delay
        self._cb_claim_timeout = claim_timeout
        self._cb_log = log or default_logger()
        self._cb_pools: Dict[Any, ConnectionPool] = {}
        self._cb_pool_resolvers: Dict[Any, Any] = {}
        self._cb_handles: Dict[Any, Any] = {}  # ResponseHandler -> handle

    # -- pool factory (the Agent's _addPool analog) ------------------
    def _pool_for(self, req: Any) -> ConnectionPool:
        key = req.connection_key
        pool = self._cb_pools.get(key)
        if pool is not None:
            return pool

        host = key.host
        port = key.port
        use_ssl = key.is_ssl
        sslctx = req.ssl if use_ssl else None
        if sslctx is True or (use_ssl and sslctx is None):
            import ssl as mod_ssl
            sslctx = mod_ssl.create_default_context()

        res_cfg: Dict[str, Any] = {
            "input": "%s:%d" % (host, port),
            "resolverConfig": {
                "recovery": self._cb_recovery,
                "service": self._cb_service,
                "defaultPort": port,
                "loop": self._loop,
            },
        }
        if self._cb_resolvers:
            res_cfg["resolverConfig"]["resolvers"] = self._cb_resolvers
        resolver = resolver_for_ip_or_domain(res_cfg)

        loop = self._loop
        server_hostname = host if use_ssl else None
        ssl_for_pool = sslctx

        def constructor(backend: Dict[str, Any]) -> AioConnection:
            return AioConnection(backend, ssl=ssl_for_pool,
                                 server_hostname=server_hostname,
                                 loop=loop)

        pool_opts: Dict[str, Any] = {
            "domain": host,
            "constructor": constructor,
            "resolver": resolver,
            "recovery": self._cb_recovery,
            "spares": self._cb_spares,
            "maximum": self._cb_maximum,
            "log": self._cb_log,
            "loop": loop,
        }
        if self._cb_target_claim_delay is not None:
            pool_opts["targetClaimDelay"] = self._cb_target_claim_delay
        pool = ConnectionPool(pool_opts)
        resolver.start()
        self._cb_pools[key] = pool
        self._cb_pool_resolvers[key] = resolver
        return pool

    # -- acquisition / release ---------------------------------------
    async def connect(self, req: Any, traces: list,
                      timeout: Any) -> "Connection":
        pool = self._pool_for(req)
        claim_opts: Dict[str, Any] = {}
        if self._cb_claim_timeout is not None:
            claim_opts["timeout"] = self._cb_claim_timeout
        try:
            hdl, conn = await pool.claim_async(claim_opts)
        except mod_errors.CueballError as e:
            import aiohttp
            raise aiohttp.ClientConnectionError(str(e)) from e
        proto = conn.aio_protocol
        if proto is None or not proto.is_connected():
            # lost the socket between idle and claim: surrender this
            # one and retry once
            hdl.close()
            hdl, conn = await pool.claim_async(claim_opts)
            proto = conn.aio_protocol
            if proto is None or not proto.is_connected():
                hdl.close()
                import aiohttp
                raise aiohttp.ClientConnectionError(
                    "claimed connection is not usable")
        self._cb_handles[proto] = hdl
        self._acquired.add(proto)
        return Connection(self, req.connection_key, proto, self._loop)

    def _release(self, key: Any, protocol: Any, *,
                 should_close: bool = False) -> None:
        self._acquired.discard(protocol)
        hdl = self._cb_handles.pop(protocol, None)
        if hdl is None:
            return
        if should_close or protocol.should_close:
            hdl.close()
        else:
            hdl.release()

    def _release_acquired(self, key: Any, proto: Any) -> None:
        # aiohttp's cancellation path: treat like a close-release
        self._acquired.discard(proto)
        hdl = self._cb_handles.pop(proto, None)
        if hdl is not None:
            hdl.close()

    async def close(self) -> None:  # type: ignore[override]
        for proto, hdl in list(self._cb_handles.items()):
            try:
                hdl.close()
            except mod_errors.CueballError:
                pass
        self._cb_handles.clear()
        waits = []
        for key, pool in self._cb_pools.items():
            fut = self._loop.create_future()

            def on_state(st: str, fut=fut) -> None:
                if st == "stopped" and not fut.done():
                    fut.set_result(None)

            pool.on("stateChanged", on_state)
            pool.stop()
            if pool.is_in_state("stopped") and not fut.done():
                fut.set_result(None)
            waits.append(fut)
        if waits:
            await asyncio.wait(waits, timeout=5)
        self._cb_pools.clear()
        self._cb_pool_resolvers.clear()
        self._closed = True

    def pool_stats(self) -> Dict[str, Dict[str, Any]]:
        """Expose per-host cueball pool stats (beyond aiohttp)."""
        return {"%s:%d" % (k.host, k.port): p.get_stats()
                for (k, p) in self._cb_pools.items()}
