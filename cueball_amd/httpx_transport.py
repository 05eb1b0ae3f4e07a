"""httpx integration: cueball-pooled transport for stock clients.

The second ecosystem adapter (alongside
:mod:`cueball_amd.aiohttp_connector`): an
:class:`httpx.AsyncBaseTransport` that routes every request through a
cueball HttpAgent/HttpsAgent, so unchanged httpx code gets pool-per-
host keep-alive sockets, DNS-SRV service discovery, recovery backoff
and Kang/Prometheus introspection.

Usage::

    from cueball_amd.httpx_transport import CueballTransport
    transport = CueballTransport(
        recovery={"default": {"timeout": 2000, "retries": 3,
                              "delay": 100, "maxDelay": 2000}},
        spares=2, maximum=16)
    async with httpx.AsyncClient(transport=transport) as client:
        r = await client.get("http://svc.example:8080/x")

Responses are buffered (the agent machinery collects the body before
completing), which matches httpx's default non-streaming usage;
``client.stream(...)`` works but does not stream incrementally.
Duplicate response headers (e.g. multiple ``Set-Cookie``) are folded
into one header, a limitation of the underlying header dict.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

try:
    import httpx
    HAVE_HTTPX = True
except ImportError:  # pragma: no cover - httpx is an optional extra
    HAVE_HTTPX = False

from . import errors as mod_errors

__all__ = ["CueballTransport", "HAVE_HTTPX"]

_BaseTransport = httpx.AsyncBaseTransport if HAVE_HTTPX else object


class CueballTransport(_BaseTransport):  # type: ignore[misc,valid-type]
    """httpx transport backed by cueball agents (one HttpAgent and one
    HttpsAgent, each with a pool per host, lib/agent.js:105-211)."""

    def __init__(self, *, recovery: Optional[Dict[str, Any]] = None,
                 spares: int = 2, maximum: int = 16,
                 resolvers: Optional[list] = None,
                 ping: Optional[str] = None,
                 ping_interval: Optional[float] = None,
                 agent_options: Optional[Dict[str, Any]] = None) -> None:
        if not HAVE_HTTPX:
            raise RuntimeError("httpx is not installed")
        self._opts: Dict[str, Any] = {
            "recovery": recovery or {
                "default": {"timeout": 5000, "retries": 3, "delay": 250,
                            "maxDelay": 5000}},
            "spares": spares,
            "maximum": maximum,
        }
        if resolvers:
            self._opts["resolvers"] = resolvers
        if ping is not None:
            self._opts["ping"] = ping
        if ping_interval is not None:
            self._opts["pingInterval"] = ping_interval
        if agent_options:
            self._opts.update(agent_options)
        self._http_agent = None
        self._https_agent = None

    def _agent(self, scheme: str):
        from .agent import HttpAgent, HttpsAgent

        if scheme == "https":
            if self._https_agent is None:
                self._https_agent = HttpsAgent(dict(self._opts))
            return self._https_agent
        if self._http_agent is None:
            self._http_agent = HttpAgent(dict(self._opts))
        return self._http_agent

    async def handle_async_request(self, request: Any) -> Any:
        url = request.url
        agent = self._agent(url.scheme)
        port = url.port or (443 if url.scheme == "https" else 80)

        body = await request.aread()
        headers: Dict[str, str] = {}
        for name, value in request.headers.items():
            # drop hop-by-hop framing; our serializer regenerates it
            if name.lower() in ("content-length", "transfer-encoding",
                                "connection"):
                continue
            headers[name] = value

        target = url.raw_path.decode("ascii")
        try:
            resp = await agent.request_async(
                {"host": url.host, "port": port},
                request.method, target, headers=headers,
                body=body or None)
        except mod_errors.CueballError as e:
            raise httpx.ConnectError(str(e), request=request) from e

        return httpx.Response(
            resp.status_code,
            headers=list(resp.headers.items()),
            content=resp.body,
            request=request,
            extensions={"http_version": (
                "HTTP/%s" % resp.http_version).encode()},
        )

    async def aclose(self) -> None:
        import asyncio

        for agent in (self._http_agent, self._https_agent):
            if agent is None:
                continue
            loop = agent._loop
            fut = loop.create_future()
            agent.stop(lambda e=None, fut=fut:
                       fut.done() or fut.set_result(None))
            try:
                await asyncio.wait_for(fut, timeout=5)
            except asyncio.TimeoutError:  # pragma: no cover
                pass
        self._http_agent = None
        self._https_agent = None

    def pool_stats(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        for label, agent in (("http", self._http_agent),
                             ("https", self._https_agent)):
            if agent is None:
                continue
            for host, pool in agent.pools.items():
                out["%s://%s" % (label, host)] = pool.get_stats()
        return out
