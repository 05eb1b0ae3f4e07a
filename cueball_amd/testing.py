"""Test harness: virtual-time event loop and fake components.

The reference's test kit (survey §4) drives multi-node behavior with
in-process fakes: a fake resolver (test/pool.test.js:45-67), fake
connections (test/pool.test.js:69-98) and a fake DNS client
(test/dns.test.js:75-110).  This module provides the same kit for the
rebuild, plus a deterministic virtual-clock event loop so backoff /
CoDel / TTL logic can be tested in milliseconds of real time.
"""

from __future__ import annotations

import asyncio
import json
import selectors
from typing import Any, Callable, Dict, List, Optional, Tuple

from . import dns_wire
from .events import EventEmitter

__all__ = [
    "VirtualLoop",
    "advance",
    "settle",
    "run",
    "DummyConnection",
    "DummyResolver",
    "MockDnsServer",
]


class VirtualLoop(asyncio.SelectorEventLoop):
    """An asyncio event loop whose clock only moves when told to.

    ``loop.time()`` returns virtual seconds.  Use ``await advance(loop, s)``
    inside a coroutine running on the loop to fast-forward: all timers due
    within the window fire in order, at their scheduled virtual times.
    """

    def __init__(self) -> None:
        super().__init__(selectors.SelectSelector())
        self._vtime = 1_000_000.0

    def time(self) -> float:  # overrides BaseEventLoop.time
        return self._vtime

    def _vt_set(self, t: float) -> None:
        if t < self._vtime:
            raise ValueError("virtual time cannot go backwards")
        self._vtime = t


async def settle(loop: Optional[asyncio.AbstractEventLoop] = None,
                 rounds: int = 1000) -> None:
    """Let all ready (call_soon) callbacks run, including cascades."""
    loop = loop or asyncio.get_running_loop()
    for _ in range(rounds):
        if not loop._ready:  # type: ignore[attr-defined]
            return
        await asyncio.sleep(0)
    raise RuntimeError("event loop did not settle after %d rounds" % rounds)


def _next_timer(loop: asyncio.AbstractEventLoop) -> Optional[float]:
    times = [h._when for h in loop._scheduled  # type: ignore[attr-defined]
             if not h._cancelled]
    return min(times) if times else None


async def advance(loop: asyncio.AbstractEventLoop, seconds: float) -> None:
    """Fast-forward a VirtualLoop by `seconds`, firing due timers in order."""
    if not isinstance(loop, VirtualLoop):
        raise TypeError("advance() requires a VirtualLoop")
    await settle(loop)
    target = loop.time() + seconds
    while True:
        nxt = _next_timer(loop)
        if nxt is None or nxt > target:
            break
        loop._vt_set(nxt)
        await asyncio.sleep(0)  # let the due timers move to ready...
        await settle(loop)      # ...and run, with any cascades
    loop._vt_set(target)
    await asyncio.sleep(0)
    await settle(loop)


def run(coro_fn: Callable[[asyncio.AbstractEventLoop], Any]) -> Any:
    """Run async test body on a fresh VirtualLoop: run(lambda loop: body(loop))."""
    loop = VirtualLoop()
    try:
        return loop.run_until_complete(coro_fn(loop))
    finally:
        loop.close()


class DummyConnection(EventEmitter):
    """Scripted fake of the user Connection interface (docs/api.adoc:580-645).

    Tests construct pools with ``constructor=lambda backend:
    DummyConnection(backend, registry)`` and then drive each instance by
    hand: ``conn.connect()``, ``conn.emit('error', e)``, etc., exactly
    like the reference's DummyConnection (test/pool.test.js:69-98).
    """

    def __init__(self, backend: Dict[str, Any],
                 registry: Optional[List["DummyConnection"]] = None) -> None:
        super().__init__()
        self.backend = backend
        self.connected = False
        self.dead = False
        self.ref_count = 0
        self.seen_unwanted = False
        if registry is not None:
            registry.append(self)

    def connect(self) -> None:
        self.connected = True
        self.emit("connect")

    def destroy(self) -> None:
        self.dead = True
        self.connected = False

    def set_unwanted(self) -> None:
        self.seen_unwanted = True

    def ref(self) -> None:
        self.ref_count += 1

    def unref(self) -> None:
        self.ref_count -= 1


def dummy_resolver_factory(registry: List["DummyResolver"]):
    """Factory usable as a monkeypatch for ``cueball_amd.resolver.Resolver``
    so a pool/set builds a DummyResolver instead of a DNS resolver — the
    rebuild's equivalent of the reference's sinon stub
    (test/pool.test.js:100-104).  Each created inner resolver is appended
    to `registry`."""
    from .resolver import ResolverFSM

    def factory(options):
        inner = DummyResolver()
        registry.append(inner)
        return ResolverFSM(inner, {"loop": options.get("loop")})

    return factory


class DummyResolver(EventEmitter):
    """Bare fake of the Resolver interface (reference test/pool.test.js:45-67).

    Drive topology from tests with ``resolver.add('b1', {"address": ...,
    "port": ...})`` / ``resolver.remove('b1')``.
    """

    def __init__(self, backends: Optional[Dict[str, Dict[str, Any]]] = None) -> None:
        super().__init__()
        self.state = "stopped"
        self.backends: Dict[str, Dict[str, Any]] = {}
        self._initial = dict(backends or {})
        self._last_error: Optional[BaseException] = None

    # Resolver interface ------------------------------------------------
    def start(self) -> None:
        self.state = "running"
        for k, b in self._initial.items():
            self.add(k, b)

    def stop(self) -> None:
        self.state = "stopped"

    def count(self) -> int:
        return len(self.backends)

    def list(self) -> Dict[str, Dict[str, Any]]:
        return dict(self.backends)

    def get_last_error(self) -> Optional[BaseException]:
        return self._last_error

    def is_in_state(self, state: str) -> bool:
        return self.state == state

    def get_state(self) -> str:
        return self.state

    # test-side controls ------------------------------------------------
    def add(self, key: str, backend: Dict[str, Any]) -> None:
        backend = dict(backend)
        backend["key"] = key
        self.backends[key] = backend
        self.emit("added", key, backend)

    def remove(self, key: str) -> None:
        self.backends.pop(key, None)
        self.emit("removed", key)

    def fail(self, err: BaseException) -> None:
        self._last_error = err
        self.state = "failed"
        self.emit("stateChanged", "failed")


class MockHttpServer:
    """Tiny asyncio HTTP/1.1 server for agent tests and benchmarks
    (stands in for the reference's local restify servers,
    test/agent.test.js:31-44).

    Routes: any path -> 200 JSON {"path":..., "count": N}; ``/ping`` ->
    200 "pong"; ``/err500`` -> 500; ``/close`` -> 200 + Connection:
    close.  ``broken=True`` accepts then immediately destroys
    connections.  Counts requests per connection to verify keep-alive
    reuse.
    """

    def __init__(self, broken: bool = False, tls_context=None) -> None:
        self.broken = broken
        self.tls_context = tls_context
        self.port: Optional[int] = None
        self.request_count = 0
        self.ping_count = 0
        self.conn_count = 0
        self.requests_per_conn: List[int] = []
        self._server: Optional[asyncio.AbstractServer] = None

    async def start(self, port: int = 0) -> int:
        self._server = await asyncio.start_server(
            self._handle, "127.0.0.1", port, ssl=self.tls_context)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            self._server = None

    async def _handle(self, reader: asyncio.StreamReader,
                      writer: asyncio.StreamWriter) -> None:
        self.conn_count += 1
        my_count = 0
        self.requests_per_conn.append(0)
        slot = len(self.requests_per_conn) - 1
        try:
            if self.broken:
                writer.close()
                return
            while True:
                line = await reader.readline()
                if not line or line == b"\r\n":
                    if not line:
                        break
                    continue
                parts = line.decode("latin-1").split()
                if len(parts) < 3:
                    break
                method, path = parts[0], parts[1]
                headers = {}
                while True:
                    hline = await reader.readline()
                    if hline in (b"\r\n", b"", b"\n"):
                        break
                    name, _, value = hline.decode("latin-1").partition(":")
                    headers[name.strip().lower()] = value.strip()
                req_body = b""
                if "chunked" in headers.get("transfer-encoding", "").lower():
                    while True:
                        szline = await reader.readline()
                        size = int(szline.split(b";")[0].strip() or b"0", 16)
                        if size == 0:
                            await reader.readline()  # trailing CRLF
                            break
                        req_body += await reader.readexactly(size)
                        await reader.readexactly(2)  # CRLF
                else:
                    clen = int(headers.get("content-length", "0"))
                    if clen:
                        req_body = await reader.readexactly(clen)

                self.request_count += 1
                my_count += 1
                self.requests_per_conn[slot] = my_count

                close = headers.get("connection", "").lower() == "close"
                if path == "/echo":
                    body = req_body
                    status = b"200 OK"
                elif path == "/ping":
                    self.ping_count += 1
                    body = b"pong"
                    status = b"200 OK"
                elif path == "/err500":
                    body = b"boom"
                    status = b"500 Internal Server Error"
                elif path == "/close":
                    body = b"bye"
                    status = b"200 OK"
                    close = True
                else:
                    body = json.dumps({"path": path,
                                       "count": my_count}).encode()
                    status = b"200 OK"
                conn_hdr = b"close" if close else b"keep-alive"
                writer.write(b"HTTP/1.1 " + status + b"\r\n"
                             b"Content-Length: " +
                             str(len(body)).encode() + b"\r\n"
                             b"Connection: " + conn_hdr + b"\r\n\r\n" + body)
                await writer.drain()
                if close:
                    break
        except (ConnectionResetError, asyncio.IncompleteReadError,
                BrokenPipeError):
            pass
        finally:
            try:
                writer.close()
            except Exception:
                pass


class MockDnsServer:
    """A real UDP DNS server on 127.0.0.1 serving a scripted zone.

    Used by the dns_client integration tests and the DNS-SRV benchmark
    config (BASELINE.json config #2).  The zone maps (name, type) to a
    list of record dicts in dns_wire format, e.g.::

        srv.add_srv("_http._tcp.svc", "b1.svc", 8080, ttl=30)
        srv.add_a("b1.svc", "127.0.0.1", ttl=30)
    """

    def __init__(self) -> None:
        self.zone: Dict[Tuple[str, str], List[Dict[str, Any]]] = {}
        self.queries: List[Tuple[str, str]] = []
        self.tcp_queries: List[Tuple[str, str]] = []
        self.port: Optional[int] = None
        self.drop_next = 0           # drop this many queries (timeouts)
        self.rcode_override: Optional[str] = None
        self.truncate_udp = False    # answer UDP with TC=1 (force TCP)
        self._transport = None
        self._tcp_server: Optional[asyncio.AbstractServer] = None

    # -- zone building --------------------------------------------------
    def add_srv(self, name: str, target: str, port: int,
                ttl: int = 60) -> None:
        self.zone.setdefault((name.lower(), "SRV"), []).append({
            "type": "SRV", "name": name, "ttl": ttl, "priority": 0,
            "weight": 10, "port": port, "target": target})

    def add_a(self, name: str, address: str, ttl: int = 60) -> None:
        self.zone.setdefault((name.lower(), "A"), []).append({
            "type": "A", "name": name, "ttl": ttl, "target": address})

    def add_aaaa(self, name: str, address: str, ttl: int = 60) -> None:
        self.zone.setdefault((name.lower(), "AAAA"), []).append({
            "type": "AAAA", "name": name, "ttl": ttl, "target": address})

    def clear(self) -> None:
        self.zone.clear()

    # -- lifecycle -------------------------------------------------------
    async def start(self, port: int = 0) -> int:
        loop = asyncio.get_running_loop()
        server = self

        class Proto(asyncio.DatagramProtocol):
            def connection_made(self, transport):
                self.transport = transport

            def datagram_received(self, data, addr):
                resp = server._handle(data)
                if resp is not None:
                    self.transport.sendto(resp, addr)

        # DNS needs the same port on UDP and TCP; with port=0 the OS
        # picks a free UDP port whose TCP twin may be taken — retry
        last_err: Optional[BaseException] = None
        for _ in range(20):
            self._transport, _ = await loop.create_datagram_endpoint(
                Proto, local_addr=("127.0.0.1", port))
            self.port = self._transport.get_extra_info("sockname")[1]
            try:
                self._tcp_server = await asyncio.start_server(
                    self._handle_tcp, "127.0.0.1", self.port)
                return self.port
            except OSError as e:
                last_err = e
                self._transport.close()
                self._transport = None
                if port != 0:
                    break
        raise last_err  # type: ignore[misc]

    async def _handle_tcp(self, reader: asyncio.StreamReader,
                          writer: asyncio.StreamWriter) -> None:
        try:
            hdr = await reader.readexactly(2)
            ln = int.from_bytes(hdr, "big")
            data = await reader.readexactly(ln)
            resp = self._handle(data, via_tcp=True)
            if resp is not None:
                writer.write(len(resp).to_bytes(2, "big") + resp)
                await writer.drain()
        except (asyncio.IncompleteReadError, ConnectionResetError):
            pass
        finally:
            writer.close()

    def stop(self) -> None:
        if self._transport is not None:
            self._transport.close()
            self._transport = None
        if self._tcp_server is not None:
            self._tcp_server.close()
            self._tcp_server = None

    @property
    def resolver_address(self) -> str:
        return "127.0.0.1@%d" % self.port

    # -- request handling -------------------------------------------------
    def _handle(self, data: bytes, via_tcp: bool = False) -> Optional[bytes]:
        try:
            q = dns_wire.decode_message(data)
        except ValueError:
            return None
        if not q.question:
            return None
        if self.drop_next > 0:
            self.drop_next -= 1
            return None
        question = q.question[0]
        name, rtype = question["name"], question["type"]
        (self.tcp_queries if via_tcp else self.queries).append((name, rtype))
        if self.truncate_udp and not via_tcp:
            return dns_wire.encode_response(q.id, question, tc=True)
        if self.rcode_override is not None:
            return dns_wire.encode_response(q.id, question,
                                            rcode=self.rcode_override)
        answers = self.zone.get((name.lower(), rtype))
        if answers is None:
            if any(k[0] == name.lower() for k in self.zone):
                # NODATA: name exists, no records of this type
                return dns_wire.encode_response(q.id, question, authority=[{
                    "type": "SOA", "name": name, "ttl": 60}])
            return dns_wire.encode_response(q.id, question, rcode="NXDOMAIN")
        return dns_wire.encode_response(q.id, question, answers=answers)
