"""DNS query engine: UDP (+TCP fallback) with multi-server failover.

Rebuild of the load-bearing parts of the ``mname-client`` npm dependency
(survey §2.2): a shared client with a concurrency cap, per-resolver
failover, an error threshold, rcode errors carrying ``.code``, and
MultiError aggregation for the resolver's rcode voting
(lib/resolver.js:385-392, :1230-1259).

The engine is asyncio-native; ``lookup(opts, cb, loop=...)`` is the
callback adapter the resolver FSM uses.  Tests stub the client class the
same way the reference stubs DnsClient with sinon
(test/dns.test.js:75-110).
"""

from __future__ import annotations

import asyncio
import random
import weakref
from typing import Any, Callable, Dict, List, Optional

from . import dns_wire

__all__ = ["DnsClient", "DnsError", "MultiError", "NoNameError",
           "NoRecordsError", "TimeoutError_"]


class DnsError(Exception):
    """A DNS-level failure; ``code`` is the rcode name (NXDOMAIN,
    SERVFAIL, REFUSED, NOTIMP...)."""

    def __init__(self, message: str, code: Optional[str] = None,
                 cause: Optional[BaseException] = None) -> None:
        super().__init__(message)
        self.code = code
        self.cause = cause
        if cause is not None:
            self.__cause__ = cause


class TimeoutError_(DnsError):
    def __init__(self, name: str, server: Optional[str] = None) -> None:
        super().__init__("Timeout while contacting resolvers for name %s%s"
                         % (name, " (server %s)" % server if server else ""))
        self.dns_name = name


class MultiError(DnsError):
    """Aggregate of per-resolver failures; the resolver votes on the
    most common ``.code`` among them."""

    def __init__(self, errors: List[BaseException]) -> None:
        super().__init__("%d errors from DNS resolvers: %s" % (
            len(errors), "; ".join(str(e) for e in errors[:4])))
        self._errors = errors

    def errors(self) -> List[BaseException]:
        return list(self._errors)


class NoNameError(DnsError):
    """NXDOMAIN: the name does not exist at all
    (lib/resolver.js:1173-1184)."""

    def __init__(self, name: str, cause: Optional[BaseException] = None) \
            -> None:
        super().__init__("No records returned for name %s" % name,
                         code="NXDOMAIN", cause=cause)
        self.dns_name = name


class NoRecordsError(DnsError):
    """NODATA: the name exists but has no records of this type; may
    carry an SOA-derived TTL (lib/resolver.js:1186-1199)."""

    def __init__(self, name: str, rtype: str,
                 ttl: Optional[float] = None) -> None:
        super().__init__("No records returned for name %s of type %s"
                         % (name, rtype))
        self.dns_name = name
        self.dns_type = rtype
        self.ttl = ttl


class _UdpProtocol(asyncio.DatagramProtocol):
    """Delivers datagrams to a replaceable future: the client re-arms
    ``fut`` after ignoring spoofed/garbage datagrams."""

    def __init__(self, fut: asyncio.Future) -> None:
        self.fut = fut

    def datagram_received(self, data: bytes, addr: Any) -> None:
        if not self.fut.done():
            self.fut.set_result(data)

    def error_received(self, exc: Exception) -> None:
        if not self.fut.done():
            self.fut.set_exception(exc)


class DnsClient:
    """Shared DNS client; one per concurrency level, cached by the
    resolver (lib/resolver.js:385-392, :411-413)."""

    def __init__(self, concurrency: int = 3) -> None:
        self.concurrency = concurrency
        # Keyed by the loop object itself through a WeakKeyDictionary so
        # that (a) destroyed loops release their semaphore and (b) id()
        # reuse can never hand a dead loop's semaphore to a new loop.
        self._sems: ("weakref.WeakKeyDictionary"
                     "[asyncio.AbstractEventLoop, asyncio.Semaphore]") = (
            weakref.WeakKeyDictionary())

    def _sem(self, loop: asyncio.AbstractEventLoop) -> asyncio.Semaphore:
        sem = self._sems.get(loop)
        if sem is None:
            sem = asyncio.Semaphore(self.concurrency)
            self._sems[loop] = sem
        return sem

    # -- callback adapter used by the resolver FSM ----------------------
    def lookup(self, opts: Dict[str, Any],
               cb: Callable[[Optional[BaseException], Any], None],
               loop: Optional[asyncio.AbstractEventLoop] = None) -> None:
        loop = loop or asyncio.get_event_loop()

        async def run() -> None:
            try:
                msg = await self.lookup_async(opts, loop=loop)
            except Exception as e:  # noqa: BLE001 - deliver to callback
                cb(e, None)
                return
            cb(None, msg)

        loop.create_task(run())

    async def lookup_async(self, opts: Dict[str, Any],
                           loop: Optional[asyncio.AbstractEventLoop] = None):
        loop = loop or asyncio.get_running_loop()
        domain: str = opts["domain"]
        rtype: str = opts["type"]
        timeout_ms: float = opts.get("timeout") or 1000.0
        resolvers: List[str] = list(opts.get("resolvers") or [])
        if not resolvers:
            raise DnsError("no resolvers configured for lookup of %s"
                           % domain)
        threshold: int = opts.get("errorThreshold") or len(resolvers)

        errors: List[BaseException] = []
        async with _sem_ctx(self._sem(loop)):
            for server in resolvers:
                try:
                    msg = await self._query_one(loop, server, domain, rtype,
                                                timeout_ms)
                except Exception as e:  # noqa: BLE001 - collect, fail over
                    errors.append(e)
                    if len(errors) >= threshold:
                        break
                    continue
                return msg
        if len(errors) == 1:
            raise errors[0]
        raise MultiError(errors)

    async def _query_one(self, loop: asyncio.AbstractEventLoop, server: str,
                         domain: str, rtype: str, timeout_ms: float):
        qid = random.randrange(0, 65536)
        query = dns_wire.encode_query(qid, domain, rtype)
        port = 53
        if "@" in server:  # "host@port" form used by tests/mocks
            server, port_s = server.split("@", 1)
            port = int(port_s)

        deadline = loop.time() + timeout_ms / 1000.0
        msg = await self._udp_round(loop, server, port, query, qid, deadline,
                                    domain)
        if msg.truncated:
            msg = await self._tcp_round(loop, server, port, query, qid,
                                        deadline, domain)
        code = msg.rcode_name
        if code != "NOERROR":
            raise DnsError("DNS server %s returned %s for %s %s"
                           % (server, code, rtype, domain), code=code)
        return msg

    async def _udp_round(self, loop, server: str, port: int, query: bytes,
                         qid: int, deadline: float, domain: str):
        fut: asyncio.Future = loop.create_future()
        proto = _UdpProtocol(fut)
        transport, _ = await loop.create_datagram_endpoint(
            lambda: proto, remote_addr=(server, port), family=0)
        try:
            transport.sendto(query)
            resends = 2
            while True:
                left = deadline - loop.time()
                if left <= 0:
                    raise TimeoutError_(domain, server)
                try:
                    data = await asyncio.wait_for(fut, timeout=left)
                except asyncio.TimeoutError:
                    raise TimeoutError_(domain, server) from None
                except OSError:
                    # connected-UDP sockets surface async ICMP errors
                    # (port unreachable etc.); these can be stale
                    # artifacts of a reused ephemeral port — resend a
                    # bounded number of times before giving up
                    if resends <= 0:
                        raise
                    resends -= 1
                    await asyncio.sleep(min(0.05, max(0.0, left)))
                    fut = loop.create_future()
                    proto.fut = fut
                    transport.sendto(query)
                    continue
                try:
                    msg = dns_wire.decode_message(data)
                except ValueError:
                    msg = None  # garbage datagram: ignore, keep waiting
                if msg is None or msg.id != qid:
                    # spoofed/stale/garbage: re-arm and keep waiting
                    fut = loop.create_future()
                    proto.fut = fut
                    continue
                return msg
        finally:
            transport.close()

    async def _tcp_round(self, loop, server: str, port: int, query: bytes,
                         qid: int, deadline: float, domain: str):
        left = deadline - loop.time()
        if left <= 0:
            raise TimeoutError_(domain, server)
        try:
            reader, writer = await asyncio.wait_for(
                asyncio.open_connection(server, port), timeout=left)
        except asyncio.TimeoutError:
            raise TimeoutError_(domain, server) from None
        try:
            writer.write(len(query).to_bytes(2, "big") + query)
            await writer.drain()
            left = deadline - loop.time()
            if left <= 0:
                raise TimeoutError_(domain, server)
            try:
                hdr = await asyncio.wait_for(reader.readexactly(2),
                                             timeout=left)
                ln = int.from_bytes(hdr, "big")
                data = await asyncio.wait_for(
                    reader.readexactly(ln),
                    timeout=max(0.001, deadline - loop.time()))
            except asyncio.TimeoutError:
                raise TimeoutError_(domain, server) from None
            msg = dns_wire.decode_message(data)
            # Same spoof/desync check as the UDP path: the response on the
            # truncation-fallback TCP stream must echo our query id.
            if msg.id != qid:
                raise DnsError(
                    "DNS server %s returned mismatched query id %d "
                    "(expected %d) for %s" % (server, msg.id, qid, domain))
            return msg
        finally:
            writer.close()


class _sem_ctx:
    def __init__(self, sem: asyncio.Semaphore) -> None:
        self.sem = sem

    async def __aenter__(self):
        await self.sem.acquire()
        return self

    async def __aexit__(self, *exc):
        self.sem.release()
        return False
