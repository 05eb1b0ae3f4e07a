"""TCP/TLS Connection implementations of the cueball Connection
interface (docs/api.adoc:580-645): an EventEmitter that emits
``connect``, ``error``, ``close`` (plus ``connectError``/``timeout``)
and implements ``destroy()``; optional ``ref()``/``unref()``/
``set_unwanted()``.

The reference leaves socket construction to the user (node's
``net.connect``); here we ship batteries: ``TcpConnection`` rides an
asyncio transport and adds node-socket-flavored ``write``/``data``/
``end`` so the HTTP agent can be built on it.
"""

from __future__ import annotations

import asyncio
import ssl as mod_ssl
from typing import Any, Dict, Optional

from .events import EventEmitter
from .fsm import get_loop

__all__ = ["TcpConnection", "tcp_constructor"]


class _Protocol(asyncio.Protocol):
    def __init__(self, conn: "TcpConnection") -> None:
        self.conn = conn

    def connection_made(self, transport: asyncio.BaseTransport) -> None:
        self.conn._on_connected(transport)

    def data_received(self, data: bytes) -> None:
        self.conn.emit("data", data)

    def eof_received(self) -> Optional[bool]:
        self.conn.emit("end")
        return False  # close the transport

    def connection_lost(self, exc: Optional[Exception]) -> None:
        self.conn._on_lost(exc)


class TcpConnection(EventEmitter):
    """One TCP (or TLS) connection to a backend.

    Lifecycle: constructing starts the connect; 'connect' fires when
    established, 'connectError'/'error' on failure, 'close' exactly once
    when the transport is gone.
    """

    def __init__(self, backend: Dict[str, Any],
                 loop: Optional[asyncio.AbstractEventLoop] = None,
                 tls: bool = False,
                 ssl_context: Optional[mod_ssl.SSLContext] = None,
                 connect_timeout: Optional[float] = None,
                 server_hostname: Optional[str] = None,
                 nodelay: bool = True) -> None:
        super().__init__()
        self.backend = backend
        self._loop = get_loop(loop)
        self.connected = False
        self.dead = False
        self.unwanted = False
        self._transport: Optional[asyncio.Transport] = None
        self._closed_emitted = False
        self._connect_task: Optional[asyncio.Task] = None
        self._tls = tls
        self._ssl_context = ssl_context
        self._server_hostname = server_hostname
        self._nodelay = nodelay
        self._connect_timeout = connect_timeout
        self.local_port: Optional[int] = None
        self._start_connect()

    # -- connection establishment ---------------------------------------
    def _start_connect(self) -> None:
        async def do_connect() -> None:
            sslctx: Any = None
            if self._tls:
                sslctx = self._ssl_context
                if sslctx is None:
                    sslctx = mod_ssl.create_default_context()
            try:
                coro = self._loop.create_connection(
                    lambda: _Protocol(self),
                    self.backend["address"], self.backend["port"],
                    ssl=sslctx,
                    server_hostname=(self._server_hostname
                                     if sslctx is not None else None))
                if self._connect_timeout is not None:
                    await asyncio.wait_for(
                        coro, timeout=self._connect_timeout / 1000.0)
                else:
                    await coro
            except asyncio.CancelledError:
                raise
            except Exception as e:  # noqa: BLE001 - report via event
                if not self.dead:
                    if isinstance(e, asyncio.TimeoutError):
                        self.emit("connectTimeout")
                    else:
                        self.emit("connectError", e)
                    self._emit_close()

        self._connect_task = self._loop.create_task(do_connect())

    def _on_connected(self, transport: asyncio.BaseTransport) -> None:
        if self.dead:
            transport.close()
            return
        self._transport = transport  # type: ignore[assignment]
        sock = transport.get_extra_info("socket")
        if sock is not None:
            try:
                if self._nodelay and not self._tls:
                    import socket as mod_socket
                    sock.setsockopt(mod_socket.IPPROTO_TCP,
                                    mod_socket.TCP_NODELAY, 1)
                self.local_port = sock.getsockname()[1]
            except OSError:
                pass
        self.connected = True
        self.emit("connect")

    def _on_lost(self, exc: Optional[Exception]) -> None:
        self.connected = False
        self._transport = None
        if exc is not None and not self.dead:
            self.emit("error", exc)
        self._emit_close()

    def _emit_close(self) -> None:
        if self._closed_emitted:
            return
        self._closed_emitted = True
        self.emit("close")

    # -- cueball Connection interface -------------------------------------
    def destroy(self) -> None:
        if self.dead:
            return
        self.dead = True
        self.connected = False
        if self._connect_task is not None and not self._connect_task.done():
            self._connect_task.cancel()
        if self._transport is not None:
            self._transport.close()
            self._transport = None
        else:
            self._emit_close()

    def set_unwanted(self) -> None:
        self.unwanted = True

    def ref(self) -> None:
        pass

    def unref(self) -> None:
        pass

    # -- socket-flavored IO (used by the HTTP agent) -----------------------
    def write(self, data: bytes) -> bool:
        if self._transport is None:
            raise ConnectionResetError("write on closed TcpConnection")
        self._transport.write(data)
        return True

    def end(self) -> None:
        if self._transport is not None:
            try:
                self._transport.write_eof()
            except (OSError, RuntimeError):
                self._transport.close()

    def pause(self) -> None:
        if self._transport is not None:
            self._transport.pause_reading()

    def resume(self) -> None:
        if self._transport is not None:
            self._transport.resume_reading()


def tcp_constructor(loop: Optional[asyncio.AbstractEventLoop] = None,
                    tls: bool = False,
                    ssl_context: Optional[mod_ssl.SSLContext] = None,
                    connect_timeout: Optional[float] = None,
                    server_hostname: Optional[str] = None):
    """Build a pool/set ``constructor`` option that opens TcpConnections."""

    def constructor(backend: Dict[str, Any]) -> TcpConnection:
        return TcpConnection(
            backend, loop=loop, tls=tls, ssl_context=ssl_context,
            connect_timeout=connect_timeout,
            server_hostname=server_hostname or backend.get("name"))

    return constructor
