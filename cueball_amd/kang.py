"""Kang debugging endpoint: serve the pool monitor's snapshot over HTTP.

The reference exposes introspection through the kang npm module mounted
on a restify server (bin/cbresolve:250-268, test/monitor.test.js:120-143);
the rebuild ships the server itself.  ``GET /kang/snapshot`` returns::

    {"service": {"name": "cueball", ...},
     "types": ["pool", "set", "dns_res"],
     "pool": {uuid: {...}}, "set": {...}, "dns_res": {...}}

with the same per-object payloads as lib/pool-monitor.js:60-216.  A
``GET /metrics`` endpoint additionally exposes the metrics collector in
Prometheus text format.
"""

from __future__ import annotations

import asyncio
import json
from typing import Any, Dict, Optional

from . import metrics as mod_metrics
from .pool_monitor import PoolMonitor, monitor as global_monitor

__all__ = ["KangServer"]


class KangServer:
    def __init__(self, monitor: Optional[PoolMonitor] = None,
                 collector: Optional[mod_metrics.Collector] = None) -> None:
        self.monitor = monitor or global_monitor
        self.collector = collector
        self.port: Optional[int] = None
        self._server: Optional[asyncio.AbstractServer] = None

    async def start(self, port: int = 0, host: str = "127.0.0.1") -> int:
        self._server = await asyncio.start_server(self._handle, host, port)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            self._server = None

    # -- snapshot assembly ------------------------------------------------
    def snapshot(self) -> Dict[str, Any]:
        opts = self.monitor.to_kang_options()
        snap: Dict[str, Any] = {
            "service": {
                "name": opts["service_name"],
                "ident": opts["ident"],
                "version": opts["version"],
            },
            "stats": opts["stats"](),
            "types": opts["list_types"](),
        }
        for type_ in snap["types"]:
            objs: Dict[str, Any] = {}
            for id_ in opts["list_objects"](type_):
                try:
                    objs[id_] = opts["get"](type_, id_)
                except Exception as e:  # noqa: BLE001 - debug endpoint
                    objs[id_] = {"error": repr(e)}
            snap[type_] = objs
        return snap

    # -- request handling ---------------------------------------------------
    async def _handle(self, reader: asyncio.StreamReader,
                      writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                line = await reader.readline()
                if not line:
                    break
                parts = line.decode("latin-1").split()
                if len(parts) < 2:
                    break
                path = parts[1]
                close = False
                while True:
                    hline = await reader.readline()
                    if hline in (b"\r\n", b"", b"\n"):
                        break
                    if hline.lower().startswith(b"connection:") and \
                            b"close" in hline.lower():
                        close = True

                if path.startswith("/kang/snapshot"):
                    body = json.dumps(self.snapshot(),
                                      default=_json_default).encode()
                    ctype = b"application/json"
                    status = b"200 OK"
                elif path == "/metrics" and self.collector is not None:
                    body = self.collector.collect().encode()
                    ctype = b"text/plain; version=0.0.4"
                    status = b"200 OK"
                else:
                    body = b'{"error": "not found"}'
                    ctype = b"application/json"
                    status = b"404 Not Found"

                writer.write(b"HTTP/1.1 " + status + b"\r\n"
                             b"Content-Type: " + ctype + b"\r\n"
                             b"Content-Length: " +
                             str(len(body)).encode() + b"\r\n"
                             b"Connection: " +
                             (b"close" if close else b"keep-alive") +
                             b"\r\n\r\n" + body)
                await writer.drain()
                if close:
                    break
        except (ConnectionResetError, BrokenPipeError):
            pass
        finally:
            try:
                writer.close()
            except Exception:
                pass


def _json_default(obj: Any) -> Any:
    if isinstance(obj, BaseException):
        return repr(obj)
    return str(obj)
