"""Minimal HTTP/1.1 client machinery for the cueball agent.

The reference plugs into node's own http stack by duck-typing
``http.Agent`` (lib/agent.js); Python has no equivalent pluggable agent
in the stdlib, so the rebuild ships the thin HTTP/1.1 client the agent
needs: request serialization, response parsing (status line, headers,
content-length / chunked / read-to-close framing), and the keep-alive
"free" protocol the agent relies on — after a response completes on a
reusable connection, the *connection* emits ``free`` so the agent can
release the claim (mirrors lib/agent.js:322-383).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from .events import EventEmitter

__all__ = ["HttpRequest", "HttpResponse", "HttpParseError"]


class HttpParseError(Exception):
    pass


class HttpResponse(EventEmitter):
    """Parsed response head + streamed body ('data'/'end' events)."""

    def __init__(self) -> None:
        super().__init__()
        self.status_code = 0
        self.reason = ""
        self.http_version = "1.1"
        self.headers: Dict[str, str] = {}
        self.complete = False
        self._body_chunks: List[bytes] = []
        self._collect = True

    def get_header(self, name: str) -> Optional[str]:
        return self.headers.get(name.lower())

    @property
    def body(self) -> bytes:
        return b"".join(self._body_chunks)

    def _push(self, data: bytes) -> None:
        if self._collect:
            self._body_chunks.append(data)
        self.emit("data", data)

    def _finish(self) -> None:
        self.complete = True
        self.emit("end")


class _ResponseParser:
    """Incremental HTTP/1.1 response parser."""

    ST_STATUS = 0
    ST_HEADERS = 1
    ST_BODY = 2
    ST_CHUNK_SIZE = 3
    ST_CHUNK_DATA = 4
    ST_CHUNK_TRAILER = 5
    ST_DONE = 6

    def __init__(self, head_request: bool = False) -> None:
        self.buf = b""
        self.state = self.ST_STATUS
        self.response = HttpResponse()
        self.head_request = head_request
        self.remaining = 0
        self.read_to_close = False

    def feed(self, data: bytes) -> None:
        self.buf += data
        while True:
            if self.state == self.ST_STATUS:
                line = self._take_line()
                if line is None:
                    return
                self._parse_status(line)
            elif self.state == self.ST_HEADERS:
                line = self._take_line()
                if line is None:
                    return
                if line == b"":
                    self._headers_done()
                    if self.state == self.ST_DONE:
                        return
                else:
                    self._parse_header(line)
            elif self.state == self.ST_BODY:
                if self.read_to_close:
                    if self.buf:
                        chunk, self.buf = self.buf, b""
                        self.response._push(chunk)
                    return
                if self.remaining > 0:
                    take = min(self.remaining, len(self.buf))
                    if take == 0:
                        return
                    chunk = self.buf[:take]
                    self.buf = self.buf[take:]
                    self.remaining -= take
                    self.response._push(chunk)
                if self.remaining == 0:
                    self._done()
                    return
            elif self.state == self.ST_CHUNK_SIZE:
                line = self._take_line()
                if line is None:
                    return
                try:
                    size = int(line.split(b";")[0].strip(), 16)
                except ValueError:
                    raise HttpParseError("bad chunk size %r" % line)
                if size == 0:
                    self.state = self.ST_CHUNK_TRAILER
                else:
                    self.remaining = size
                    self.state = self.ST_CHUNK_DATA
            elif self.state == self.ST_CHUNK_DATA:
                if len(self.buf) < self.remaining + 2:
                    return
                chunk = self.buf[:self.remaining]
                if self.buf[self.remaining:self.remaining + 2] != b"\r\n":
                    raise HttpParseError("missing chunk terminator")
                self.buf = self.buf[self.remaining + 2:]
                self.remaining = 0
                self.response._push(chunk)
                self.state = self.ST_CHUNK_SIZE
            elif self.state == self.ST_CHUNK_TRAILER:
                line = self._take_line()
                if line is None:
                    return
                if line == b"":
                    self._done()
                    return
            else:  # ST_DONE
                return

    def eof(self) -> None:
        """Transport closed: with read-to-close framing that's 'end'."""
        if self.state == self.ST_BODY and self.read_to_close:
            self._done()
        elif self.state != self.ST_DONE:
            raise HttpParseError("connection closed mid-response")

    # -- helpers --------------------------------------------------------
    def _take_line(self) -> Optional[bytes]:
        idx = self.buf.find(b"\r\n")
        if idx == -1:
            if len(self.buf) > 65536:
                raise HttpParseError("header line too long")
            return None
        line = self.buf[:idx]
        self.buf = self.buf[idx + 2:]
        return line

    def _parse_status(self, line: bytes) -> None:
        parts = line.split(b" ", 2)
        if len(parts) < 2 or not parts[0].startswith(b"HTTP/"):
            raise HttpParseError("bad status line %r" % line)
        self.response.http_version = parts[0][5:].decode("ascii", "replace")
        try:
            self.response.status_code = int(parts[1])
        except ValueError:
            raise HttpParseError("bad status code %r" % parts[1])
        self.response.reason = (parts[2].decode("latin-1")
                                if len(parts) > 2 else "")
        self.state = self.ST_HEADERS

    def _parse_header(self, line: bytes) -> None:
        idx = line.find(b":")
        if idx == -1:
            raise HttpParseError("bad header line %r" % line)
        name = line[:idx].strip().lower().decode("latin-1")
        value = line[idx + 1:].strip().decode("latin-1")
        if name in self.response.headers:
            self.response.headers[name] += ", " + value
        else:
            self.response.headers[name] = value

    def _headers_done(self) -> None:
        r = self.response
        code = r.status_code
        if self.head_request or code in (204, 304) or 100 <= code < 200:
            self._done()
            return
        te = r.headers.get("transfer-encoding", "")
        if "chunked" in te.lower():
            self.state = self.ST_CHUNK_SIZE
            return
        cl = r.headers.get("content-length")
        if cl is not None:
            try:
                self.remaining = int(cl)
            except ValueError:
                raise HttpParseError("bad content-length %r" % cl)
            if self.remaining == 0:
                self._done()
            else:
                self.state = self.ST_BODY
            return
        self.read_to_close = True
        self.state = self.ST_BODY

    def _done(self) -> None:
        self.state = self.ST_DONE
        self.response._finish()


class HttpRequest(EventEmitter):
    """One HTTP request bound to a cueball connection.

    Protocol with the agent (mirrors node ClientRequest/Agent):
    the agent claims a connection and calls ``req.on_socket(conn)``; the
    request writes itself, parses the response, emits ``response`` (and
    ``error``), and afterwards makes the connection emit ``free`` if it
    is reusable (keep-alive) — the agent's listener then releases the
    claim.  ``abort()`` emits 'abort' so the agent can cancel/close.
    """

    def __init__(self, method: str, path: str,
                 headers: Optional[Dict[str, str]] = None,
                 body: Optional[bytes] = None,
                 host: Optional[str] = None,
                 streaming: bool = False) -> None:
        super().__init__()
        self.method = method.upper()
        self.path = path
        self.headers = dict(headers or {})
        self.body = body
        self.host = host
        self.aborted = False
        self.conn: Any = None
        self._parser: Optional[_ResponseParser] = None
        self._response: Optional[HttpResponse] = None
        self._finished = False
        # streaming upload (node ClientRequest write()/end() parity):
        # body chunks written before the socket arrives are buffered
        self.streaming = streaming
        self._ended = not streaming
        self._pending_chunks: List[bytes] = []
        self._head_sent = False
        self._chunked_upload = False

    # -- user API -------------------------------------------------------
    def abort(self) -> None:
        if self.aborted or self._finished:
            return
        self.aborted = True
        self.emit("abort")

    def write(self, data: bytes) -> None:
        """Stream a body chunk (requires streaming=True).  Uses chunked
        transfer-encoding unless Content-Length was given up front."""
        if not self.streaming:
            raise RuntimeError("HttpRequest.write() requires streaming=True")
        if self._ended:
            raise RuntimeError("write() after end()")
        if self.conn is None or not self._head_sent:
            self._pending_chunks.append(data)
        else:
            self._send_chunk(data)

    def end(self, data: Optional[bytes] = None) -> None:
        """Finish a streaming request body."""
        if not self.streaming or self._ended:
            if data:
                raise RuntimeError("end(data) on a non-streaming request")
            self._ended = True
            return
        if data:
            self.write(data)
        self._ended = True
        if self.conn is not None and self._head_sent:
            self._send_trailer()

    def _send_chunk(self, data: bytes) -> None:
        if not data:
            return
        try:
            if self._chunked_upload:
                self.conn.write(b"%x\r\n" % len(data) + data + b"\r\n")
            else:
                self.conn.write(data)
        except (ConnectionResetError, OSError) as e:
            self._fail(e)

    def _send_trailer(self) -> None:
        if self._chunked_upload:
            try:
                self.conn.write(b"0\r\n\r\n")
            except (ConnectionResetError, OSError) as e:
                self._fail(e)

    # -- agent protocol ---------------------------------------------------
    def on_socket(self, conn: Any) -> None:
        """The agent hands us a connected cueball connection (or a
        FakeSocket that will emit 'error')."""
        self.conn = conn
        if getattr(conn, "_is_fake_socket", False):
            conn.on("error", lambda e: self._fail(e))
            return
        self._parser = _ResponseParser(
            head_request=(self.method == "HEAD"))

        conn.on("data", self._on_data)
        conn.on("close", self._on_close)
        conn.on("error", self._on_conn_error)

        try:
            conn.write(self._serialize())
            self._head_sent = True
            if self.streaming:
                for chunk in self._pending_chunks:
                    self._send_chunk(chunk)
                self._pending_chunks = []
                if self._ended:
                    self._send_trailer()
        except (ConnectionResetError, OSError) as e:
            self._fail(e)

    def _serialize(self) -> bytes:
        hdrs = {k.lower(): (k, v) for k, v in self.headers.items()}
        lines = ["%s %s HTTP/1.1" % (self.method, self.path)]
        if "host" not in hdrs:
            hdrs["host"] = ("Host", self.host or "localhost")
        if "connection" not in hdrs:
            hdrs["connection"] = ("Connection", "keep-alive")
        body = self.body or b""
        if self.streaming:
            if body:
                raise RuntimeError("streaming request cannot also have "
                                   "a fixed body")
            if "content-length" not in hdrs and \
                    "transfer-encoding" not in hdrs:
                hdrs["transfer-encoding"] = ("Transfer-Encoding", "chunked")
                self._chunked_upload = True
        elif body and "content-length" not in hdrs:
            hdrs["content-length"] = ("Content-Length", str(len(body)))
        for key, (name, value) in hdrs.items():
            lines.append("%s: %s" % (name, value))
        head = ("\r\n".join(lines) + "\r\n\r\n").encode("latin-1")
        return head + body

    # -- connection events -------------------------------------------------
    def _on_data(self, data: bytes) -> None:
        if self._finished:
            return
        parser = self._parser
        try:
            parser.feed(data)
        except HttpParseError as e:
            self._fail(e)
            return
        if parser.response.status_code and self._response is None and \
                parser.state >= _ResponseParser.ST_BODY:
            self._response = parser.response
            self.emit("response", parser.response)
        if parser.state == _ResponseParser.ST_DONE:
            if self._response is None:
                self._response = parser.response
                self.emit("response", parser.response)
            self._complete()

    def _on_close(self) -> None:
        if self._finished:
            return
        parser = self._parser
        try:
            if parser is not None:
                parser.eof()
        except HttpParseError as e:
            self._fail(e)
            return
        if parser is not None and parser.state == _ResponseParser.ST_DONE:
            if self._response is None:
                self._response = parser.response
                self.emit("response", parser.response)
            self._complete(reusable=False)
        else:
            self._fail(ConnectionResetError("connection closed before "
                                            "response"))

    def _on_conn_error(self, err: BaseException) -> None:
        self._fail(err)

    # -- completion ---------------------------------------------------------
    def _reusable(self) -> bool:
        r = self._response
        if r is None:
            return False
        if self._parser is not None and self._parser.read_to_close:
            return False
        conn_hdr = (r.headers.get("connection") or "").lower()
        if r.http_version == "1.0":
            return "keep-alive" in conn_hdr
        return "close" not in conn_hdr

    def _cleanup(self) -> None:
        conn = self.conn
        if conn is None:
            return
        conn.remove_listener("data", self._on_data)
        conn.remove_listener("close", self._on_close)
        conn.remove_listener("error", self._on_conn_error)

    def _complete(self, reusable: Optional[bool] = None) -> None:
        if self._finished:
            return
        self._finished = True
        self._cleanup()
        conn = self.conn
        if reusable is None:
            reusable = self._reusable()
        if reusable:
            # keep-alive: hand the connection back (agent releases it)
            conn.emit("free")
        else:
            # not reusable: nudge the transport shut; the agent's
            # 'close' handler releases the claim when it dies
            if getattr(conn, "connected", False):
                try:
                    conn.end()
                except (OSError, RuntimeError):
                    pass

    def _fail(self, err: BaseException) -> None:
        if self._finished:
            return
        self._finished = True
        self._cleanup()
        self.emit("error", err)


# exclude the request's own connection-error listener from the claim
# handle's leak accounting (see connection_fsm.count_listeners)
HttpRequest._on_conn_error._cueball_internal = True  # type: ignore[attr-defined]
