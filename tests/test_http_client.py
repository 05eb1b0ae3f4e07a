"""HTTP/1.1 response parser + request serialization unit tests."""

import pytest

from cueball_amd.http_client import (HttpParseError, HttpRequest,
                                     _ResponseParser)


def feed_all(parser, data, chunk=7):
    for i in range(0, len(data), chunk):
        parser.feed(data[i:i + chunk])


def test_content_length_body():
    p = _ResponseParser()
    feed_all(p, b"HTTP/1.1 200 OK\r\nContent-Length: 5\r\n"
                b"X-Foo: bar\r\n\r\nhello")
    assert p.state == p.ST_DONE
    r = p.response
    assert r.status_code == 200
    assert r.reason == "OK"
    assert r.headers["x-foo"] == "bar"
    assert r.body == b"hello"
    assert r.complete


def test_chunked_body():
    p = _ResponseParser()
    feed_all(p, b"HTTP/1.1 200 OK\r\nTransfer-Encoding: chunked\r\n\r\n"
                b"5\r\nhello\r\n6\r\n world\r\n0\r\n\r\n")
    assert p.state == p.ST_DONE
    assert p.response.body == b"hello world"


def test_read_to_close_body():
    p = _ResponseParser()
    feed_all(p, b"HTTP/1.0 200 OK\r\n\r\npartial data")
    assert p.state == p.ST_BODY
    assert p.read_to_close
    p.eof()
    assert p.state == p.ST_DONE
    assert p.response.body == b"partial data"


def test_no_body_statuses():
    for code in (204, 304):
        p = _ResponseParser()
        feed_all(p, b"HTTP/1.1 %d X\r\nContent-Length: 10\r\n\r\n"
                 % code)
        assert p.state == p.ST_DONE
        assert p.response.body == b""


def test_head_request_no_body():
    p = _ResponseParser(head_request=True)
    feed_all(p, b"HTTP/1.1 200 OK\r\nContent-Length: 100\r\n\r\n")
    assert p.state == p.ST_DONE


def test_eof_mid_response_raises():
    p = _ResponseParser()
    feed_all(p, b"HTTP/1.1 200 OK\r\nContent-Length: 10\r\n\r\nabc")
    with pytest.raises(HttpParseError):
        p.eof()


def test_bad_status_line():
    p = _ResponseParser()
    with pytest.raises(HttpParseError):
        p.feed(b"NONSENSE\r\n\r\n")


def test_duplicate_headers_joined():
    p = _ResponseParser()
    feed_all(p, b"HTTP/1.1 200 OK\r\nSet-Thing: a\r\nSet-Thing: b\r\n"
                b"Content-Length: 0\r\n\r\n")
    assert p.response.headers["set-thing"] == "a, b"


def test_request_serialization():
    req = HttpRequest("get", "/x/y?q=1", headers={"X-Custom": "v"},
                      body=b"data", host="example.com")
    wire = req._serialize()
    head, _, body = wire.partition(b"\r\n\r\n")
    assert body == b"data"
    lines = head.split(b"\r\n")
    assert lines[0] == b"GET /x/y?q=1 HTTP/1.1"
    joined = b"\n".join(lines).lower()
    assert b"host: example.com" in joined
    assert b"connection: keep-alive" in joined
    assert b"content-length: 4" in joined
    assert b"x-custom: v" in joined


def test_keep_alive_decision():
    req = HttpRequest("GET", "/", host="h")
    p = _ResponseParser()
    feed_all(p, b"HTTP/1.1 200 OK\r\nContent-Length: 0\r\n\r\n")
    req._parser = p
    req._response = p.response
    assert req._reusable() is True

    p2 = _ResponseParser()
    feed_all(p2, b"HTTP/1.1 200 OK\r\nConnection: close\r\n"
                 b"Content-Length: 0\r\n\r\n")
    req2 = HttpRequest("GET", "/", host="h")
    req2._parser = p2
    req2._response = p2.response
    assert req2._reusable() is False

    # HTTP/1.0 defaults to not reusable without keep-alive
    p3 = _ResponseParser()
    feed_all(p3, b"HTTP/1.0 200 OK\r\nContent-Length: 0\r\n\r\n")
    req3 = HttpRequest("GET", "/", host="h")
    req3._parser = p3
    req3._response = p3.response
    assert req3._reusable() is False
