"""Parser fuzzing: HTTP/1.1 response parser and DNS wire codec.

Properties:
- the HTTP parser's result is invariant under arbitrary re-chunking of
  the byte stream;
- arbitrary garbage raises HttpParseError (never hangs or crashes);
- DNS messages round-trip encode->decode for arbitrary record sets, and
  the decoder never crashes on mutated/truncated packets.
"""

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from cueball_amd import dns_wire
from cueball_amd.http_client import HttpParseError, _ResponseParser

SETTINGS = dict(max_examples=300, deadline=None,
                suppress_health_check=[HealthCheck.too_slow])


def parse_all_at_once(data, head=False):
    p = _ResponseParser(head_request=head)
    p.feed(data)
    return p


@settings(**SETTINGS)
@given(
    body=st.binary(max_size=2000),
    chunks=st.lists(st.integers(min_value=1, max_value=64), max_size=60),
    chunked_te=st.booleans(),
    extra_headers=st.lists(
        st.tuples(
            st.text(alphabet="abcdefghij-", min_size=1, max_size=10),
            st.text(alphabet="klmnopqrst 0123456789", max_size=12)),
        max_size=5),
)
def test_http_parser_rechunking_invariance(body, chunks, chunked_te,
                                           extra_headers):
    hdrs = "".join("X-%s: %s\r\n" % (n, v) for n, v in extra_headers)
    if chunked_te:
        # split body into HTTP chunks of <=97 bytes
        parts = [body[i:i + 97] for i in range(0, len(body), 97)]
        payload = b"".join(b"%x\r\n%s\r\n" % (len(pt), pt)
                           for pt in parts) + b"0\r\n\r\n"
        wire = (("HTTP/1.1 200 OK\r\n%sTransfer-Encoding: chunked\r\n\r\n"
                 % hdrs).encode("latin-1") + payload)
    else:
        wire = (("HTTP/1.1 200 OK\r\n%sContent-Length: %d\r\n\r\n"
                 % (hdrs, len(body))).encode("latin-1") + body)

    ref = parse_all_at_once(wire)
    assert ref.state == ref.ST_DONE
    assert ref.response.body == body

    # feed the same bytes in arbitrary fragments
    p = _ResponseParser()
    pos = 0
    for c in chunks:
        if pos >= len(wire):
            break
        p.feed(wire[pos:pos + c])
        pos += c
    p.feed(wire[pos:])
    assert p.state == p.ST_DONE
    assert p.response.body == ref.response.body
    assert p.response.headers == ref.response.headers
    assert p.response.status_code == 200


@settings(**SETTINGS)
@given(data=st.binary(max_size=400))
def test_http_parser_never_hangs_on_garbage(data):
    p = _ResponseParser()
    try:
        p.feed(data)
        p.eof()
    except HttpParseError:
        pass  # rejecting garbage loudly is the contract
    # no crash, no unbounded state


RR_STRAT = st.one_of(
    st.builds(lambda n, t: {"type": "A", "name": n, "ttl": t,
                            "target": "10.1.2.3"},
              st.sampled_from(["a.b", "x.y.z", "host-1.example.com"]),
              st.integers(min_value=0, max_value=2**31 - 1)),
    st.builds(lambda n, t: {"type": "AAAA", "name": n, "ttl": t,
                            "target": "fe80::42"},
              st.sampled_from(["a.b", "v6.example"]),
              st.integers(min_value=0, max_value=3600)),
    st.builds(lambda n, p, t, tgt: {"type": "SRV", "name": n, "ttl": t,
                                    "priority": 1, "weight": 2,
                                    "port": p, "target": tgt},
              st.sampled_from(["_s._tcp.a.b"]),
              st.integers(min_value=0, max_value=65535),
              st.integers(min_value=0, max_value=3600),
              st.sampled_from(["b1.a.b", "b2.a.b"])),
)


@settings(**SETTINGS)
@given(answers=st.lists(RR_STRAT, max_size=6),
       additionals=st.lists(RR_STRAT, max_size=3),
       qid=st.integers(min_value=0, max_value=65535),
       rcode=st.sampled_from(["NOERROR", "NXDOMAIN", "SERVFAIL",
                              "REFUSED", "NOTIMP"]))
def test_dns_wire_roundtrip(answers, additionals, qid, rcode):
    wire = dns_wire.encode_response(
        qid, {"name": "q.example", "type": "SRV"}, rcode=rcode,
        answers=answers, additionals=additionals)
    msg = dns_wire.decode_message(wire)
    assert msg.id == qid
    assert msg.rcode_name == rcode
    assert len(msg.answers) == len(answers)
    for sent, got in zip(answers, msg.answers):
        assert got["type"] == sent["type"]
        assert got["name"] == sent["name"]
        assert got["ttl"] == sent["ttl"]
        if sent["type"] == "SRV":
            assert got["port"] == sent["port"]
            assert got["target"] == sent["target"]
        else:
            # address normalization (fe80::42 stays canonical here)
            assert got["target"] == sent["target"]


@settings(**SETTINGS)
@given(data=st.binary(max_size=200))
def test_dns_decoder_never_crashes(data):
    try:
        dns_wire.decode_message(data)
    except ValueError:
        pass


@settings(**SETTINGS)
@given(cut=st.integers(min_value=0, max_value=100),
       flip=st.integers(min_value=0, max_value=10**6))
def test_dns_decoder_survives_mutations(cut, flip):
    wire = dns_wire.encode_response(
        7, {"name": "q.example", "type": "SRV"},
        answers=[{"type": "SRV", "name": "q.example", "ttl": 9,
                  "priority": 0, "weight": 0, "port": 80,
                  "target": "b.example"}],
        additionals=[{"type": "A", "name": "b.example", "ttl": 9,
                      "target": "10.0.0.1"}])
    mutated = bytearray(wire[:max(0, len(wire) - cut)])
    if mutated:
        mutated[flip % len(mutated)] ^= 0xFF
    try:
        dns_wire.decode_message(bytes(mutated))
    except ValueError:
        pass
