"""DNS wire-format encoding/decoding (RFC 1035 + SRV RFC 2782 + EDNS0).

The reference outsources the DNS protocol to the ``mname-client`` npm
module (survey §2.2); this is our own implementation of the subset
cueball's resolver needs: query encoding and response decoding for
SRV / AAAA / A (plus CNAME/DNAME/SOA/OPT handling and name
compression), with TTLs, rcodes, and the additionals section.
"""

from __future__ import annotations

import socket
import struct
from typing import Any, Dict, List, Optional, Tuple

__all__ = [
    "encode_query", "decode_message", "DnsMessage",
    "RCODES", "QTYPES", "encode_response",
]

QTYPES = {"A": 1, "NS": 2, "CNAME": 5, "SOA": 6, "PTR": 12, "MX": 15,
          "TXT": 16, "AAAA": 28, "SRV": 33, "OPT": 41, "DNAME": 39}
TYPE_NAMES = {v: k for k, v in QTYPES.items()}

RCODES = {0: "NOERROR", 1: "FORMERR", 2: "SERVFAIL", 3: "NXDOMAIN",
          4: "NOTIMP", 5: "REFUSED"}
RCODE_NUMS = {v: k for k, v in RCODES.items()}

QCLASS_IN = 1

FLAG_QR = 0x8000
FLAG_AA = 0x0400
FLAG_TC = 0x0200
FLAG_RD = 0x0100
FLAG_RA = 0x0080


def _encode_name(name: str) -> bytes:
    out = b""
    name = name.rstrip(".")
    if name:
        for label in name.split("."):
            raw = label.encode("idna") if any(ord(c) > 127 for c in label) \
                else label.encode("ascii")
            if len(raw) > 63:
                raise ValueError("DNS label too long: %r" % label)
            out += struct.pack("B", len(raw)) + raw
    return out + b"\x00"


def encode_query(qid: int, name: str, qtype: str, edns: bool = True,
                 rd: bool = True) -> bytes:
    """Encode one question; EDNS0 OPT advertises a 1400-byte UDP payload."""
    flags = FLAG_RD if rd else 0
    arcount = 1 if edns else 0
    hdr = struct.pack(">HHHHHH", qid, flags, 1, 0, 0, arcount)
    q = _encode_name(name) + struct.pack(">HH", QTYPES[qtype], QCLASS_IN)
    msg = hdr + q
    if edns:
        # OPT pseudo-RR: root name, type 41, class = udp payload size
        msg += b"\x00" + struct.pack(">HHIH", 41, 1400, 0, 0)
    return msg


def _decode_name(buf: bytes, off: int, depth: int = 0) -> Tuple[str, int]:
    if depth > 16:
        raise ValueError("DNS name compression loop")
    labels: List[str] = []
    while True:
        if off >= len(buf):
            raise ValueError("truncated DNS name")
        ln = buf[off]
        if ln == 0:
            off += 1
            break
        if (ln & 0xC0) == 0xC0:
            ptr = struct.unpack(">H", buf[off:off + 2])[0] & 0x3FFF
            off += 2
            tail, _ = _decode_name(buf, ptr, depth + 1)
            if tail:
                labels.append(tail)
            break
        off += 1
        labels.append(buf[off:off + ln].decode("ascii", "replace"))
        off += ln
    return ".".join(labels), off


class DnsMessage:
    """Decoded DNS message with the accessor methods the resolver uses
    (get_answers/get_authority/get_additionals; mname-client duck type).
    """

    def __init__(self) -> None:
        self.id = 0
        self.flags = 0
        self.rcode = 0
        self.question: List[Dict[str, Any]] = []
        self.answers: List[Dict[str, Any]] = []
        self.authority: List[Dict[str, Any]] = []
        self.additionals: List[Dict[str, Any]] = []

    @property
    def rcode_name(self) -> str:
        return RCODES.get(self.rcode, "RCODE%d" % self.rcode)

    @property
    def truncated(self) -> bool:
        return bool(self.flags & FLAG_TC)

    def get_answers(self) -> List[Dict[str, Any]]:
        return self.answers

    def get_authority(self) -> List[Dict[str, Any]]:
        return self.authority

    def get_additionals(self) -> List[Dict[str, Any]]:
        return self.additionals


def _decode_rr(buf: bytes, off: int) -> Tuple[Dict[str, Any], int]:
    name, off = _decode_name(buf, off)
    rtype, rclass, ttl, rdlen = struct.unpack(">HHIH", buf[off:off + 10])
    off += 10
    rdata = buf[off:off + rdlen]
    rdstart = off
    off += rdlen
    rr: Dict[str, Any] = {
        "name": name,
        "type": TYPE_NAMES.get(rtype, str(rtype)),
        "class": rclass,
        "ttl": ttl,
    }
    if rtype == QTYPES["A"] and rdlen == 4:
        rr["target"] = socket.inet_ntop(socket.AF_INET, rdata)
    elif rtype == QTYPES["AAAA"] and rdlen == 16:
        rr["target"] = socket.inet_ntop(socket.AF_INET6, rdata)
    elif rtype == QTYPES["SRV"]:
        prio, weight, port = struct.unpack(">HHH", rdata[:6])
        target, _ = _decode_name(buf, rdstart + 6)
        rr.update(priority=prio, weight=weight, port=port, target=target)
    elif rtype in (QTYPES["CNAME"], QTYPES["DNAME"], QTYPES["NS"],
                   QTYPES["PTR"]):
        target, _ = _decode_name(buf, rdstart)
        rr["target"] = target
    elif rtype == QTYPES["SOA"]:
        mname, o2 = _decode_name(buf, rdstart)
        rname, o2 = _decode_name(buf, o2)
        serial, refresh, retry, expire, minimum = struct.unpack(
            ">IIIII", buf[o2:o2 + 20])
        rr.update(mname=mname, rname=rname, serial=serial, refresh=refresh,
                  retry=retry, expire=expire, minimum=minimum)
    elif rtype == QTYPES["OPT"]:
        rr["udp_payload"] = rclass
    else:
        rr["rdata"] = rdata
    return rr, off


def decode_message(buf: bytes) -> DnsMessage:
    """Decode; malformed/truncated input raises ValueError (never a
    bare struct.error — the client relies on the exception type to
    treat garbage datagrams as noise)."""
    if len(buf) < 12:
        raise ValueError("DNS message too short")
    try:
        msg = DnsMessage()
        (msg.id, msg.flags, qd, an, ns, ar) = struct.unpack(
            ">HHHHHH", buf[:12])
        msg.rcode = msg.flags & 0x0F
        off = 12
        for _ in range(qd):
            qname, off = _decode_name(buf, off)
            qtype, qclass = struct.unpack(">HH", buf[off:off + 4])
            off += 4
            msg.question.append({"name": qname,
                                 "type": TYPE_NAMES.get(qtype, str(qtype)),
                                 "class": qclass})
        for _ in range(an):
            rr, off = _decode_rr(buf, off)
            msg.answers.append(rr)
        for _ in range(ns):
            rr, off = _decode_rr(buf, off)
            msg.authority.append(rr)
        for _ in range(ar):
            rr, off = _decode_rr(buf, off)
            msg.additionals.append(rr)
    except struct.error as e:
        raise ValueError("truncated DNS message: %s" % e) from e
    return msg


# -- response encoding (used by the in-repo mock DNS server in tests and
#    benchmarks; real servers are of course remote) -----------------------

def _encode_rr(rr: Dict[str, Any]) -> bytes:
    rtype = QTYPES[rr["type"]]
    out = _encode_name(rr["name"])
    if rr["type"] == "A":
        rdata = socket.inet_pton(socket.AF_INET, rr["target"])
    elif rr["type"] == "AAAA":
        rdata = socket.inet_pton(socket.AF_INET6, rr["target"])
    elif rr["type"] == "SRV":
        rdata = struct.pack(">HHH", rr.get("priority", 0),
                            rr.get("weight", 0), rr["port"]) + \
            _encode_name(rr["target"])
    elif rr["type"] in ("CNAME", "DNAME", "NS", "PTR"):
        rdata = _encode_name(rr["target"])
    elif rr["type"] == "SOA":
        rdata = _encode_name(rr.get("mname", "ns0")) + \
            _encode_name(rr.get("rname", "root")) + \
            struct.pack(">IIIII", rr.get("serial", 1),
                        rr.get("refresh", 60), rr.get("retry", 60),
                        rr.get("expire", 60), rr.get("minimum", 60))
    else:
        rdata = rr.get("rdata", b"")
    out += struct.pack(">HHIH", rtype, QCLASS_IN, rr.get("ttl", 60),
                       len(rdata))
    return out + rdata


def encode_response(qid: int, question: Dict[str, Any], rcode: str = "NOERROR",
                    answers: Optional[List[Dict[str, Any]]] = None,
                    authority: Optional[List[Dict[str, Any]]] = None,
                    additionals: Optional[List[Dict[str, Any]]] = None,
                    aa: bool = True, tc: bool = False) -> bytes:
    answers = answers or []
    authority = authority or []
    additionals = additionals or []
    flags = FLAG_QR | FLAG_RA | (FLAG_AA if aa else 0) | (FLAG_TC if tc else 0)
    flags |= RCODE_NUMS[rcode] & 0x0F
    hdr = struct.pack(">HHHHHH", qid, flags, 1, len(answers), len(authority),
                      len(additionals))
    q = _encode_name(question["name"]) + struct.pack(
        ">HH", QTYPES[question["type"]], QCLASS_IN)
    body = b"".join(_encode_rr(rr) for rr in answers + authority + additionals)
    return hdr + q + body
