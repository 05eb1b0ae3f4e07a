"""Network-interface enumeration: os.networkInterfaces() equivalent.

The reference resolver skips the whole AAAA stage when no interface has
an IPv6 address other than ::1 (lib/resolver.js:738-772, using node's
``os.networkInterfaces()``).  This module provides the same view of the
host's interfaces:

1. ``getifaddrs(3)`` via ctypes (authoritative, both families);
2. fallback: ``/proc/net/if_inet6`` (IPv6) plus a UDP-connect probe
   for the primary IPv4 address;
3. last resort: ``getaddrinfo(gethostname())`` (the round-1 heuristic,
   which is wrong on hosts whose hostname does not resolve to their
   interface addresses — kept only when everything else fails).

``network_interfaces()`` returns ``{ifname: [{family: 'IPv4'|'IPv6',
address: str, internal: bool}]}``; scope-ids are stripped from
link-local addresses like node does.
"""

from __future__ import annotations

import ctypes
import ctypes.util
import ipaddress
import os
import socket
from typing import Any, Dict, List

__all__ = ["network_interfaces", "have_non_loopback_v6",
           "parse_proc_if_inet6"]


# -- getifaddrs via ctypes --------------------------------------------------

class _sockaddr(ctypes.Structure):
    _fields_ = [("sa_family", ctypes.c_ushort),
                ("sa_data", ctypes.c_ubyte * 14)]


class _sockaddr_in(ctypes.Structure):
    _fields_ = [("sin_family", ctypes.c_ushort),
                ("sin_port", ctypes.c_uint16),
                ("sin_addr", ctypes.c_ubyte * 4)]


class _sockaddr_in6(ctypes.Structure):
    _fields_ = [("sin6_family", ctypes.c_ushort),
                ("sin6_port", ctypes.c_uint16),
                ("sin6_flowinfo", ctypes.c_uint32),
                ("sin6_addr", ctypes.c_ubyte * 16),
                ("sin6_scope_id", ctypes.c_uint32)]


class _ifaddrs(ctypes.Structure):
    pass


_ifaddrs._fields_ = [
    ("ifa_next", ctypes.POINTER(_ifaddrs)),
    ("ifa_name", ctypes.c_char_p),
    ("ifa_flags", ctypes.c_uint),
    ("ifa_addr", ctypes.POINTER(_sockaddr)),
    ("ifa_netmask", ctypes.POINTER(_sockaddr)),
    ("ifa_ifu", ctypes.POINTER(_sockaddr)),
    ("ifa_data", ctypes.c_void_p),
]

_IFF_LOOPBACK = 0x8


def _getifaddrs() -> Dict[str, List[Dict[str, Any]]]:
    libc_name = ctypes.util.find_library("c") or "libc.so.6"
    libc = ctypes.CDLL(libc_name, use_errno=True)
    head = ctypes.POINTER(_ifaddrs)()
    if libc.getifaddrs(ctypes.byref(head)) != 0:
        raise OSError(ctypes.get_errno(), "getifaddrs failed")
    nics: Dict[str, List[Dict[str, Any]]] = {}
    try:
        node = head
        while node:
            ifa = node.contents
            name = (ifa.ifa_name or b"?").decode()
            nics.setdefault(name, [])
            internal = bool(ifa.ifa_flags & _IFF_LOOPBACK)
            if ifa.ifa_addr:
                fam = ifa.ifa_addr.contents.sa_family
                if fam == socket.AF_INET:
                    sa = ctypes.cast(
                        ifa.ifa_addr,
                        ctypes.POINTER(_sockaddr_in)).contents
                    addr = socket.inet_ntop(socket.AF_INET,
                                            bytes(sa.sin_addr))
                    nics[name].append({"family": "IPv4",
                                       "address": addr,
                                       "internal": internal})
                elif fam == socket.AF_INET6:
                    sa6 = ctypes.cast(
                        ifa.ifa_addr,
                        ctypes.POINTER(_sockaddr_in6)).contents
                    addr = socket.inet_ntop(socket.AF_INET6,
                                            bytes(sa6.sin6_addr))
                    nics[name].append({"family": "IPv6",
                                       "address": addr,
                                       "internal": internal})
            node = ifa.ifa_next
    finally:
        libc.freeifaddrs(head)
    # drop interfaces with no addresses, like node does
    return {k: v for (k, v) in nics.items() if v}


# -- /proc/net/if_inet6 fallback --------------------------------------------

def parse_proc_if_inet6(text: str) -> Dict[str, List[Dict[str, Any]]]:
    """Parse /proc/net/if_inet6 content: each line is
    ``<32-hex-addr> <ifindex> <prefixlen> <scope> <flags> <ifname>``."""
    nics: Dict[str, List[Dict[str, Any]]] = {}
    for line in text.splitlines():
        parts = line.split()
        if len(parts) != 6 or len(parts[0]) != 32:
            continue
        hexaddr, _, _, _, _, ifname = parts
        try:
            packed = bytes.fromhex(hexaddr)
            addr = socket.inet_ntop(socket.AF_INET6, packed)
        except (ValueError, OSError):
            continue
        nics.setdefault(ifname, []).append({
            "family": "IPv6",
            "address": addr,
            "internal": addr == "::1",
        })
    return nics


def _proc_v6() -> Dict[str, List[Dict[str, Any]]]:
    with open("/proc/net/if_inet6") as f:
        return parse_proc_if_inet6(f.read())


def _probe_v4() -> Dict[str, List[Dict[str, Any]]]:
    """Primary IPv4 via a UDP connect (no packets sent)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("192.0.2.1", 9))  # TEST-NET-1: never routed
        return {"_v4": [{"family": "IPv4",
                         "address": s.getsockname()[0],
                         "internal": False}]}
    except OSError:
        return {}
    finally:
        s.close()


def _hostname_heuristic() -> Dict[str, List[Dict[str, Any]]]:
    nics: Dict[str, List[Dict[str, Any]]] = {}
    try:
        infos = socket.getaddrinfo(socket.gethostname(), None)
    except OSError:
        return nics
    for family, _, _, _, sockaddr in infos:
        if family == socket.AF_INET6:
            nics.setdefault("_host", []).append(
                {"family": "IPv6", "address": sockaddr[0],
                 "internal": sockaddr[0] == "::1"})
        elif family == socket.AF_INET:
            nics.setdefault("_host", []).append(
                {"family": "IPv4", "address": sockaddr[0],
                 "internal": sockaddr[0].startswith("127.")})
    return nics


def network_interfaces() -> Dict[str, List[Dict[str, Any]]]:
    try:
        return _getifaddrs()
    except (OSError, AttributeError):
        pass
    nics: Dict[str, List[Dict[str, Any]]] = {}
    try:
        nics.update(_proc_v6())
    except OSError:
        pass
    nics.update(_probe_v4())
    if nics:
        return nics
    return _hostname_heuristic()


def have_non_loopback_v6(nics: Dict[str, List[Dict[str, Any]]]) -> bool:
    """The reference's AAAA-stage gate: any IPv6 address that is not
    ::1 counts — including link-locals (lib/resolver.js:749-754)."""
    for addrs in nics.values():
        for addr in addrs:
            if addr.get("family") == "IPv6" and \
                    addr.get("address") != "::1":
                return True
    return False


def have_global_v6(nics: Dict[str, List[Dict[str, Any]]]) -> bool:
    """Stricter variant (not what the reference does, offered for
    callers that want it): a global-scope IPv6 address exists."""
    for addrs in nics.values():
        for addr in addrs:
            if addr.get("family") != "IPv6":
                continue
            try:
                ip = ipaddress.IPv6Address(addr["address"])
            except ValueError:
                continue
            if not (ip.is_loopback or ip.is_link_local):
                return True
    return False
