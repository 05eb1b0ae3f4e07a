"""cbresolve: resolve a name via the cueball resolver and print/follow
backends (reference bin/cbresolve).

Usage:
    cbresolve HOSTNAME[:PORT]                # DNS-based lookup
    cbresolve -S IP[:PORT]...                # static IPs

Options (DNS lookups):
    -f/--follow        periodically re-resolve and report changes
    -p/--port PORT     default backend port
    -r/--resolvers     comma-separated DNS resolver IPs
    -s/--service       SRV "service" name (e.g. _http._tcp)
    -t/--timeout       lookup timeout (e.g. 5000, 5s, 1m)
    -k/--kang-port     start a kang snapshot listener on this port
"""

from __future__ import annotations

import argparse
import asyncio
import logging
import os
import re
import sys
import time
from typing import Any, Dict, Optional

from .logutil import CueballLogger
from .resolver import StaticIpResolver, resolver_for_ip_or_domain

__all__ = ["main"]


def parse_ip_port(s: str) -> Dict[str, Any]:
    import ipaddress
    colon = s.rfind(":")
    if colon == -1:
        raise ValueError('not an "IP:port" pair: %s' % s)
    ip = s[:colon]
    try:
        ipaddress.ip_address(ip)
        port = int(s[colon + 1:])
    except ValueError:
        raise ValueError('not an "IP:port" pair: %s' % s)
    return {"address": ip, "port": port}


def parse_time_interval(s: str) -> int:
    """'5000', '5s', '2m', '150ms' -> milliseconds."""
    m = re.match(r"^([1-9][0-9]*)(s|ms|m)?$", s)
    if m is None:
        raise ValueError("invalid time interval: %s" % s)
    n = int(m.group(1))
    unit = m.group(2)
    if unit == "s":
        return n * 1000
    if unit == "m":
        return n * 60000
    return n


def main(argv: Optional[list] = None) -> int:
    ap = argparse.ArgumentParser(
        prog="cbresolve",
        description="Locate services in DNS using the cueball resolver.")
    ap.add_argument("-S", "--static", action="store_true",
                    help="treat arguments as static IP[:PORT] backends")
    ap.add_argument("-f", "--follow", action="store_true",
                    help="periodically re-resolve and report changes")
    ap.add_argument("-p", "--port", type=int, default=None,
                    help="default backend port")
    ap.add_argument("-r", "--resolvers", default=None,
                    help="comma-separated list of DNS resolvers")
    ap.add_argument("-s", "--service", default=None,
                    help='SRV "service" name')
    ap.add_argument("-t", "--timeout", default="5000",
                    help="timeout for lookups (e.g. 5000, 5s, 1m)")
    ap.add_argument("-k", "--kang-port", type=int, default=None,
                    help="start kang listener on this port")
    ap.add_argument("names", nargs="+", metavar="HOSTNAME[:PORT]")
    args = ap.parse_args(argv)

    level = os.environ.get("LOG_LEVEL", "CRITICAL").upper()
    logging.basicConfig(level=getattr(logging, level, logging.CRITICAL))
    log = CueballLogger(logging.getLogger("cbresolve"))

    try:
        timeout = parse_time_interval(args.timeout)
    except ValueError as e:
        ap.error(str(e))

    resolver_conf: Dict[str, Any] = {"log": log}
    if args.port is not None:
        if not (0 <= args.port <= 65535):
            ap.error("bad value for -p/--port: %d" % args.port)
        resolver_conf["defaultPort"] = args.port
    if args.resolvers:
        resolver_conf["resolvers"] = [
            ip for ip in args.resolvers.split(",") if ip]
    if args.service:
        resolver_conf["service"] = args.service

    rc = {"code": 0}

    async def run() -> None:
        backends: Dict[str, Dict[str, Any]] = {}
        done = asyncio.get_running_loop().create_future()

        if args.static:
            if args.follow:
                print("-f/--follow cannot be used with -S/--static",
                      file=sys.stderr)
                rc["code"] = 2
                return
            try:
                resolver_conf["backends"] = [parse_ip_port(p)
                                             for p in args.names]
            except ValueError as e:
                print(str(e), file=sys.stderr)
                rc["code"] = 2
                return
            print("using static IP resolver", file=sys.stderr)
            resolver = StaticIpResolver(resolver_conf)
        else:
            if len(args.names) != 1:
                print("exactly one HOSTNAME[:PORT] required",
                      file=sys.stderr)
                rc["code"] = 2
                return
            print("domain: %s" % args.names[0], file=sys.stderr)
            print("timeout: %d milliseconds" % timeout, file=sys.stderr)
            resolver_conf["recovery"] = {
                "default": {"retries": 0, "timeout": timeout,
                            "maxTimeout": timeout, "delay": 0,
                            "maxDelay": 0},
            }
            resolver = resolver_for_ip_or_domain({
                "input": args.names[0],
                "resolverConfig": resolver_conf,
            })
            if isinstance(resolver, Exception):
                print(str(resolver), file=sys.stderr)
                rc["code"] = 2
                return

        def on_added(key: str, backend: Dict[str, Any]) -> None:
            backends[key] = backend
            if args.follow:
                print("%s added   %16s:%-5d (%s)" % (
                    time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
                    backend["address"], backend["port"], key))
            else:
                print("%-16s %5d %s" % (backend["address"],
                                        backend["port"], key))

        def on_removed(key: str) -> None:
            old = backends.pop(key, None)
            if args.follow and old is not None:
                print("%s removed %16s:%-5d (%s)" % (
                    time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
                    old["address"], old["port"], key))

        resolver.on("added", on_added)
        resolver.on("removed", on_removed)

        def on_state(st: str) -> None:
            if args.follow:
                return
            if st == "running":
                resolver.stop()
                if not done.done():
                    done.set_result(None)
            elif st == "failed":
                err = resolver.get_last_error()
                print("error: %s" % err, file=sys.stderr)
                rc["code"] = 1
                if not done.done():
                    done.set_result(None)

        resolver.on("stateChanged", on_state)

        kang_server = None
        if args.kang_port is not None:
            from .kang import KangServer
            kang_server = KangServer()
            await kang_server.start(args.kang_port)
            print("kang listener on port %d" % kang_server.port,
                  file=sys.stderr)

        resolver.start()
        if args.follow:
            while True:
                await asyncio.sleep(3600)
        else:
            await done
        if kang_server is not None:
            kang_server.stop()

    try:
        asyncio.run(run())
    except KeyboardInterrupt:
        pass
    return rc["code"]


if __name__ == "__main__":
    sys.exit(main())
