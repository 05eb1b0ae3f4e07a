"""Service discovery: DNS SRV/AAAA/A resolver, static resolver, factory.

Re-design of reference lib/resolver.js.

Resolvers take a domain (+ service, default port) and emit ``added(key,
backend)`` / ``removed(key)`` as hosts appear and disappear.  The DNS
resolver runs the staged pipeline SRV -> AAAA -> A -> process -> sleep,
recording a TTL expiry per stage and waking only the stage whose TTL
expired (lib/resolver.js:152-241 has the state diagram).  "Bootstrap"
(Dynamic Resolver) mode discovers the nameservers themselves via DNS
(_dns._udp), sharing one refcounted bootstrap resolver per name
(lib/resolver.js:411, :476-491).

Public surface (lib/index.js): ``Resolver`` (DNS), ``DNSResolver``,
``StaticIpResolver``, ``resolver_for_ip_or_domain``; plus
``ResolverFSM``, ``config_for_ip_or_domain``, ``parse_ip_or_domain``
exposed for testing.
"""

from __future__ import annotations

import base64
import hashlib
import ipaddress
import math
import random
import uuid as mod_uuid
from typing import Any, Dict, List, Optional

from . import utils as mod_utils
from .dns_client import (DnsClient, MultiError, NoNameError, NoRecordsError,
                         TimeoutError_)
from .events import EventEmitter
from .fsm import FSM, StateScope, get_loop
from .logutil import CueballLogger, default_logger
from .pool_monitor import monitor as global_monitor

__all__ = [
    "Resolver",
    "DNSResolver",
    "StaticIpResolver",
    "ResolverFSM",
    "resolver_for_ip_or_domain",
    "config_for_ip_or_domain",
    "parse_ip_or_domain",
    "srv_key",
    "NoNameError",
    "NoRecordsError",
]


def _is_ip(s: str) -> int:
    """net.isIP() equivalent: 0 / 4 / 6."""
    try:
        return 4 if isinstance(ipaddress.ip_address(s),
                               ipaddress.IPv4Address) else 6
    except ValueError:
        return 0


def srv_key(srv: Dict[str, Any]) -> str:
    """Stable backend key: sha1(name || port || normalized address),
    base64 (lib/resolver.js:1157-1171)."""
    h = hashlib.sha1()
    h.update(str(srv["name"]).encode())
    h.update(b"||")
    h.update(str(srv["port"]).encode())
    h.update(b"||")
    try:
        addr = str(ipaddress.ip_address(srv["address"]))
    except ValueError:
        addr = str(srv["address"])
    h.update(addr.encode())
    return base64.b64encode(h.digest()).decode()


class ResolverFSM(FSM):
    """Wrapper FSM presenting the uniform Resolver interface over an
    inner implementation (lib/resolver.js:66-150).

    States: stopped -> starting -> running <-> failed ; stopping.
    """

    def __init__(self, fsm: Any, options: Dict[str, Any]) -> None:
        self.r_fsm = fsm
        self.r_last_error: Optional[BaseException] = None
        log: CueballLogger = options.get("log") or default_logger()
        self.r_log = log.child(component="ResolverFSM")
        super().__init__("stopped", loop=options.get("loop"))
        fsm.on("added", lambda k, b: self.emit("added", k, b))
        fsm.on("removed", lambda k: self.emit("removed", k))

    def start(self) -> None:
        self.emit("startAsserted")

    def stop(self) -> None:
        self.emit("stopAsserted")

    def count(self) -> int:
        return self.r_fsm.count()

    def list(self) -> Dict[str, Dict[str, Any]]:
        return self.r_fsm.list()

    def get_last_error(self) -> Optional[BaseException]:
        return self.r_last_error

    def state_stopped(self, S: StateScope) -> None:
        S.on(self, "startAsserted", lambda: S.goto_state("starting"))

    def state_starting(self, S: StateScope) -> None:
        self.r_fsm.start()

        def on_updated(err: Optional[BaseException] = None) -> None:
            if err:
                self.r_last_error = err
                S.goto_state("failed")
            else:
                S.goto_state("running")

        S.on(self.r_fsm, "updated", on_updated)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_running(self, S: StateScope) -> None:
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_failed(self, S: StateScope) -> None:
        def on_updated(err: Optional[BaseException] = None) -> None:
            if not err:
                S.goto_state("running")

        S.on(self.r_fsm, "updated", on_updated)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_stopping(self, S: StateScope) -> None:
        self.r_fsm.stop()
        S.immediate(lambda: S.goto_state("stopped"))


class _StaticResolver(EventEmitter):
    """Inner static-IP implementation (lib/resolver.js:1387-1456)."""

    def __init__(self, options: Dict[str, Any]) -> None:
        super().__init__()
        backends = options.get("backends")
        if not isinstance(backends, list):
            raise TypeError("options.backends (list) is required")
        self.sr_backends: List[Dict[str, Any]] = []
        for i, backend in enumerate(backends):
            addr = backend.get("address")
            if not isinstance(addr, str) or not _is_ip(addr):
                raise ValueError(
                    "options.backends[%d].address must be an IP address" % i)
            port = backend.get("port")
            if port is None:
                port = options.get("defaultPort")
            if not isinstance(port, int):
                raise TypeError("options.backends[%d].port is required" % i)
            self.sr_backends.append({
                "name": "%s:%d" % (addr, port),
                "address": addr,
                "port": port,
            })
        self.sr_state = "idle"
        self._loop = get_loop(options.get("loop"))

    def start(self) -> None:
        if self.sr_state != "idle":
            raise AssertionError(
                "cannot call start() again without calling stop()")
        self.sr_state = "started"

        def fire() -> None:
            for be in self.sr_backends:
                self.emit("added", srv_key(be), dict(be))
            self.emit("updated")

        self._loop.call_soon(fire)

    def stop(self) -> None:
        if self.sr_state != "started":
            raise AssertionError(
                "cannot call stop() again without calling start()")
        self.sr_state = "idle"

    def count(self) -> int:
        return len(self.sr_backends)

    def list(self) -> Dict[str, Dict[str, Any]]:
        return {srv_key(be): dict(be) for be in self.sr_backends}


def StaticIpResolver(options: Dict[str, Any]) -> ResolverFSM:
    """Emit a fixed list of IPs (development/debugging)."""
    return ResolverFSM(_StaticResolver(options), options)


class DNSResolverFSM(FSM):
    """Inner DNS resolution machine (lib/resolver.js:242-1377).

    States: init -> check_ns [-> bootstrap_ns] -> srv -> srv_try
    [-> srv_error] -> aaaa -> aaaa_next/aaaa_try [-> aaaa_error] -> a ->
    a_next/a_try [-> a_error] -> process -> sleep -> (stage with expired
    TTL).
    """

    #: shared bootstrap resolvers by DNS name (Dynamic Resolver mode)
    bootstrap_resolvers: Dict[str, "DNSResolverFSM"] = {}
    #: shared DnsClients keyed by concurrency
    global_ns_clients: Dict[int, DnsClient] = {}

    #: NIC-cache TTL in ms: skip AAAA entirely when the host has no
    #: global v6 address (lib/resolver.js:738-772)
    NIC_CACHE_TTL = 60000.0
    _nic_cache: Optional[Dict[str, List[Dict[str, Any]]]] = None
    _nic_cache_updated: Optional[float] = None

    def __init__(self, options: Dict[str, Any]) -> None:
        self.r_uuid = str(mod_uuid.uuid4())
        self.r_resolvers: List[str] = list(options.get("resolvers") or [])
        self.r_domain: str = options["domain"]
        self.r_service: str = options.get("service") or "_http._tcp"
        self.r_maxres: int = options.get("maxDNSConcurrency") or 3
        self.r_defport: int = options.get("defaultPort") or 80
        self.r_is_bootstrap = bool(options.get("_isBootstrap"))
        if self.r_is_bootstrap:
            # Bootstrap resolvers look up the DNS service itself and try
            # all possible resolvers (lib/resolver.js:264-278).
            self.r_service = "_dns._udp"
            self.r_defport = 53
            self.r_maxres = options.get("maxDNSConcurrency") or 10
            self.r_ref_count = 0

        log: CueballLogger = options.get("log") or default_logger()
        self.r_log = log.child(component="DNSResolverFSM",
                               domain=self.r_domain)

        recovery = options.get("recovery")
        if not isinstance(recovery, dict):
            raise TypeError("options.recovery is required")
        self.r_recovery = recovery

        dns_srv_recov = recovery["default"]
        dns_recov = recovery["default"]
        if recovery.get("dns") is not None:
            dns_srv_recov = recovery["dns"]
            dns_recov = recovery["dns"]
        if recovery.get("dns_srv") is not None:
            dns_srv_recov = recovery["dns_srv"]
        mod_utils.assert_recovery(dns_srv_recov, "recovery.dns_srv")
        mod_utils.assert_recovery(dns_recov, "recovery.dns")

        def mkretry(r: Dict[str, Any]) -> Dict[str, Any]:
            return {
                "max": r["retries"],
                "count": r["retries"],
                "timeout": r["timeout"],
                "minDelay": r["delay"],
                "delay": r["delay"],
                # default only when absent; explicit 0.0 spread is kept
                "delaySpread": (0.2 if r.get("delaySpread") is None
                                else r["delaySpread"]),
                "maxDelay": r.get("maxDelay") or math.inf,
            }

        self.r_srv_retry = mkretry(dns_srv_recov)
        self.r_retry = mkretry(dns_recov)

        loop = get_loop(options.get("loop"))

        # Next refresh times per stage, ms on the loop clock.  They
        # normally track TTL expiry, but in error cases they are the
        # next-attempt time (lib/resolver.js:327-339).
        now = loop.time() * 1000.0
        self.r_next_service: Optional[float] = now
        self.r_next_v6: Optional[float] = now
        self.r_next_v4: Optional[float] = now

        self.r_last_srv_ttl = 60.0
        self.r_last_ttl = 60.0
        self.r_last_error: Optional[BaseException] = None

        self.r_srvs: List[Dict[str, Any]] = []
        self.r_srv_rem: List[Dict[str, Any]] = []
        self.r_srv: Optional[Dict[str, Any]] = None
        self.r_backends: Dict[str, Dict[str, Any]] = {}

        self.r_bootstrap: Optional["DNSResolverFSM"] = None
        self.r_bootstrap_res: Dict[str, Dict[str, Any]] = {}

        # tests inject a scripted client here, like the reference's sinon
        # stub of mname-client (test/dns.test.js:75-110)
        nsclient = options.get("_nsclient")
        if nsclient is None:
            nsclient = DNSResolverFSM.global_ns_clients.get(self.r_maxres)
            if nsclient is None:
                nsclient = DnsClient(concurrency=self.r_maxres)
                DNSResolverFSM.global_ns_clients[self.r_maxres] = nsclient
        self.r_nsclient = nsclient

        self.r_stopping = False
        # anti-flap flags (lib/resolver.js:396-401)
        self.r_have_seen_srv = False
        self.r_have_seen_addr = False
        self.r_counters: Dict[str, float] = {}
        self.r_last_processed: Optional[Dict[str, List[str]]] = None

        super().__init__("init", loop=loop)

    # -- counters -------------------------------------------------------
    def _incr_counter(self, counter: str) -> None:
        self.r_counters[counter] = self.r_counters.get(counter, 0) + 1

    def _hwm_counter(self, counter: str, val: float) -> None:
        if self.r_counters.get(counter, -math.inf) < val:
            self.r_counters[counter] = val

    def _now_ms(self) -> float:
        return self._loop.time() * 1000.0

    # -- interface ------------------------------------------------------
    def start(self) -> None:
        self.emit("startAsserted")

    def stop(self) -> None:
        self.r_stopping = True
        self.emit("stopAsserted")

    def count(self) -> int:
        return len(self.r_backends)

    def list(self) -> Dict[str, Dict[str, Any]]:
        return dict(self.r_backends)

    # -- startup / bootstrap states --------------------------------------
    def state_init(self, S: StateScope) -> None:
        self.r_stopping = False
        global_monitor.register_dns_resolver(self)
        if self.r_bootstrap is not None:
            self.r_bootstrap.r_ref_count -= 1
            if self.r_bootstrap.r_ref_count <= 0:
                self.r_bootstrap.stop()
            self.r_bootstrap = None
        S.on(self, "startAsserted", lambda: S.goto_state("check_ns"))

    def state_check_ns(self, S: StateScope) -> None:
        if self.r_resolvers:
            # resolvers may use "ip@port" to target a non-53 port
            # (tests/mock servers); only a bare non-IP string means
            # Dynamic Resolver mode
            not_ip = [r for r in self.r_resolvers
                      if not _is_ip(r.split("@", 1)[0])]
            if not not_ip:
                S.goto_state("srv")
                return
            if len(not_ip) != 1:
                raise ValueError("at most one DNS name allowed in resolvers")
            # Dynamic Resolver mode: nameservers themselves come from DNS
            self.r_resolvers = []
            boot = DNSResolverFSM.bootstrap_resolvers.get(not_ip[0])
            if boot is None:
                boot = DNSResolverFSM({
                    "domain": not_ip[0],
                    "log": self.r_log,
                    "recovery": self.r_recovery,
                    "_isBootstrap": True,
                    "loop": self._loop,
                })
                DNSResolverFSM.bootstrap_resolvers[not_ip[0]] = boot
            self.r_bootstrap = boot
            boot.r_ref_count += 1
            S.goto_state("bootstrap_ns")
        else:
            def on_read() -> None:
                try:
                    with open("/etc/resolv.conf") as f:
                        content = f.read()
                except OSError:
                    self.r_resolvers = ["8.8.8.8", "8.8.4.4"]
                    S.goto_state("srv")
                    return
                self.r_resolvers = []
                for line in content.splitlines():
                    parts = line.split()
                    if len(parts) == 2 and parts[0] == "nameserver" and \
                            _is_ip(parts[1]):
                        self.r_resolvers.append(parts[1])
                if not self.r_resolvers:
                    self.r_resolvers = ["8.8.8.8", "8.8.4.4"]
                S.goto_state("srv")

            S.immediate(on_read)

    def state_bootstrap_ns(self, S: StateScope) -> None:
        boot = self.r_bootstrap

        def on_added(k: str, srv: Dict[str, Any]) -> None:
            self.r_bootstrap_res[k] = srv
            self.r_resolvers.append(srv["address"])

        def on_removed(k: str) -> None:
            srv = self.r_bootstrap_res.pop(k)
            self.r_resolvers.remove(srv["address"])

        # NB: unscoped on purpose — bootstrap membership must keep
        # updating while we are in srv/aaaa/... (lib/resolver.js:515-526)
        boot.on("added", on_added)
        boot.on("removed", on_removed)

        if boot.count() > 0:
            srvs = boot.list()
            self.r_bootstrap_res = srvs
            for k in srvs:
                self.r_resolvers.append(srvs[k]["address"])
            S.goto_state("srv")
        else:
            S.on(boot, "added", lambda *a: S.goto_state("srv"))
            boot.start()

    # -- SRV stage --------------------------------------------------------
    def state_srv(self, S: StateScope) -> None:
        r = self.r_srv_retry
        r["delay"] = r["minDelay"]
        r["count"] = r["max"]
        S.goto_state("srv_try")

    def state_srv_try(self, S: StateScope) -> None:
        name = self.r_service + "." + self.r_domain
        req = self.resolve(name, "SRV", self.r_srv_retry["timeout"])

        def on_answers(ans: List[Dict[str, Any]], ttl: float) -> None:
            self.r_next_service = self._now_ms() + 1000.0 * ttl
            self.r_last_srv_ttl = ttl
            self.r_last_ttl = ttl
            self.r_have_seen_srv = True

            # carry over cached A/AAAA results for unchanged SRV targets
            old_lookup: Dict[str, Dict[int, Dict[str, Any]]] = {}
            for srv in self.r_srvs:
                old_lookup.setdefault(srv["name"], {})[srv["port"]] = srv
            for srv in ans:
                old = old_lookup.get(srv["name"], {}).get(srv["port"])
                if old is None:
                    continue
                for fld in ("expiry_v4", "addresses_v4",
                            "expiry_v6", "addresses_v6"):
                    if old.get(fld) is not None:
                        srv[fld] = old[fld]

            self.r_srvs = ans
            S.goto_state("aaaa")

        S.on(req, "answers", on_answers)

        def on_error(err: BaseException) -> None:
            self.r_last_error = _verror(
                err, 'SRV lookup for "%s" failed' % name)
            self._incr_counter("srv-failure")
            code = getattr(err, "code", None)

            if isinstance(err, (NoRecordsError, NoNameError)) or \
                    code == "NOTIMP":
                # No SRV records (NXDOMAIN/NODATA/NOTIMP): look up the
                # base domain as a plain name instead, and don't retry
                # SRV for a while (lib/resolver.js:591-644).
                self.r_srvs = [{
                    "name": self.r_domain,
                    "port": self.r_defport,
                }]
                ttl = 60.0 * 60.0
                if code == "NOTIMP":
                    self.r_log.info("SRV got NOTIMP for %s; retry in %d "
                                    "seconds", self.r_service, ttl)
                else:
                    if getattr(err, "ttl", None):
                        ttl = err.ttl
                    self.r_log.info("no SRV records for %s; retry in %d "
                                    "seconds", self.r_service, ttl)
                self.r_next_service = self._now_ms() + ttl * 1000.0
                self._incr_counter("srv-skipped")
                S.goto_state("aaaa")
            elif code == "REFUSED":
                # retrying is pointless (lib/resolver.js:645-652)
                self.r_srv_retry["count"] = 0
                S.goto_state("srv_error")
            else:
                S.goto_state("srv_error")

        S.on(req, "error", on_error)
        req.send()

    def state_srv_error(self, S: StateScope) -> None:
        r = self.r_srv_retry
        r["count"] -= 1
        if r["count"] > 0:
            delay = mod_utils.gen_delay(r["delay"], r["delaySpread"])
            S.timeout(delay, lambda: S.goto_state("srv_try"))
            r["delay"] *= 2
            if r["delay"] > r["maxDelay"]:
                r["delay"] = r["maxDelay"]
            return

        self.r_log.trace("repeated error during SRV resolution for service "
                         "%s, will retry in %d sec", self.r_service,
                         self.r_last_srv_ttl)
        self.r_srvs = [{"name": self.r_domain, "port": self.r_defport}]
        d = self._now_ms() + 1000.0 * self.r_last_srv_ttl
        self.r_next_service = d

        # Anti-flap: only fall back to plain A/AAAA if SRV has *never*
        # succeeded.  The initial-failure fallthrough is load-bearing —
        # node-moray sets a 1ms SRV timeout and expects it
        # (lib/resolver.js:687-723).
        if not self.r_have_seen_srv and not self.r_have_seen_addr:
            self.r_log.debug("no SRV records found for service %s, trying "
                             "as a plain name", self.r_service)
            S.goto_state("aaaa")
            return
        elif not self.r_have_seen_srv:
            self.r_log.info("no SRV records found for service %s, falling "
                            "back to A/AAAA for 15min", self.r_service)
            self.r_next_service = self._now_ms() + 1000.0 * 60 * 15
            S.goto_state("aaaa")
            return

        # make sure the next wake-up is for SRV, not A/AAAA
        if self.r_next_v6 is not None and self.r_next_v6 < d:
            self.r_next_v6 = d
        if self.r_next_v4 is not None and self.r_next_v4 < d:
            self.r_next_v4 = d
        S.goto_state("sleep")

    # -- AAAA stage -------------------------------------------------------
    @classmethod
    def _get_nics(cls) -> Dict[str, List[Dict[str, Any]]]:
        """os.networkInterfaces() equivalent (lib/resolver.js:738-772):
        real interface enumeration via getifaddrs(3) with
        /proc/net/if_inet6 and hostname-lookup fallbacks (netif.py)."""
        from . import netif
        return netif.network_interfaces()

    def state_aaaa(self, S: StateScope) -> None:
        now = self._now_ms()
        cls = DNSResolverFSM
        if cls._nic_cache_updated is None or \
                now - cls._nic_cache_updated > cls.NIC_CACHE_TTL:
            cls._nic_cache = self._get_nics()
            cls._nic_cache_updated = now
        nics = cls._nic_cache or {}
        have_v6 = any(
            addr.get("family") == "IPv6" and addr.get("address") != "::1"
            for addrs in nics.values() for addr in addrs)
        if have_v6:
            self.r_next_v6 = None
            self.r_srv_rem = list(self.r_srvs)
            S.goto_state("aaaa_next")
        else:
            # come back after the NIC cache has definitely expired
            self.r_next_v6 = cls._nic_cache_updated + cls.NIC_CACHE_TTL + 1
            S.goto_state("a")

    def state_aaaa_next(self, S: StateScope) -> None:
        r = self.r_retry
        r["delay"] = r["minDelay"]
        r["count"] = r["max"]
        if self.r_srv_rem:
            self.r_srv = self.r_srv_rem.pop(0)
            S.goto_state("aaaa_try")
        else:
            S.goto_state("a")

    def state_aaaa_try(self, S: StateScope) -> None:
        srv = self.r_srv

        if srv.get("additionals"):
            self.r_log.trace("skipping v6 lookup for %s, using additionals "
                             "from SRV", srv["name"])
            srv["addresses_v6"] = [a for a in srv["additionals"]
                                   if _is_ip(a) == 6]
            S.goto_state("aaaa_next")
            return

        now = self._now_ms()
        if srv.get("expiry_v6") is not None and srv["expiry_v6"] > now:
            if self.r_next_v6 is None or srv["expiry_v6"] <= self.r_next_v6:
                self.r_next_v6 = srv["expiry_v6"]
            S.goto_state("aaaa_next")
            return

        req = self.resolve(srv["name"], "AAAA", self.r_retry["timeout"])

        def on_answers(ans: List[Dict[str, Any]], ttl: float) -> None:
            d = self._now_ms() + 1000.0 * ttl
            if self.r_next_v6 is None or d <= self.r_next_v6:
                self.r_next_v6 = d
            self.r_last_ttl = ttl
            self.r_have_seen_addr = True
            srv["expiry_v6"] = d
            srv["addresses_v6"] = [v["address"] for v in ans]
            S.goto_state("aaaa_next")

        S.on(req, "answers", on_answers)

        def on_error(err: BaseException) -> None:
            code = getattr(err, "code", None)
            if isinstance(err, NoRecordsError) or code == "NOTIMP":
                # NODATA => name probably has only A records: skip, and
                # cache the skip for NIC_CACHE_TTL
                srv["expiry_v6"] = self._now_ms() + self.NIC_CACHE_TTL
                S.goto_state("aaaa_next")
                return
            elif code == "REFUSED":
                self.r_retry["count"] = 0
            self.r_last_error = _verror(
                err, 'IPv6 (AAAA) lookup failed for "%s"' % srv["name"])
            S.goto_state("aaaa_error")

        S.on(req, "error", on_error)
        req.send()

    def state_aaaa_error(self, S: StateScope) -> None:
        r = self.r_retry
        r["count"] -= 1
        if r["count"] > 0:
            delay = mod_utils.gen_delay(r["delay"], r["delaySpread"])
            S.timeout(delay, lambda: S.goto_state("aaaa_try"))
            r["delay"] *= 2
            if r["delay"] > r["maxDelay"]:
                r["delay"] = r["maxDelay"]
            return
        self.r_log.trace("repeated error during AAAA resolution for name "
                         "%s, proceeding", self.r_srv["name"])
        d = self._now_ms() + 1000.0 * 60 * 60
        if self.r_next_v6 is None or d <= self.r_next_v6:
            self.r_next_v6 = d
        S.goto_state("aaaa_next")

    # -- A stage ----------------------------------------------------------
    def state_a(self, S: StateScope) -> None:
        self.r_next_v4 = None
        self.r_srv_rem = list(self.r_srvs)
        S.goto_state("a_next")

    def state_a_next(self, S: StateScope) -> None:
        r = self.r_retry
        r["delay"] = r["minDelay"]
        r["count"] = r["max"]
        if self.r_srv_rem:
            self.r_srv = self.r_srv_rem.pop(0)
            S.goto_state("a_try")
        else:
            S.goto_state("process")

    def state_a_try(self, S: StateScope) -> None:
        srv = self.r_srv

        if srv.get("additionals"):
            self.r_log.trace("skipping v4 lookup for %s, using additionals "
                             "from SRV", srv["name"])
            srv["addresses_v4"] = [a for a in srv["additionals"]
                                   if _is_ip(a) == 4]
            S.goto_state("a_next")
            return

        now = self._now_ms()
        if srv.get("expiry_v4") is not None and srv["expiry_v4"] > now:
            if self.r_next_v4 is None or srv["expiry_v4"] <= self.r_next_v4:
                self.r_next_v4 = srv["expiry_v4"]
            S.goto_state("a_next")
            return

        req = self.resolve(srv["name"], "A", self.r_retry["timeout"])

        def on_answers(ans: List[Dict[str, Any]], ttl: float) -> None:
            d = self._now_ms() + 1000.0 * ttl
            if self.r_next_v4 is None or d <= self.r_next_v4:
                self.r_next_v4 = d
            self.r_last_ttl = ttl
            self.r_have_seen_addr = True
            srv["expiry_v4"] = d
            srv["addresses_v4"] = [v["address"] for v in ans]
            S.goto_state("a_next")

        S.on(req, "answers", on_answers)

        def on_error(err: BaseException) -> None:
            code = getattr(err, "code", None)
            if isinstance(err, NoRecordsError):
                # NODATA for A: fine if we got AAAAs, else not retryable
                if srv.get("addresses_v6"):
                    S.goto_state("a_next")
                    return
                self.r_retry["count"] = 0
            elif isinstance(err, NoNameError):
                self.r_retry["count"] = 0
            elif code == "REFUSED":
                self.r_retry["count"] = 0
            self.r_last_error = _verror(
                err, 'IPv4 (A) lookup for "%s" failed' % srv["name"])
            S.goto_state("a_error")

        S.on(req, "error", on_error)
        req.send()

    def state_a_error(self, S: StateScope) -> None:
        r = self.r_retry
        r["count"] -= 1
        if r["count"] > 0:
            delay = mod_utils.gen_delay(r["delay"], r["delaySpread"])
            S.timeout(delay, lambda: S.goto_state("a_try"))
            r["delay"] *= 2
            if r["delay"] > r["maxDelay"]:
                r["delay"] = r["maxDelay"]
            return
        self.r_log.debug("repeated error during A resolution for name %s, "
                         "proceeding", self.r_srv["name"])
        d = self._now_ms() + 1000.0 * self.r_last_ttl
        if self.r_next_v4 is None or d <= self.r_next_v4:
            self.r_next_v4 = d
        S.goto_state("a_next")

    # -- process + sleep ---------------------------------------------------
    def state_process(self, S: StateScope) -> None:
        old_backends = self.r_backends
        new_backends: Dict[str, Dict[str, Any]] = {}
        all_addrs: List[str] = []
        for srv in self.r_srvs:
            srv["addresses"] = list(srv.get("addresses_v6") or []) + \
                list(srv.get("addresses_v4") or [])
            for addr in srv["addresses"]:
                final = {"name": srv["name"], "port": srv["port"],
                         "address": addr}
                all_addrs.append(addr)
                new_backends[srv_key(final)] = final

        if not new_backends:
            err = _verror(self.r_last_error,
                          "failed to find any DNS records for (%s.)%s"
                          % (self.r_service, self.r_domain))
            self._incr_counter("empty-set")
            self.r_log.warn("finished processing with empty set")
            self.emit("updated", err)
            S.goto_state("sleep")
            return

        removed = [k for k in old_backends if k not in new_backends]
        added = [k for k in new_backends if k not in old_backends]

        self.r_backends = new_backends

        if old_backends and (removed or added):
            self.r_log.info("records changed in DNS", added=len(added),
                            removed=len(removed))

        for k in removed:
            self.r_log.trace("host removed: %s", k)
            self.emit("removed", k)
            self._incr_counter("backend-removed")
        for k in added:
            self.r_log.trace("host added: %s", k)
            self.emit("added", k, new_backends[k])
            self._incr_counter("backend-added")

        if self.r_is_bootstrap:
            gone = [r for r in self.r_resolvers if r not in all_addrs]
            self.r_resolvers = all_addrs
            if gone:
                self.r_log.info("removed %d resolvers from bootstrap",
                                len(gone))

        self.emit("updated", None)
        self.r_last_processed = {"added": added, "removed": removed}
        S.goto_state("sleep")

    def state_sleep(self, S: StateScope) -> None:
        if self.r_stopping:
            S.goto_state("init")
            return

        now = self._now_ms()
        min_delay = (self.r_next_service or math.inf) - now
        state = "srv"
        if (self.r_next_v6 or math.inf) - now < min_delay:
            min_delay = (self.r_next_v6 or math.inf) - now
            state = "aaaa"
        if (self.r_next_v4 or math.inf) - now < min_delay:
            min_delay = (self.r_next_v4 or math.inf) - now
            state = "a"

        self._hwm_counter("max-sleep", min_delay)

        if min_delay < 0:
            S.goto_state(state)
        else:
            # TTL expiries spread forwards only: retrying early just
            # re-hits the cache (lib/resolver.js:1136-1144)
            delay = round(min_delay *
                          (1 + random.random() * self.r_retry["delaySpread"]))
            self.r_log.trace("sleeping until next TTL expiry",
                             state=state, delay=delay)
            S.timeout(delay, lambda: S.goto_state(state))
            S.on(self, "stopAsserted", lambda: S.goto_state("init"))

    # -- the DNS query adapter (lib/resolver.js:1210-1377) -----------------
    def resolve(self, domain: str, rtype: str, timeout: float):
        opts: Dict[str, Any] = {
            "domain": domain,
            "type": rtype,
            "timeout": timeout,
            "resolvers": self.r_resolvers,
        }
        if self.r_is_bootstrap:
            opts["errorThreshold"] = min(self.r_maxres,
                                         len(self.r_resolvers)) or 1

        em = _DnsRequest()

        def on_lookup(err: Optional[BaseException], msg: Any) -> None:
            # Vote on the most common rcode across a MultiError
            if isinstance(err, MultiError):
                codes: Dict[str, int] = {}
                for e in err.errors():
                    if isinstance(e, TimeoutError_):
                        self._incr_counter("timeout")
                        continue
                    code = getattr(e, "code", None)
                    if code is None:
                        continue
                    codes[code] = codes.get(code, 0) + 1
                    self._incr_counter("rcode-" + code.lower())
                if codes:
                    err.code = sorted(codes, key=lambda c: -codes[c])[0]
            if err is not None and getattr(err, "code", None) == "NXDOMAIN":
                err = NoNameError(domain, cause=err)

            # NODATA: a successful response with zero answers; pick up a
            # TTL from an SOA in the authority section if present
            if err is None and msg is not None and not msg.get_answers():
                ttl = None
                for v in msg.get_authority():
                    if v.get("type") == "SOA" and v.get("ttl", 0) > 0:
                        ttl = v["ttl"]
                err = NoRecordsError(domain, rtype, ttl)

            if err is not None:
                code = getattr(err, "code", None)
                if code:
                    self._incr_counter("rcode-" + str(code).lower())
                em.emit("error", err)
                return

            answers = msg.get_answers()
            min_ttl: Optional[float] = None
            self._incr_counter("rcode-ok")
            ans: List[Dict[str, Any]] = []

            if rtype in ("A", "AAAA"):
                for a in answers:
                    if a.get("type") != rtype:
                        if a.get("type") in ("CNAME", "DNAME"):
                            self._incr_counter("cname")
                            continue
                        self._incr_counter("unknown-rrtype")
                        self.r_log.warn("got unsupported answer rrtype: %s",
                                        a.get("type"))
                        continue
                    if min_ttl is None or a["ttl"] < min_ttl:
                        min_ttl = a["ttl"]
                    ans.append({"name": a["name"], "address": a["target"]})
            elif rtype == "SRV":
                cache: Dict[str, List[str]] = {}
                for rr in msg.get_additionals():
                    if rr.get("type") not in ("A", "AAAA"):
                        if rr.get("type") in ("CNAME", "DNAME", "OPT"):
                            continue
                        self._incr_counter("unknown-rrtype")
                        self.r_log.warn("got unsupported additional rrtype: "
                                        "%s", rr.get("type"))
                        continue
                    if rr.get("target"):
                        if min_ttl is None or rr["ttl"] < min_ttl:
                            min_ttl = rr["ttl"]
                        cache.setdefault(rr["name"], []).append(rr["target"])
                for a in answers:
                    if a.get("type") != rtype:
                        if a.get("type") in ("CNAME", "DNAME"):
                            self._incr_counter("cname")
                            continue
                        self._incr_counter("unknown-rrtype")
                        self.r_log.warn("got unsupported answer rrtype: %s",
                                        a.get("type"))
                        continue
                    if min_ttl is None or a["ttl"] < min_ttl:
                        min_ttl = a["ttl"]
                    obj: Dict[str, Any] = {"name": a["target"],
                                           "port": a["port"]}
                    if a["target"] in cache:
                        self._incr_counter("additionals-used")
                        obj["additionals"] = cache[a["target"]]
                    ans.append(obj)
            else:
                raise ValueError("Invalid record type " + rtype)

            if not ans:
                em.emit("error", NoRecordsError(domain, rtype))
                return
            em.emit("answers", ans, min_ttl)

        def send() -> None:
            self.r_nsclient.lookup(opts, on_lookup, loop=self._loop)

        em.send = send  # type: ignore[attr-defined]
        return em


class _DnsRequest(EventEmitter):
    """One in-flight DNS question: emits 'answers'(ans, ttl) or
    'error'(err) after .send()."""

    send: Any = None


def _verror(cause: Optional[BaseException], msg: str) -> BaseException:
    from .errors import CueballError
    return CueballError(msg, cause)


def DNSResolver(options: Dict[str, Any]) -> ResolverFSM:
    """DNS SRV/AAAA/A resolver (the primary Resolver implementation)."""
    return ResolverFSM(DNSResolverFSM(options), options)


#: compatibility alias — "Resolver" is the DNS resolver (lib/resolver.js:9-13)
Resolver = DNSResolver


def parse_ip_or_domain(s: str):
    """Parse "HOSTNAME[:PORT]" into a resolver kind + config
    (lib/resolver.js:1533-1573).  Returns an Error-like ValueError
    instance (not raised) on invalid input, like the reference."""
    colon = s.rfind(":")
    if colon == -1:
        first = s
        port = None
    else:
        first = s[:colon]
        try:
            port = int(s[colon + 1:])
        except ValueError:
            return ValueError("unsupported port in input: " + s)
        if port < 0 or port > 65535:
            return ValueError("unsupported port in input: " + s)

    if _is_ip(first) == 0:
        ret = {
            "kind": "dns",
            "cons": DNSResolver,
            "config": {"domain": first},
        }
        if port is not None:
            ret["config"]["defaultPort"] = port
    else:
        ret = {
            "kind": "static",
            "cons": StaticIpResolver,
            "config": {"backends": [{"address": first, "port": port}]},
        }
    return ret


def config_for_ip_or_domain(args: Dict[str, Any]):
    if not isinstance(args.get("input"), str):
        raise TypeError("args.input (string) is required")
    rcfg = dict(args.get("resolverConfig") or {})
    speccfg = parse_ip_or_domain(args["input"])
    if isinstance(speccfg, Exception):
        return speccfg
    rcfg.update(speccfg["config"])
    speccfg["mergedConfig"] = rcfg
    return speccfg


def resolver_for_ip_or_domain(args: Dict[str, Any]):
    """Build a static or DNS resolver from user input "HOST[:PORT]";
    returns an Error instance (not raised) if the input is invalid
    (lib/resolver.js:1485-1497)."""
    speccfg = config_for_ip_or_domain(args)
    if isinstance(speccfg, Exception):
        return speccfg
    return speccfg["cons"](speccfg["mergedConfig"])
