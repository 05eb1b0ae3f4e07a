"""Process-wide registry of pools/sets/DNS resolvers + kang snapshots.

Equivalent of reference lib/pool-monitor.js: a global singleton where
every pool, set and DNS resolver registers itself on start and
unregisters on stop; ``to_kang_options()`` serializes internal state for
the kang debugging endpoint (served by cueball_amd.kang).
"""

from __future__ import annotations

import socket
import time
from typing import Any, Dict, List

__all__ = ["PoolMonitor", "monitor"]


class PoolMonitor:
    def __init__(self) -> None:
        self.pm_pools: Dict[str, Any] = {}
        self.pm_sets: Dict[str, Any] = {}
        self.pm_dns_res: Dict[str, Any] = {}

    # -- registration ---------------------------------------------------
    def register_pool(self, pool: Any) -> None:
        self.pm_pools[pool.p_uuid] = pool

    def unregister_pool(self, pool: Any) -> None:
        if pool.p_uuid not in self.pm_pools:
            raise KeyError("pool %s not registered" % pool.p_uuid)
        del self.pm_pools[pool.p_uuid]

    def register_set(self, cset: Any) -> None:
        self.pm_sets[cset.cs_uuid] = cset

    def unregister_set(self, cset: Any) -> None:
        if cset.cs_uuid not in self.pm_sets:
            raise KeyError("set %s not registered" % cset.cs_uuid)
        del self.pm_sets[cset.cs_uuid]

    def register_dns_resolver(self, res: Any) -> None:
        self.pm_dns_res[res.r_uuid] = res

    def unregister_dns_resolver(self, res: Any) -> None:
        if res.r_uuid not in self.pm_dns_res:
            raise KeyError("resolver %s not registered" % res.r_uuid)
        del self.pm_dns_res[res.r_uuid]

    # -- kang serialization (lib/pool-monitor.js:60-216) ----------------
    def list_types(self) -> List[str]:
        return ["pool", "set", "dns_res"]

    def list_objects(self, type_: str) -> List[str]:
        if type_ == "pool":
            return list(self.pm_pools.keys())
        if type_ == "set":
            return list(self.pm_sets.keys())
        if type_ == "dns_res":
            return list(self.pm_dns_res.keys())
        raise ValueError('Invalid type "%s"' % type_)

    def get(self, type_: str, id_: str) -> Dict[str, Any]:
        if type_ == "pool":
            return self._get_pool(id_)
        if type_ == "set":
            return self._get_set(id_)
        if type_ == "dns_res":
            return self._get_dns_resolver(id_)
        raise ValueError('Invalid type "%s"' % type_)

    def _get_pool(self, id_: str) -> Dict[str, Any]:
        pool = self.pm_pools[id_]
        obj: Dict[str, Any] = {}
        obj["backends"] = {k: _backend_json(b)
                           for k, b in pool.p_backends.items()}
        obj["connections"] = {}
        ks = list(pool.p_keys)
        for k in pool.p_connections.keys():
            if k not in ks:
                ks.append(k)
        for k in ks:
            conns = pool.p_connections.get(k, [])
            per_state: Dict[str, int] = {}
            for fsm in conns:
                s = fsm.get_state()
                per_state[s] = per_state.get(s, 0) + 1
            obj["connections"][k] = per_state
        obj["dead_backends"] = list(pool.p_dead.keys())
        if pool.p_last_rebalance is not None:
            obj["last_rebalance"] = int(round(pool.p_last_rebalance))
        obj["resolvers"] = getattr(pool.p_resolver, "r_resolvers", None)
        obj["state"] = pool.get_state()
        obj["counters"] = dict(pool.p_counters)
        opts: Dict[str, Any] = {}
        res = getattr(pool.p_resolver, "r_fsm", None)
        opts["domain"] = getattr(res, "r_domain", None) or pool.p_domain
        opts["service"] = getattr(res, "r_service", None)
        opts["defaultPort"] = getattr(res, "r_defport", None)
        opts["spares"] = pool.p_spares
        opts["maximum"] = pool.p_max
        obj["options"] = opts
        return obj

    def _get_set(self, id_: str) -> Dict[str, Any]:
        cset = self.pm_sets[id_]
        obj: Dict[str, Any] = {}
        obj["backends"] = {k: _backend_json(b)
                           for k, b in cset.cs_backends.items()}
        obj["fsms"] = {}
        obj["connections"] = list(cset.cs_connections.keys())
        ks = list(cset.cs_keys)
        for k in cset.cs_fsm.keys():
            if k not in ks:
                ks.append(k)
        for k in ks:
            fsm = cset.cs_fsm.get(k)
            if fsm is None:
                continue
            s = fsm.get_state()
            obj["fsms"][k] = {s: 1}
        obj["dead_backends"] = list(cset.cs_dead.keys())
        if cset.cs_last_rebalance is not None:
            obj["last_rebalance"] = int(round(cset.cs_last_rebalance))
        obj["resolvers"] = getattr(cset.cs_resolver, "r_resolvers", None)
        obj["state"] = cset.get_state()
        obj["counters"] = dict(cset.cs_counters)
        obj["target"] = cset.cs_target
        obj["maximum"] = cset.cs_max
        opts: Dict[str, Any] = {}
        res = getattr(cset.cs_resolver, "r_fsm", None)
        opts["domain"] = getattr(res, "r_domain", None) or cset.cs_domain
        opts["service"] = getattr(res, "r_service", None)
        opts["defaultPort"] = getattr(res, "r_defport", None)
        obj["options"] = opts
        return obj

    def _get_dns_resolver(self, id_: str) -> Dict[str, Any]:
        res = self.pm_dns_res[id_]
        obj: Dict[str, Any] = {}
        obj["domain"] = res.r_domain
        obj["service"] = res.r_service
        obj["resolvers"] = res.r_resolvers
        obj["defaultPort"] = res.r_defport
        obj["state"] = res.get_state()
        # r_next_* are ms on the (monotonic) event-loop clock; convert
        # to epoch so the ISO times are real wall-clock expiries like
        # the reference's Date objects (lib/pool-monitor.js:189-195)
        loop = getattr(res, "_loop", None)
        if loop is not None:
            off = time.time() - loop.time()
        else:
            off = 0.0
        nxt: Dict[str, Any] = {}
        if getattr(res, "r_next_service", None):
            nxt["srv"] = _iso(res.r_next_service / 1000.0 + off)
        if getattr(res, "r_next_v6", None):
            nxt["v6"] = _iso(res.r_next_v6 / 1000.0 + off)
        if getattr(res, "r_next_v4", None):
            nxt["v4"] = _iso(res.r_next_v4 / 1000.0 + off)
        obj["next"] = nxt
        obj["backends"] = {k: _backend_json(b)
                           for k, b in res.r_backends.items()}
        obj["counters"] = dict(res.r_counters)
        return obj

    def to_kang_options(self) -> Dict[str, Any]:
        return {
            "uri_base": "/kang",
            "service_name": "cueball",
            "version": "1.0.0",
            "ident": socket.gethostname(),
            "list_types": self.list_types,
            "list_objects": self.list_objects,
            "get": self.get,
            "stats": lambda: {},
        }


def _backend_json(b: Any) -> Any:
    if isinstance(b, dict):
        return {k: v for k, v in b.items()}
    return b


def _iso(ts: float) -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime(ts))


#: process-wide singleton (lib/pool-monitor.js:9)
monitor = PoolMonitor()
