"""ConnectionSet: at most one connection per backend, added/removed contract.

Re-design of reference lib/set.js.  Unlike a pool, a Set hands every
connection to the consumer as soon as it is up (for protocols that
multiplex many requests over one connection): it emits ``added(ckey,
conn, handle)`` — which MUST be handled — and later ``removed(ckey,
conn, handle)`` when the connection should be drained; the consumer then
calls ``handle.release()`` (drained cleanly) or ``handle.close()``.

Each connection key is serial-numbered ``<backend-key>.<n>``; a
LogicalConnection FSM tracks one ckey through init -> advertised ->
draining -> stopped (state diagram at lib/set.js:632-675).
"""

from __future__ import annotations

import math
import random
import time as mod_time
import uuid as mod_uuid
from typing import Any, Dict, List, Optional

from . import utils as mod_utils
from .connection_fsm import ClaimHandle, ConnectionSlotFSM
from .fsm import FSM, StateScope
from .logutil import CueballLogger, default_logger
from .pool import _IntervalTimer
from .pool_monitor import monitor as global_monitor

__all__ = ["ConnectionSet", "LogicalConnection"]


class ConnectionSet(FSM):
    def __init__(self, options: Dict[str, Any]) -> None:
        if not callable(options.get("constructor")):
            raise TypeError("options.constructor (callable) is required")
        loop_opt = options.get("loop")

        self.cs_uuid = str(mod_uuid.uuid4())
        self.cs_constructor = options["constructor"]

        if options.get("resolver") is None:
            raise TypeError("options.resolver is required")
        self.cs_resolver = options["resolver"]

        recovery = options.get("recovery")
        mod_utils.assert_recovery_set(recovery)
        self.cs_recovery = recovery

        self.cs_conn_handles_err = bool(options.get("connectionHandlesError"))

        log: CueballLogger = options.get("log") or default_logger()
        self.cs_log = log.child(
            component="ConnectionSet",
            domain=options.get("domain"),
            service=options.get("service"),
            cset=self.cs_uuid,
        )
        self.cs_domain = options.get("domain")

        self.cs_collector = mod_utils.create_error_metrics(options)

        target = options.get("target")
        maximum = options.get("maximum")
        if not isinstance(target, int) or not isinstance(maximum, int):
            raise TypeError("options.target and options.maximum are required")
        self.cs_target = target
        self.cs_max = maximum

        self.cs_keys: List[str] = []
        self.cs_backends: Dict[str, Dict[str, Any]] = {}
        self.cs_fsm: Dict[str, ConnectionSlotFSM] = {}
        self.cs_dead: Dict[str, bool] = {}

        # serial numbers generate per-connection keys "key.N"
        self.cs_serials: Dict[str, int] = {}
        self.cs_connections: Dict[str, Any] = {}
        self.cs_connection_keys: Dict[str, List[str]] = {}
        self.cs_lconns: Dict[str, "LogicalConnection"] = {}

        self.cs_last_rebalance: Optional[float] = None
        self.cs_in_rebalance = False
        self.cs_rebal_scheduled = False
        self.cs_counters: Dict[str, int] = {}
        self.cs_last_error: Optional[BaseException] = None

        super().__init__("starting", loop=loop_opt)

        self.cs_rebal_timer = _IntervalTimer(self._loop, 10_000)
        shuffle_intvl = options.get("decoherenceInterval")
        if shuffle_intvl is None or shuffle_intvl < 60:
            shuffle_intvl = 60
        self.cs_shuffle_timer = _IntervalTimer(self._loop,
                                               shuffle_intvl * 1000)

    # -- resolver events -------------------------------------------------
    def _on_resolver_added(self, k: str, backend: Dict[str, Any]) -> None:
        backend["key"] = k
        if k in self.cs_keys:
            raise AssertionError("Resolver key is a duplicate")
        idx = random.randrange(len(self.cs_keys) + 1)
        self.cs_keys.insert(idx, k)
        self.cs_backends[k] = backend
        self.rebalance()

    def _on_resolver_removed(self, k: str) -> None:
        try:
            self.cs_keys.remove(k)
        except ValueError:
            raise AssertionError(
                "Resolver removed key that is not present in cs_keys")
        self.cs_backends.pop(k, None)
        self.cs_dead.pop(k, None)

        fsm = self.cs_fsm.get(k)
        if fsm is not None:
            fsm.set_unwanted()
        for ck in list(self.cs_connection_keys.get(k, ())):
            lconn = self.cs_lconns.get(ck)
            if lconn is not None and not lconn.is_in_state("stopped"):
                lconn.drain()

    def is_declared_dead(self, backend: str) -> bool:
        return self.cs_dead.get(backend) is True

    def should_retry_backend(self, backend: str) -> bool:
        return backend in self.cs_backends

    # -- states -----------------------------------------------------------
    def state_starting(self, S: StateScope) -> None:
        S.valid_transitions(["failed", "running", "stopping"])
        global_monitor.register_set(self)

        S.on(self.cs_resolver, "added", self._on_resolver_added)
        S.on(self.cs_resolver, "removed", self._on_resolver_removed)

        if self.cs_resolver.is_in_state("failed"):
            self.cs_log.warn('resolver has already failed, cset will start '
                             'up in "failed" state')
            self.cs_last_error = self.cs_resolver.get_last_error()
            S.goto_state("failed")
            return

        def on_res_state(st: str) -> None:
            if st == "failed":
                self.cs_log.warn('underlying resolver failed, moving cset '
                                 'to "failed" state')
                self.cs_last_error = self.cs_resolver.get_last_error()
                S.goto_state("failed")

        S.on(self.cs_resolver, "stateChanged", on_res_state)

        if self.cs_resolver.is_in_state("running"):
            for k, backend in self.cs_resolver.list().items():
                self._on_resolver_added(k, backend)

        S.on(self, "connectedToBackend", lambda *a: S.goto_state("running"))

        def on_closed_backend(*a: Any) -> None:
            dead = len(self.cs_dead)
            if dead >= len(self.cs_keys):
                self.cs_log.warn('cset has exhausted all retries, now '
                                 'moving to "failed" state', dead=dead)
                S.goto_state("failed")

        S.on(self, "closedBackend", on_closed_backend)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_failed(self, S: StateScope) -> None:
        S.valid_transitions(["running", "stopping"])
        S.on(self.cs_resolver, "added", self._on_resolver_added)
        S.on(self.cs_resolver, "removed", self._on_resolver_removed)
        S.on(self.cs_shuffle_timer, "timeout", self.reshuffle)

        def on_connected(*a: Any) -> None:
            if self.cs_resolver.is_in_state("failed"):
                raise AssertionError("resolver failed while cset recovering")
            self.cs_log.info("successfully connected to a backend, moving "
                             "back to running state")
            S.goto_state("running")

        S.on(self, "connectedToBackend", on_connected)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_running(self, S: StateScope) -> None:
        S.valid_transitions(["failed", "stopping"])
        S.on(self.cs_resolver, "added", self._on_resolver_added)
        S.on(self.cs_resolver, "removed", self._on_resolver_removed)
        S.on(self.cs_rebal_timer, "timeout", self.rebalance)
        S.on(self.cs_shuffle_timer, "timeout", self.reshuffle)

        def on_closed_backend(*a: Any) -> None:
            dead = len(self.cs_dead)
            if dead >= len(self.cs_keys):
                self.cs_log.warn('cset has exhausted all retries, now '
                                 'moving to "failed" state', dead=dead)
                S.goto_state("failed")

        S.on(self, "closedBackend", on_closed_backend)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_stopping(self, S: StateScope) -> None:
        S.valid_transitions(["stopped"])
        self.cs_backends = {}
        fsms = list(self.cs_fsm.values())
        remaining = {"n": len(fsms)}

        def one_done() -> None:
            remaining["n"] -= 1
            if remaining["n"] == 0:
                S.goto_state("stopped")

        if not fsms:
            S.goto_state("stopped")
            return

        for fsm in fsms:
            if fsm.is_in_state("stopped") or fsm.is_in_state("failed"):
                one_done()
                continue
            k = fsm.csf_backend["key"]
            cks = list(self.cs_connection_keys.get(k, ()))
            fired = {"done": False}

            def make_cb(f=fired):
                def cb(s: str) -> None:
                    if f["done"]:
                        return
                    if s in ("stopped", "failed"):
                        f["done"] = True
                        one_done()
                return cb

            # Scoped: auto-removed when the set leaves this state (see
            # the matching fix in pool.state_stopping_backends).
            S.on(fsm, "stateChanged", make_cb())
            fsm.set_unwanted()
            for ck in cks:
                # async: avoid FSM loops when .stop() was called from an
                # 'added' handler (lib/set.js:307-318)
                lconn = self.cs_lconns.get(ck)

                def drain_later(lc=lconn):
                    if lc is not None and not lc.is_in_state("stopped"):
                        lc.drain()

                self._loop.call_soon(drain_later)

    def state_stopped(self, S: StateScope) -> None:
        S.valid_transitions([])
        global_monitor.unregister_set(self)
        self.cs_keys = []
        self.cs_fsm = {}
        self.cs_connections = {}
        self.cs_backends = {}
        self.cs_rebal_timer.cancel()
        self.cs_shuffle_timer.cancel()

    # -- public API --------------------------------------------------------
    def reshuffle(self) -> None:
        if len(self.cs_keys) <= 1:
            return
        taken = self.cs_keys.pop()
        idx = random.randrange(len(self.cs_keys) + 1)
        if len(self.cs_keys) > self.cs_target and idx < self.cs_target:
            self.cs_log.info('random shuffle puts backend "%s" at idx %d',
                             taken, idx)
        self.cs_keys.insert(idx, taken)
        self.rebalance()

    def stop(self) -> None:
        self.emit("stopAsserted")

    def set_target(self, target: int) -> None:
        self.cs_target = target
        self.rebalance()

    def get_last_error(self) -> Optional[BaseException]:
        return self.cs_last_error

    def get_connections(self) -> List[Any]:
        """Currently-advertised connections (lib/set.js:613-623 intent;
        the reference's own implementation of this accessor is broken —
        it references fields that don't exist — so this returns what the
        docs promise: connections 'added' and not yet 'removed')."""
        return list(self.cs_connections.values())

    def get_stats(self) -> Dict[str, Any]:
        return {
            "counters": dict(self.cs_counters),
            "totalConnections": len(self.cs_fsm),
            "connections": len(self.cs_connections),
            "deadBackends": len(self.cs_dead),
        }

    # -- rebalancing --------------------------------------------------------
    def rebalance(self) -> None:
        if len(self.cs_keys) < 1:
            return
        if self.is_in_state("stopping") or self.is_in_state("stopped"):
            return
        if self.cs_rebal_scheduled:
            return
        self.cs_rebal_scheduled = True
        self._loop.call_soon(self._rebalance)

    def _rebalance(self) -> None:
        if self.cs_in_rebalance:
            return
        self.cs_in_rebalance = True
        self.cs_rebal_scheduled = False

        conns: Dict[str, List[ConnectionSlotFSM]] = {}
        total = 0
        working = 0
        for k in self.cs_keys:
            conns[k] = []
            fsm = self.cs_fsm.get(k)
            if fsm is not None:
                conns[k].append(fsm)
                if fsm.is_in_state("busy") or fsm.is_in_state("idle"):
                    working += 1
                total += 1

        plan = mod_utils.plan_rebalance(conns, self.cs_dead, self.cs_target,
                                        self.cs_max, singleton=True)

        if plan["remove"] or plan["add"]:
            self.cs_log.trace("rebalancing cset, remove %d, add %d "
                              "(target = %d, total = %d)",
                              len(plan["remove"]), len(plan["add"]),
                              self.cs_target, total)

        for fsm in plan["remove"]:
            # Never deliberately remove the last working connection: wait
            # for a replacement to come up first (lib/set.js:417-429).
            if (fsm.is_in_state("busy") or fsm.is_in_state("idle")) and \
                    working <= 1:
                continue
            k = fsm.csf_backend["key"]
            if fsm.is_in_state("busy") or fsm.is_in_state("idle"):
                working -= 1
            fsm.set_unwanted()
            if fsm.is_in_state("stopped") or fsm.is_in_state("failed"):
                self.cs_fsm.pop(k, None)
                total -= 1
            # drain any advertised connections from this FSM
            for ck in list(self.cs_connection_keys.get(k, ())):
                lconn = self.cs_lconns.get(ck)
                if lconn is not None and not lconn.is_in_state("stopped"):
                    lconn.drain()

        for k in plan["add"]:
            total += 1
            if total > self.cs_max + 1:
                continue
            if k in self.cs_fsm:  # never >1 slot per backend
                continue
            self.add_connection(k)

        self.cs_in_rebalance = False
        self.cs_last_rebalance = mod_time.time()

    def assert_emit(self, event: str, *args: Any) -> bool:
        """emit() that throws if nobody is listening — the added/removed
        contract is mandatory (lib/set.js:471-479)."""
        if self.listener_count(event) < 1:
            raise RuntimeError('Event "%s" on ConnectionSet must be handled'
                               % event)
        return self.emit(event, *args)

    def create_logi_conn(self, key: str) -> None:
        fsm = self.cs_fsm[key]
        if key not in self.cs_serials:
            self.cs_serials[key] = 1
        self.cs_connection_keys.setdefault(key, [])

        serial = self.cs_serials[key]
        self.cs_serials[key] += 1
        ckey = "%s.%d" % (key, serial)
        self.cs_connection_keys[key].append(ckey)

        lconn = LogicalConnection({
            "set": self,
            "log": self.cs_log,
            "key": key,
            "ckey": ckey,
            "fsm": fsm,
            "loop": self._loop,
        })
        self.cs_lconns[ckey] = lconn

        def on_lconn_state(st: str) -> None:
            if st != "stopped":
                return
            self.cs_lconns.pop(ckey, None)
            cks = self.cs_connection_keys.get(key, [])
            if ckey in cks:
                cks.remove(ckey)
            # Chain the next serial if this slot will contribute another
            # connection (lib/set.js:514-533).
            if key not in self.cs_backends:
                return
            if fsm.is_in_state("failed") or fsm.is_in_state("stopped"):
                return
            self.create_logi_conn(key)

        lconn.on("stateChanged", on_lconn_state)

    def add_connection(self, key: str) -> None:
        if self.is_in_state("stopping") or self.is_in_state("stopped"):
            return

        backend = self.cs_backends[key]
        backend["key"] = key

        fsm = ConnectionSlotFSM({
            "constructor": self.cs_constructor,
            "backend": backend,
            "log": self.cs_log,
            "pool": self,
            "recovery": self.cs_recovery,
            "monitor": self.cs_dead.get(key) is True,
            "loop": self._loop,
        })
        if key in self.cs_fsm:
            raise AssertionError("slot for %s already exists" % key)
        self.cs_fsm[key] = fsm

        self.create_logi_conn(key)

        # rebalance when a slot reaches idle or leaves it — those are the
        # points where the plan can meaningfully change (lib/set.js:559-565)
        was_idle = {"v": False}

        def on_slot_state(new_state: str) -> None:
            if new_state == "idle":
                self.emit("connectedToBackend", key, fsm)
                if key in self.cs_dead:
                    del self.cs_dead[key]
                self.rebalance()
                was_idle["v"] = True
                return

            if was_idle["v"]:
                was_idle["v"] = False
                self.rebalance()

            if new_state == "failed":
                if key in self.cs_backends:
                    self.cs_dead[key] = True
                    err = fsm.get_socket_mgr().get_last_error()
                    if err is not None:
                        self.cs_last_error = err

            if new_state in ("stopped", "failed"):
                self.cs_fsm.pop(key, None)
                self.emit("closedBackend", fsm)
                self.rebalance()

        fsm.on("stateChanged", on_slot_state)
        fsm.start()

    def _incr_counter(self, counter: str) -> None:
        mod_utils.update_error_metrics(self.cs_collector, self.cs_uuid,
                                       counter)
        self.cs_counters[counter] = self.cs_counters.get(counter, 0) + 1


class LogicalConnection(FSM):
    """Tracks one connection key through its advertised lifetime
    (lib/set.js:632-820): init -> advertised -> draining -> stopped.
    """

    def __init__(self, options: Dict[str, Any]) -> None:
        self.lc_set: ConnectionSet = options["set"]
        self.lc_key: str = options["key"]
        self.lc_fsm: ConnectionSlotFSM = options["fsm"]
        self.lc_smgr = options["fsm"].get_socket_mgr()
        self.lc_conn: Any = None
        self.lc_ckey: str = options["ckey"]
        self.lc_hdl: Optional[ClaimHandle] = None
        self.lc_log: CueballLogger = options["log"]
        super().__init__("init", loop=options.get("loop"))

    def drain(self) -> None:
        if self.is_in_state("stopped"):
            raise AssertionError("drain() on stopped LogicalConnection")
        self.emit("drainAsserted")

    def state_init(self, S: StateScope) -> None:
        S.valid_transitions(["advertised", "stopped"])

        def on_claimed(err, hdl=None, conn=None):
            if err is not None:
                raise AssertionError("cset claim handle failed: %r" % err)
            if hdl is not self.lc_hdl:
                raise AssertionError("claimed foreign handle")
            self.lc_conn = conn
            S.goto_state("advertised")

        self.lc_hdl = ClaimHandle({
            "pool": self.lc_set,
            "claimStack": ["claim", "ConnectionSet.add_connection",
                           "ConnectionSet.add_connection"],
            "callback": S.callback(on_claimed),
            "log": self.lc_log,
            "throwError": not self.lc_set.cs_conn_handles_err,
            "claimTimeout": math.inf,
            "loop": self._loop,
        })

        # Keep trying the slot until we get a claim; multiple attempts in
        # this state are fine — 'added' has not been emitted yet.
        def on_hdl_state(st: str) -> None:
            if st == "waiting" and self.lc_hdl.is_in_state("waiting"):
                if self.lc_fsm.is_in_state("idle"):
                    self.lc_hdl.try_(self.lc_fsm)
            elif st in ("failed", "cancelled"):
                S.goto_state("stopped")

        S.on(self.lc_hdl, "stateChanged", on_hdl_state)

        def on_fsm_state(st: str) -> None:
            if st == "idle" and self.lc_fsm.is_in_state("idle"):
                if self.lc_hdl.is_in_state("waiting"):
                    self.lc_hdl.try_(self.lc_fsm)
            elif st == "failed":
                S.goto_state("stopped")

        S.on(self.lc_fsm, "stateChanged", on_fsm_state)

        # drain before advertising: go straight to stopped
        S.on(self, "drainAsserted", lambda: S.goto_state("stopped"))

    def state_advertised(self, S: StateScope) -> None:
        S.valid_transitions(["draining", "stopped"])

        def on_hdl_state(st: str) -> None:
            # user may .close() at any time, but .release() only after
            # 'removed' has been emitted (docs/api.adoc; lib/set.js:760-773)
            if st == "closed":
                S.goto_state("stopped")
            if st == "released":
                raise RuntimeError(
                    "The .release() method may not be called on a "
                    'ConnectionSet handle before "removed" has been emitted')

        S.on(self.lc_hdl, "stateChanged", on_hdl_state)

        def on_smgr_state(st: str) -> None:
            if st != "connected":
                S.goto_state("draining")

        S.on(self.lc_smgr, "stateChanged", on_smgr_state)
        S.on(self, "drainAsserted", lambda: S.goto_state("draining"))

        self.lc_set.cs_connections[self.lc_ckey] = self.lc_conn
        self.lc_set.assert_emit("added", self.lc_ckey, self.lc_conn,
                                self.lc_hdl)

    def state_draining(self, S: StateScope) -> None:
        S.valid_transitions(["stopped"])
        self.lc_set.cs_connections.pop(self.lc_ckey, None)

        hdl = self.lc_hdl
        if not hdl.is_in_state("claimed"):
            # Divergence from the reference (bug fix, property suite):
            # the user already relinquished the handle (e.g. close()
            # racing the socket's own death event, both pending in the
            # same spin).  Emitting 'removed' now would hand the
            # consumer a dead handle whose release() throws
            # (lib/set.js:792-811 emits unconditionally).  The
            # connection is already gone: stop directly.
            S.goto_state("stopped")
            return

        def on_hdl_state(st: str) -> None:
            if st in ("closed", "released", "cancelled"):
                S.goto_state("stopped")

        S.on(hdl, "stateChanged", on_hdl_state)
        self.lc_set.assert_emit("removed", self.lc_ckey, self.lc_conn,
                                hdl)

    def state_stopped(self, S: StateScope) -> None:
        S.valid_transitions([])
        self.lc_set.cs_connections.pop(self.lc_ckey, None)
        if self.lc_hdl is not None and (
                self.lc_hdl.is_in_state("waiting")
                or self.lc_hdl.is_in_state("claiming")):
            self.lc_hdl.cancel()
