"""ConnectionPool: claim/release pooling over resolver-discovered backends.

Re-design of reference lib/pool.js.  The pool owns one ConnectionSlotFSM
per connection, keeps idle/init/waiter queues (intrusive, O(1) unlink),
and continuously rebalances towards `spares` idle connections spread
evenly over the live backends, clamped by an EMA low-pass filter so the
pool does not shrink against recent load transients
(lib/pool.js:37-100), rate-limited by `maxChurnRate`, and periodically
"decoherence-shuffled" so that many clients don't converge on the same
backend ordering (lib/pool.js:501-519, docs/internals.adoc:276-386).

States: starting -> running <-> failed ; stopping -> stopping.backends
-> stopped (lib/pool.js:315-487).
"""

from __future__ import annotations

import functools
import math
import os as _os
import random
import time as mod_time
import uuid as mod_uuid
from typing import Any, Callable, Dict, List, Optional

from . import codel as mod_codel
from . import errors as mod_errors
from . import utils as mod_utils
from .connection_fsm import ClaimHandle, ConnectionSlotFSM
from .events import EventEmitter
from .fsm import FSM, StateScope
from .logutil import CueballLogger, default_logger
from .pool_monitor import monitor as global_monitor
from .queue import Queue

__all__ = ["ConnectionPool", "FIRFilter", "gen_taps"]

# EMA/low-pass filter parameters (lib/pool.js:37-62): 5 Hz sampling,
# 128-tap EMA with time constant -0.2 => pass band ~0.25 Hz.
LP_RATE = 5
LP_INT_MS = round(1000.0 / LP_RATE)


def gen_taps(count: int, tc: float) -> List[float]:
    taps = [math.exp(tc * i) for i in range(count)]
    s = sum(taps)
    return [t / s for t in taps]


LP_TAPS = gen_taps(128, -0.2)


class FIRFilter:
    """Simple FIR filter over a circular buffer (lib/pool.js:77-100)."""

    __slots__ = ("f_taps", "f_buf", "f_ptr")

    def __init__(self, taps: List[float]) -> None:
        self.f_taps = taps
        self.f_buf = [0.0] * len(taps)
        self.f_ptr = 0

    def put(self, v: float) -> None:
        self.f_buf[self.f_ptr] = v
        self.f_ptr += 1
        if self.f_ptr == len(self.f_taps):
            self.f_ptr = 0

    def get(self) -> float:
        i = self.f_ptr - 1
        n = len(self.f_taps)
        if i < 0:
            i += n
        acc = 0.0
        buf = self.f_buf
        for tap in self.f_taps:
            acc += buf[i] * tap
            i -= 1
            if i < 0:
                i += n
        return acc


class _ClaimTicket:
    """Per-claim driver: retries the claim whenever the handle returns
    to 'waiting' (registered as the handle's stateChanged listener; the
    closure-free form of lib/pool.js:922-968, on the claim hot path).

    When the native core is loaded, ``_speed.ClaimTicket`` replaces
    this class on the claim path: it performs the idle-queue handoff
    entirely in C and delegates everything else to the same
    ``pool._ticket_slow`` this class uses."""

    __slots__ = ("pool", "handle", "err_on_empty")

    #: lets the handle's terminal cleanup unregister us (see
    #: ClaimHandle._fsm_terminal_settled)
    _cueball_ticket = True

    def __init__(self, pool: "ConnectionPool", handle: ClaimHandle,
                 err_on_empty: bool) -> None:
        self.pool = pool
        self.handle = handle
        self.err_on_empty = err_on_empty

    def __call__(self, st: str) -> None:
        if st == "waiting":
            self.pool._ticket_slow(self.handle, self.err_on_empty)


_NativeClaimTicket = None
_native_claim_fast = None
_native_pool_claim = None
_NativeSlotDispatch = None
if not _os.environ.get("CUEBALL_PURE"):
    try:
        from ._speed import ClaimTicket as _NativeClaimTicket  # noqa: F811
        from ._speed import claim_fast as _native_claim_fast
        from ._speed import pool_claim as _native_pool_claim
        from ._speed import SlotDispatch as _NativeSlotDispatch
    except ImportError:
        pass


class _CancelStub:
    """claim() return value when the pool is stopping/failed: supports
    only .cancel() (lib/pool.js:895-897)."""

    __slots__ = ("_state",)

    def __init__(self, state):
        self._state = state

    def cancel(self) -> None:
        self._state["done"] = True


class _IntervalTimer(EventEmitter):
    """An EventEmitter that fires 'timeout' every `ms` (the pool's FSM
    states subscribe/unsubscribe to these with their state scope, like
    the reference's unref()'d setInterval wrappers, lib/pool.js:228-263).
    """

    def __init__(self, loop, ms: float) -> None:
        super().__init__()
        self._loop = loop
        self._ms = ms
        self._handle = None
        self._stopped = False
        self._schedule()

    def _schedule(self) -> None:
        self._handle = self._loop.call_later(self._ms / 1000.0, self._fire)

    def _fire(self) -> None:
        if self._stopped:
            return
        self._schedule()
        self.emit("timeout")

    def cancel(self) -> None:
        self._stopped = True
        if self._handle is not None:
            self._handle.cancel()


class ConnectionPool(FSM):
    def __init__(self, options: Dict[str, Any]) -> None:
        if not callable(options.get("constructor")):
            raise TypeError("options.constructor (callable) is required")

        loop_opt = options.get("loop")
        self.p_uuid = str(mod_uuid.uuid4())
        self.p_constructor = options["constructor"]

        if not isinstance(options.get("domain"), str):
            raise TypeError("options.domain (string) is required")
        self.p_domain = options["domain"]
        mod_utils.assert_claim_delay(options.get("targetClaimDelay"))

        recovery = options.get("recovery")
        mod_utils.assert_recovery_set(recovery)
        self.p_recovery = recovery

        log: CueballLogger = options.get("log") or default_logger()
        self.p_log = log.child(
            component="ConnectionPool",
            domain=options.get("domain"),
            service=options.get("service"),
            pool=self.p_uuid,
        )
        # shared child logger for claim handles (hot path)
        self.p_claim_log = self.p_log.child(component="ClaimHandle")

        self.p_collector = mod_utils.create_error_metrics(options)
        try:
            self._gauge = self.p_collector.gauge(
                name="cueball_pool_connections",
                help="Live connection counts per pool and state")
            self._gauge_labels = {"pool": self.p_uuid.split("-")[0],
                                  "domain": options["domain"]}
        except (AttributeError, ValueError):
            self._gauge = None  # foreign collector without gauges
            self._gauge_labels = {}

        spares = options.get("spares")
        maximum = options.get("maximum")
        if not isinstance(spares, int) or not isinstance(maximum, int):
            raise TypeError("options.spares and options.maximum are required")
        self.p_spares = spares
        self.p_max = maximum

        self.p_checker = options.get("checker")
        self.p_check_timeout = options.get("checkTimeout")

        self.p_keys: List[str] = []
        self.p_backends: Dict[str, Dict[str, Any]] = {}
        self.p_connections: Dict[str, List[ConnectionSlotFSM]] = {}
        self.p_dead: Dict[str, bool] = {}
        self.p_lastrate: Dict[str, Dict[str, float]] = {}

        max_churn = options.get("maxChurnRate")
        self.p_maxrate = float(max_churn) if max_churn is not None else math.inf

        self.p_last_rebalance: Optional[float] = None
        self.p_in_rebalance = False
        self.p_rebal_scheduled = False
        self.p_started_resolver = False
        self.p_lpf = FIRFilter(LP_TAPS)
        self.p_last_rebal_clamped = False
        self.p_rate_delay_timer = None
        # Census signature of the last no-op rebalance: plan_rebalance
        # is a pure function of (key order, per-key counts, dead set,
        # target, max), so if none of those changed since a rebalance
        # that planned nothing, the plan is empty again and we can skip
        # building it.  Claim-heavy steady state schedules a rebalance
        # every loop turn (each idle->busy claim does), making this the
        # hot no-op path.
        self.p_noop_census: Optional[tuple] = None
        # Peak demand (busy + extras) and peak busy seen since the last
        # rebalance / LPF sample.  The reference sizes from an
        # instantaneous sample taken when its setImmediate/timer fires
        # (lib/pool.js:560-588, :251-263); with this runtime's batched
        # event dispatch those callbacks tend to run at quiescent
        # points between claim bursts, where busy == waiters == 0, so
        # an instantaneous sample would chronically under-size the
        # pool.  Tracking the high-water mark on the claim path sizes
        # for the peak concurrent demand of the interval instead,
        # which is the reference's intent.
        self.p_demand_hwm = 0
        self.p_busy_hwm = 0
        self.p_total_conns = 0

        self.p_idleq = Queue()
        self.p_initq = Queue()
        self.p_waiters = Queue()

        self.p_codel: Optional[mod_codel.ControlledDelay] = None
        tcd = options.get("targetClaimDelay")
        if tcd is not None and math.isfinite(tcd):
            self.p_codel = mod_codel.ControlledDelay(tcd, loop=loop_opt)

        self.p_last_error: Optional[BaseException] = None
        self.p_counters: Dict[str, int] = {}

        if options.get("resolver") is not None:
            self.p_resolver = options["resolver"]
            self.p_resolver_custom = True
        else:
            from . import resolver as mod_resolver
            self.p_resolver = mod_resolver.Resolver({
                "resolvers": options.get("resolvers"),
                "domain": options["domain"],
                "service": options.get("service"),
                "maxDNSConcurrency": options.get("maxDNSConcurrency"),
                "defaultPort": options.get("defaultPort"),
                "log": self.p_log,
                "recovery": recovery,
                "loop": loop_opt,
            })
            self.p_resolver_custom = False

        super().__init__("starting", loop=loop_opt)

        # Periodic timers.  Created after FSM init so self._loop exists;
        # states subscribe to them scoped (lib/pool.js:228-263).
        self.p_rebal_timer = _IntervalTimer(self._loop, 10_000)
        shuffle_intvl = options.get("decoherenceInterval")
        if shuffle_intvl is None or shuffle_intvl < 60:
            shuffle_intvl = 60
        self.p_shuffle_timer = _IntervalTimer(self._loop, shuffle_intvl * 1000)
        self.p_lp_timer = _IntervalTimer(self._loop, LP_INT_MS)
        self.p_lp_timer.on("timeout", self._lp_tick)

    # -- counters -------------------------------------------------------
    def _incr_counter(self, counter: str) -> None:
        # only tracked error events reach the metrics collector; the
        # hot counters (claim, queued-claim) skip the call entirely
        if counter in mod_utils.TRACKED_ERROR_EVENTS:
            mod_utils.update_error_metrics(self.p_collector, self.p_uuid,
                                           counter)
        self.p_counters[counter] = self.p_counters.get(counter, 0) + 1

    def _hwm_counter(self, counter: str, val: int) -> None:
        if self.p_counters.get(counter, -1) < val:
            self.p_counters[counter] = val

    def _note_demand(self) -> None:
        """O(1) demand sample on the claim path: fold the current
        busy/extras into the high-water marks consumed by _rebalance
        and _lp_tick (see p_demand_hwm comment in __init__)."""
        nw = self.p_waiters._len
        ni = self.p_initq._len
        spares = self.p_idleq._len + ni - nw
        if spares < 0:
            spares = 0
        busy = self.p_total_conns - spares
        if busy < 0:
            busy = 0
        extras = nw - ni
        if extras < 0:
            extras = 0
        if busy > self.p_busy_hwm:
            self.p_busy_hwm = busy
        if busy + extras > self.p_demand_hwm:
            self.p_demand_hwm = busy + extras

    # -- LPF anti-shrink sampling (lib/pool.js:251-263) ------------------
    def _lp_tick(self) -> None:
        conns = sum(len(v) for v in self.p_connections.values())
        spares = len(self.p_idleq) + len(self.p_initq)
        busy = conns - spares
        if busy < self.p_busy_hwm:
            busy = self.p_busy_hwm
        self.p_busy_hwm = conns - spares if conns > spares else 0
        self.p_lpf.put(busy + self.p_spares)
        # piggyback live gauges on the 5 Hz sample (beyond the
        # reference, which exposes state through kang only)
        g = self._gauge
        if g is not None:
            labels = self._gauge_labels
            g.set(conns, {**labels, "state": "total"})
            g.set(len(self.p_idleq), {**labels, "state": "idle"})
            g.set(len(self.p_initq), {**labels, "state": "pending"})
            g.set(len(self.p_waiters), {**labels, "state": "waiting"})
        if self.p_last_rebal_clamped:
            self.rebalance()

    # -- resolver events ------------------------------------------------
    def _on_resolver_added(self, k: str, backend: Dict[str, Any]) -> None:
        backend["key"] = k
        idx = random.randrange(len(self.p_keys) + 1)
        self.p_keys.insert(idx, k)
        self.p_backends[k] = backend
        self.rebalance()

    def _on_resolver_removed(self, k: str) -> None:
        try:
            self.p_keys.remove(k)
        except ValueError:
            raise AssertionError("resolver key %s not found" % k)
        self.p_backends.pop(k, None)
        self.p_dead.pop(k, None)
        # Slots unlink themselves (and rebalance) from the stateChanged
        # handler in add_connection once they stop; here we only flag
        # them unwanted (lib/pool.js:300-313).
        for fsm in list(self.p_connections.get(k, ())):
            fsm.set_unwanted()

    # -- states ----------------------------------------------------------
    def state_starting(self, S: StateScope) -> None:
        S.valid_transitions(["failed", "running", "stopping"])
        global_monitor.register_pool(self)

        S.on(self.p_resolver, "added", self._on_resolver_added)
        S.on(self.p_resolver, "removed", self._on_resolver_removed)

        if self.p_resolver.is_in_state("failed"):
            self.p_log.warn('pre-provided resolver has already failed, '
                            'pool will start up in "failed" state')
            self.p_last_error = mod_errors.CueballError(
                'Pool resolver entered state "failed"',
                self.p_resolver.get_last_error())
            S.goto_state("failed")
            return

        def on_res_state(state: str) -> None:
            if state == "failed":
                self.p_log.warn('underlying resolver failed, moving pool '
                                'to "failed" state')
                self.p_last_error = mod_errors.CueballError(
                    'Pool resolver entered state "failed"',
                    self.p_resolver.get_last_error())
                S.goto_state("failed")

        S.on(self.p_resolver, "stateChanged", on_res_state)

        if self.p_resolver.is_in_state("running"):
            for k, backend in self.p_resolver.list().items():
                self._on_resolver_added(k, backend)
        elif self.p_resolver.is_in_state("stopped") and \
                not self.p_resolver_custom:
            self.p_resolver.start()
            self.p_started_resolver = True

        S.on(self, "connectedToBackend", lambda *a: S.goto_state("running"))

        def on_closed_backend(*a: Any) -> None:
            dead = len(self.p_dead)
            self._hwm_counter("max-dead-backends", dead)
            if dead >= len(self.p_keys):
                self.p_log.warn("pool has exhausted all retries, now moving "
                                'to "failed" state', dead=dead)
                S.goto_state("failed")

        S.on(self, "closedBackend", on_closed_backend)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_failed(self, S: StateScope) -> None:
        S.valid_transitions(["running", "stopping"])
        S.on(self.p_resolver, "added", self._on_resolver_added)
        S.on(self.p_resolver, "removed", self._on_resolver_removed)
        S.on(self.p_shuffle_timer, "timeout", self.reshuffle)

        def on_connected(*a: Any) -> None:
            if self.p_resolver.is_in_state("failed"):
                raise AssertionError("resolver failed while pool recovering")
            self.p_log.info("successfully connected to a backend, moving "
                            "back to running state")
            S.goto_state("running")

        S.on(self, "connectedToBackend", on_connected)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

        self._incr_counter("failed-state")

        # Fail all outstanding claims that wait for a connection.
        while not self.p_waiters.is_empty():
            hdl = self.p_waiters.shift()
            if hdl.is_in_state("waiting"):
                hdl.fail(mod_errors.PoolFailedError(self, self.p_last_error))

    def state_running(self, S: StateScope) -> None:
        S.valid_transitions(["failed", "stopping"])
        S.on(self.p_resolver, "added", self._on_resolver_added)
        S.on(self.p_resolver, "removed", self._on_resolver_removed)
        S.on(self.p_rebal_timer, "timeout", self.rebalance)
        S.on(self.p_shuffle_timer, "timeout", self.reshuffle)

        def on_closed_backend(*a: Any) -> None:
            dead = len(self.p_dead)
            self._hwm_counter("max-dead-backends", dead)
            if dead >= len(self.p_keys):
                self.p_log.warn("pool has exhausted all retries, now moving "
                                'to "failed" state', dead=dead)
                S.goto_state("failed")

        S.on(self, "closedBackend", on_closed_backend)
        S.on(self, "stopAsserted", lambda: S.goto_state("stopping"))

    def state_stopping(self, S: StateScope) -> None:
        S.valid_transitions(["stopping.backends"])
        # Divergence from the reference (bug fix): fail queued waiters
        # with PoolStoppingError instead of leaving them pending forever
        # (the reference's state_stopping never touches p_waiters, so an
        # infinite-timeout claim outstanding at stop() hangs;
        # lib/pool.js:433-448 vs the failed-state drain at :398-405).
        while not self.p_waiters.is_empty():
            hdl = self.p_waiters.shift()
            if hdl.is_in_state("waiting"):
                hdl.fail(mod_errors.PoolStoppingError(self))
        if self.p_started_resolver:
            def on_res_state(s: str) -> None:
                if s == "stopped":
                    S.goto_state("stopping.backends")

            S.on(self.p_resolver, "stateChanged", on_res_state)
            self.p_resolver.stop()
            if self.p_resolver.is_in_state("stopped"):
                S.goto_state("stopping.backends")
        else:
            S.goto_state("stopping.backends")

    def state_stopping_backends(self, S: StateScope) -> None:
        S.valid_transitions(["stopped"])
        fsms: List[ConnectionSlotFSM] = []
        for conns in self.p_connections.values():
            fsms.extend(conns)

        remaining = {"n": len(fsms)}

        def one_done() -> None:
            remaining["n"] -= 1
            if remaining["n"] == 0:
                S.goto_state("stopped")

        if not fsms:
            S.goto_state("stopped")
            return

        for fsm in fsms:
            fsm.set_unwanted()
            if fsm.is_in_state("stopped") or fsm.is_in_state("failed"):
                one_done()
            else:
                def make_cb():
                    fired = {"done": False}

                    def cb(st: str) -> None:
                        if fired["done"]:
                            return
                        if st in ("stopped", "failed"):
                            fired["done"] = True
                            one_done()
                    return cb

                # Scoped: auto-removed from the slot FSM when the pool
                # leaves stopping.backends, so repeated stop cycles (or
                # slots that outlive the pool) cannot accumulate
                # listeners (round-1 review finding).
                S.on(fsm, "stateChanged", make_cb())

    def state_stopped(self, S: StateScope) -> None:
        S.valid_transitions([])
        global_monitor.unregister_pool(self)
        self.p_keys = []
        self.p_connections = {}
        self.p_backends = {}
        self.p_rebal_timer.cancel()
        self.p_shuffle_timer.cancel()
        self.p_lp_timer.cancel()
        if self.p_rate_delay_timer is not None:
            self.p_rate_delay_timer.cancel()

    # -- public helpers --------------------------------------------------
    def should_retry_backend(self, backend: str) -> bool:
        return backend in self.p_backends

    def is_declared_dead(self, backend: str) -> bool:
        return self.p_dead.get(backend) is True

    def get_last_error(self) -> Optional[BaseException]:
        return self.p_last_error

    def stop(self) -> None:
        self.emit("stopAsserted")

    def reshuffle(self) -> None:
        """Decoherence shuffle: move the last preference-list entry to a
        random position (lib/pool.js:501-519)."""
        if len(self.p_keys) <= 1:
            return
        taken = self.p_keys.pop()
        idx = random.randrange(len(self.p_keys) + 1)
        conns = sum(len(v) for v in self.p_connections.values())
        if len(self.p_keys) > conns and idx < conns:
            self.p_log.info('random shuffle puts backend "%s" at idx %d',
                            taken, idx)
        self.p_keys.insert(idx, taken)
        self.rebalance()

    # -- rebalancing ------------------------------------------------------
    def rebalance(self) -> None:
        if len(self.p_keys) < 1:
            return
        if self.is_in_state("stopping") or self.is_in_state("stopped"):
            return
        if self.p_rebal_scheduled:
            return
        self.p_rebal_scheduled = True
        self._loop.call_soon(self._rebalance)

    def _rebalance(self) -> None:
        if self.p_in_rebalance:
            return
        self.p_in_rebalance = True
        self.p_rebal_scheduled = False

        total = 0
        counts = []
        for k in self.p_keys:
            n = len(self.p_connections.get(k, ()))
            counts.append(n)
            total += n
        spares = len(self.p_idleq) + len(self.p_initq) - len(self.p_waiters)
        if spares < 0:
            spares = 0
        busy = total - spares
        if busy < 0:
            busy = 0
        extras = len(self.p_waiters) - len(self.p_initq)
        if extras < 0:
            extras = 0

        # Size for the peak demand of the interval, not just this
        # instant (see p_demand_hwm comment in __init__).
        demand = busy + extras
        if demand < self.p_demand_hwm:
            demand = self.p_demand_hwm
        self.p_demand_hwm = busy + extras

        target = demand + self.p_spares

        # Anti-shrink clamp from the low-pass filter (lib/pool.js:579-588)
        lpf_min = math.ceil(self.p_lpf.get())
        if target < lpf_min * 1.05:
            target = lpf_min
            self.p_last_rebal_clamped = True
        else:
            self.p_last_rebal_clamped = False

        if target > self.p_max:
            target = self.p_max

        census = (target, tuple(counts), tuple(self.p_keys),
                  tuple(sorted(self.p_dead)))
        if census == self.p_noop_census:
            self.p_in_rebalance = False
            self.p_last_rebalance = mod_time.time()
            return

        conns: Dict[str, List[ConnectionSlotFSM]] = {}
        for k in self.p_keys:
            conns[k] = list(self.p_connections.get(k, ()))

        plan = mod_utils.plan_rebalance(conns, self.p_dead, target, self.p_max)
        self.p_noop_census = None if (plan["remove"] or plan["add"]) \
            else census

        if plan["remove"] or plan["add"]:
            self.p_log.trace(
                "rebalancing pool, remove %d, add %d (busy = %d, spares = "
                "%d, target = %d)", len(plan["remove"]), len(plan["add"]),
                busy, spares, target)

        now = self._loop.time()
        rate_delay: Optional[float] = None

        for fsm in plan["remove"]:
            k = fsm.get_backend()["key"]
            lastrate = self.p_lastrate.get(k)
            n = len(self.p_connections.get(k, ())) - 1
            if lastrate:
                tdelta = now - lastrate["time"]
                ndelta = n - lastrate["count"]
                rate = abs(ndelta / tdelta) if tdelta else math.inf
                if rate > self.p_maxrate:
                    tnext = lastrate["time"] + abs(ndelta) / self.p_maxrate
                    delay = tnext - now
                    if rate_delay is None or delay < rate_delay:
                        rate_delay = delay
                    continue
            self.p_lastrate[k] = {"time": now, "count": n}

            fsm.set_unwanted()
            # The slot may have gone stopped/failed synchronously; if so
            # it no longer counts against the cap (lib/pool.js:623-631).
            if fsm.is_in_state("stopped") or fsm.is_in_state("failed"):
                total -= 1

        for k in plan["add"]:
            lastrate = self.p_lastrate.get(k)
            n = len(self.p_connections.get(k, ())) + 1
            if lastrate:
                tdelta = now - lastrate["time"]
                ndelta = n - lastrate["count"]
                rate = abs(ndelta / tdelta) if tdelta else math.inf
                if rate > self.p_maxrate:
                    tnext = lastrate["time"] + abs(ndelta) / self.p_maxrate
                    delay = tnext - now
                    if rate_delay is None or delay < rate_delay:
                        rate_delay = delay
                    continue
            self.p_lastrate[k] = {"time": now, "count": n}
            total += 1
            if total > self.p_max:  # never exceed the socket limit
                continue
            self.add_connection(k)

        if rate_delay is not None:
            if self.p_rate_delay_timer is not None:
                self.p_rate_delay_timer.cancel()
            self.p_rate_delay_timer = self._loop.call_later(
                rate_delay + 0.01, self.rebalance)

        self.p_in_rebalance = False
        self.p_last_rebalance = mod_time.time()

    # -- slot lifecycle ----------------------------------------------------
    def add_connection(self, key: str) -> None:
        if self.is_in_state("stopping") or self.is_in_state("stopped"):
            return

        backend = self.p_backends[key]
        backend["key"] = key

        fsm = ConnectionSlotFSM({
            "constructor": self.p_constructor,
            "backend": backend,
            "log": self.p_log,
            "pool": self,
            "checker": self.p_checker,
            "checkTimeout": self.p_check_timeout,
            "recovery": self.p_recovery,
            "monitor": self.p_dead.get(key) is True,
            "loop": self._loop,
        })
        self.p_connections.setdefault(key, []).append(fsm)
        self.p_total_conns += 1

        fsm.p_initq_node = self.p_initq.push(fsm)
        fsm.p_idleq_node = None

        # Native dispatcher when available (idle-feed/busy fast paths
        # in C, everything else falls back to _slot_state_changed);
        # else functools.partial, which dispatches at C level.
        if _NativeSlotDispatch is not None:
            fsm.on("stateChanged", _NativeSlotDispatch(
                self, fsm, key, self.p_codel is not None))
        else:
            fsm.on("stateChanged",
                   functools.partial(self._slot_state_changed, fsm, key))
        fsm.start()

    def _slot_state_changed(self, fsm: ConnectionSlotFSM, key: str,
                            new_state: str) -> None:
        """The big per-slot dispatcher (lib/pool.js:692-807)."""
        if fsm.p_initq_node is not None:
            if new_state in ("init", "connecting", "retrying"):
                return  # still starting up
            fsm.p_initq_node.remove()
            fsm.p_initq_node = None

        if new_state == "idle":
            self.emit("connectedToBackend", key, fsm)
            if key in self.p_dead:
                del self.p_dead[key]
                self.rebalance()

            if fsm._fsm_state == "idle":
                # Just became available: either released by its user
                # or done connecting.
                if key not in self.p_backends:
                    fsm.set_unwanted()
                    return

                # Feed waiters, with the CoDel drop check on each.
                while self.p_waiters._len > 0:
                    hdl = self.p_waiters.shift()
                    drop = self.p_codel is not None and \
                        self.p_codel.overloaded(hdl.ch_started)
                    if not hdl.is_in_state("waiting"):
                        continue
                    if drop:
                        hdl.timeout()
                        continue
                    hdl.try_(fsm)
                    return

                if self.p_codel is not None:
                    self.p_codel.empty()

                fsm.p_idleq_node = self.p_idleq.push(fsm)
                return

        elif new_state == "busy":
            # Health-checking connections ride the initq so they don't
            # count as busy (lib/pool.js:762-769); inline of
            # fsm.is_running_ping() with the cheap checks first
            hdl = fsm.csf_handle
            if hdl is not None and fsm.p_initq_node is None and \
                    hdl.ch_pinger and fsm._fsm_state == "busy":
                fsm.p_initq_node = self.p_initq.push(fsm)

        elif new_state == "failed":
            if key in self.p_backends:
                self.p_dead[key] = True
            err = fsm.get_socket_mgr().get_last_error()
            if err is not None:
                self.p_last_error = err

        if new_state in ("stopped", "failed"):
            conns = self.p_connections.get(key)
            if conns is not None:
                conns.remove(fsm)
                self.p_total_conns -= 1
                if not conns:
                    del self.p_connections[key]
            self.emit("closedBackend", key, fsm)
            self.rebalance()

        if fsm.p_idleq_node is not None:
            # Was idle, now isn't: unlink and rebalance in case we were
            # closed or died.
            fsm.p_idleq_node.remove()
            fsm.p_idleq_node = None
            self.rebalance()

    def print_connections(self) -> None:
        """Debugging helper: dump live connection states and dead set
        (lib/pool.js:812-832)."""
        obj: Dict[str, Dict[str, int]] = {}
        ks = list(self.p_keys)
        for k in self.p_connections.keys():
            if k not in ks:
                ks.append(k)
        for k in ks:
            per_state: Dict[str, int] = {}
            for fsm in self.p_connections.get(k, ()):
                s = fsm.get_state()
                per_state[s] = per_state.get(s, 0) + 1
            obj[k] = per_state
        print("live:", obj)
        print("dead:", dict(self.p_dead))

    # -- stats / claim ------------------------------------------------------
    def get_stats(self) -> Dict[str, Any]:
        tconns = sum(len(v) for v in self.p_connections.values())
        return {
            "counters": dict(self.p_counters),
            "totalConnections": tconns,
            "idleConnections": len(self.p_idleq),
            "pendingConnections": len(self.p_initq),
            "waiterCount": len(self.p_waiters),
        }

    def _ticket_slow(self, handle: ClaimHandle,
                     err_on_empty: bool) -> None:
        """The full claim-retry logic (lib/pool.js:922-968): invoked on
        every handle return to 'waiting' — directly in pure mode, and
        as the native ClaimTicket's off-hot-path fallback (pool not in
        'running', or the idle queue empty)."""
        if not handle.is_in_state("waiting"):
            return
        # The first try runs on the next loop turn; the pool may have
        # started stopping (or failed) in between — fail now rather
        # than queueing a waiter nothing will ever feed (companion to
        # the stopping-state waiter drain).
        if self.is_in_state("stopping") or self.is_in_state("stopped"):
            handle.fail(mod_errors.PoolStoppingError(self))
            return
        if self.is_in_state("failed"):
            handle.fail(mod_errors.PoolFailedError(
                self, self.p_last_error))
            return
        # Idle connections sitting around?  Take one.  Entries may be
        # stale ('stateChanged' is async): just unlink and skip them;
        # the slot dispatcher copes (lib/pool.js:934-951).
        idleq = self.p_idleq
        while idleq._len > 0:
            fsm = idleq.shift()
            fsm.p_idleq_node = None
            if not fsm.is_in_state("idle"):
                continue
            handle.try_(fsm)
            self._note_demand()
            return

        if err_on_empty and self.p_resolver.count() < 1:
            handle.fail(mod_errors.NoBackendsError(
                self, self.p_resolver.get_last_error()))

        # Keep the node handle so the claim can unlink itself the
        # moment it leaves 'waiting' (timeout/cancel): a sustained
        # overload no longer accumulates dead entries between feeds.
        # (Deliberate divergence: the reference leaves timed-out
        # entries linked until a dequeue walks past them,
        # lib/pool.js:934-951, which grows without bound when claims
        # time out faster than connections free up.)
        handle.ch_waiter_node = self.p_waiters.push(handle)
        self._note_demand()
        self._hwm_counter("max-claim-queue", self.p_waiters._len)
        self._incr_counter("queued-claim")
        self.rebalance()

    def claim(self, options: Any = None, cb: Optional[Callable] = None):
        """Claim a connection: cb(err, handle, connection).

        Returns the ClaimHandle (or a cancel-only stub when the pool is
        stopping/failed, lib/pool.js:889-910).
        """
        # fast path: claim({}, cb) / claim(cb) with no CoDel — the
        # overwhelmingly common call shape on the hot path
        if not options and cb is not None and self.p_codel is None:
            if _native_pool_claim is not None:
                # counter + state check + stack + handle/ticket in C;
                # True means stopping/stopped/failed (counter already
                # bumped): build the short-circuit error here
                h = _native_pool_claim(ClaimHandle, self, cb,
                                       self.p_claim_log, self._loop)
                if h is not True:
                    return h
                if self._fsm_state == "failed":
                    return self._claim_shortcircuit(
                        cb, mod_errors.PoolFailedError(
                            self, self.p_last_error))
                return self._claim_shortcircuit(
                    cb, mod_errors.PoolStoppingError(self))
            err_on_empty = False
            timeout = math.inf
        else:
            if callable(options) and cb is None:
                cb = options
                options = {}
            options = options or {}
            if cb is None:
                raise TypeError("claim() requires a callback")
            err_on_empty = bool(options.get("errorOnEmpty", False))

            if self.p_codel is not None:
                if options.get("timeout") is not None:
                    raise ValueError("options.timeout not allowed when "
                                     "targetClaimDelay has been set")
                timeout = self.p_codel.get_max_idle()
            elif options.get("timeout") is not None:
                timeout = options["timeout"]
            else:
                timeout = math.inf

        counters = self.p_counters
        counters["claim"] = counters.get("claim", 0) + 1

        st = self._fsm_state
        if st == "stopping" or st == "stopped" or \
                st == "stopping.backends":
            return self._claim_shortcircuit(
                cb, mod_errors.PoolStoppingError(self))
        if st == "failed":
            return self._claim_shortcircuit(
                cb, mod_errors.PoolFailedError(self, self.p_last_error))

        stack = mod_utils.maybe_capture_stack_trace()
        # The handle's construction already queues its async
        # stateChanged('waiting'); the ticket receives it on the next
        # loop turn and runs the first try_next then — claim() never
        # fires the callback synchronously (lib/pool.js:922-968).
        if _native_claim_fast is not None:
            return _native_claim_fast(ClaimHandle, self, stack, cb,
                                      self.p_claim_log, timeout,
                                      self._loop, err_on_empty)
        handle = ClaimHandle.fast(self, stack, cb, self.p_claim_log,
                                  timeout, self._loop)
        handle.on("stateChanged", _ClaimTicket(self, handle, err_on_empty))
        return handle

    def _claim_shortcircuit(self, cb: Callable, err: BaseException):
        state = {"done": False}

        def fire() -> None:
            if not state["done"]:
                cb(err)
            state["done"] = True

        self._loop.call_soon(fire)
        return _CancelStub(state)

    async def claim_async(self, options: Any = None):
        """Coroutine sugar over claim(): returns (handle, connection)."""
        import asyncio
        fut = self._loop.create_future()

        def cb(err, hdl=None, conn=None):
            if fut.cancelled():
                if hdl is not None:
                    hdl.release()
                return
            if err is not None:
                fut.set_exception(err)
            else:
                fut.set_result((hdl, conn))

        res = self.claim(options or {}, cb)
        try:
            return await fut
        except asyncio.CancelledError:
            res.cancel()
            raise
