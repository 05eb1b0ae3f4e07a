"""Moore finite-state-machine runtime.

This is the foundation the entire framework is written against — the
equivalent of the ``mooremachine`` npm module that every class in the
reference extends (reference package.json:13; every lib/*.js file calls
``FSM.call(this, <state>)``).  It is a fresh design for asyncio, but it
reproduces the precise observable semantics the reference's correctness
depends on (survey §7 "hard parts" #1):

1. **State-entry functions.**  A subclass defines ``state_<name>(self, S)``
   methods.  Dotted sub-states like ``stopping.backends`` map to
   ``state_stopping_backends`` (reference lib/pool.js:450 defines the
   sub-state as a property of the parent state function).

2. **Scoped resources.**  The ``S`` handle passed to a state-entry
   function registers event listeners, timeouts and immediates that are
   all torn down automatically when the state is exited
   (lib/connection-fsm.js passim).

3. **Synchronous chained transitions, deferred within entry.**  Signal
   functions (``emit`` + listener + ``S.goto_state``) cause transitions
   that complete synchronously from the caller's point of view — e.g.
   ``fsm.set_unwanted()`` can leave the slot in state ``stopped`` by the
   time it returns (relied on at lib/pool.js:624-631).  A transition
   requested *while the entry function of the same FSM is still running*
   is deferred until that entry function returns (this ordering is what
   makes the claim double-handshake hand the user callback a slot that is
   already ``busy``).

4. **Asynchronous ``stateChanged`` emission.**  ``stateChanged`` events
   are queued and delivered on the next loop turn, to the listeners
   registered *at delivery time*.  Several reference code paths tolerate
   — and depend on — observing a stale state on a queue while the real
   transition event is still pending (lib/pool.js:937-946,
   lib/connection-fsm.js:881-890, :1209-1216).

The runtime sits on the claim hot path (a claim/release cycle walks
6+ state transitions across three FSMs), so it is written for low
allocation: per-class state-entry method caches, tuple-based scope
disposal, and a copy-free single-listener emit fast path.
"""

from __future__ import annotations

import asyncio
from typing import Any, Callable, Dict, List, Optional, Sequence

from .events import EventEmitter

__all__ = ["FSM", "StateScope", "FSMError", "get_loop"]


def get_loop(loop: Optional[asyncio.AbstractEventLoop] = None) -> asyncio.AbstractEventLoop:
    if loop is not None:
        return loop
    try:
        return asyncio.get_running_loop()
    except RuntimeError:
        return asyncio.get_event_loop()


#: optional transition tracer: fn(fsm, new_state) called synchronously on
#: every state transition — the analog of mooremachine's DTrace probes
#: on FSM transitions (docs/internals.adoc:125-131).  None => zero cost.
_TRACER = None


def set_transition_tracer(fn) -> None:
    """Install (or clear, with None) a global FSM transition tracer."""
    global _TRACER
    _TRACER = fn
    if NATIVE:
        from . import _speed as _sp
        _sp._set_tracer(fn)


def get_transition_tracer():
    return _TRACER


class FSMError(AssertionError):
    """Invalid use of an FSM (bad transition, signal in wrong state...)."""


class StateScope:
    """The ``S`` handle given to state-entry functions.

    Everything registered through the scope is disconnected when the FSM
    leaves the state that created it.
    """

    __slots__ = ("_fsm", "_listeners", "_timers", "_active")

    def __init__(self, fsm: "FSM") -> None:
        self._fsm = fsm
        self._listeners: Optional[List] = None   # flat [em, evt, cb, ...]
        self._timers: Optional[List] = None      # cancellables
        self._active = True

    # -- queries ------------------------------------------------------
    @property
    def active(self) -> bool:
        return self._active

    # -- scoped registrations -----------------------------------------
    def on(self, emitter: EventEmitter, event: str, cb: Callable) -> None:
        if not self._active:
            raise FSMError("S.on() used on exited state scope")
        emitter.on(event, cb)
        ls = self._listeners
        if ls is None:
            ls = self._listeners = []
        ls.append(emitter)
        ls.append(event)
        ls.append(cb)

    def _add_timer(self, handle: Any) -> None:
        ts = self._timers
        if ts is None:
            ts = self._timers = []
        ts.append(handle)

    def timeout(self, ms: float, cb: Callable[[], None]) -> None:
        """Run cb after `ms` milliseconds unless the state is exited first."""
        if not self._active:
            raise FSMError("S.timeout() used on exited state scope")
        self._add_timer(self._fsm._loop.call_later(
            ms / 1000.0, self._guarded(cb)))

    def interval(self, ms: float, cb: Callable[[], None]) -> None:
        if not self._active:
            raise FSMError("S.interval() used on exited state scope")
        state = {"h": None, "stop": False}

        def tick() -> None:
            if state["stop"] or not self._active:
                return
            cb()
            if self._active and not state["stop"]:
                state["h"] = self._fsm._loop.call_later(ms / 1000.0, tick)

        state["h"] = self._fsm._loop.call_later(ms / 1000.0, tick)

        class _IntervalHandle:
            __slots__ = ()

            @staticmethod
            def cancel() -> None:
                state["stop"] = True
                if state["h"] is not None:
                    state["h"].cancel()

        self._add_timer(_IntervalHandle)

    def immediate(self, cb: Callable[[], None]) -> None:
        if not self._active:
            raise FSMError("S.immediate() used on exited state scope")
        self._add_timer(self._fsm._loop.call_soon(self._guarded(cb)))

    def callback(self, cb: Callable) -> Callable:
        """Wrap cb so it becomes a no-op once the state has been exited."""

        def wrapper(*args: Any) -> None:
            if self._active:
                cb(*args)

        return wrapper

    def _guarded(self, cb: Callable[[], None]) -> Callable[[], None]:
        def wrapper() -> None:
            if self._active:
                cb()

        return wrapper

    # -- transitions ---------------------------------------------------
    def valid_transitions(self, states: Sequence[str]) -> None:
        self._fsm._fsm_valid = states

    def goto_state(self, state: str) -> None:
        if not self._active:
            # A handler belonging to an already-exited state fired during
            # the same synchronous cascade; the transition it wanted is
            # obsolete.
            return
        self._fsm.goto_state(state)

    def goto_state_on(self, emitter: EventEmitter, event: str, state: str) -> None:
        self.on(emitter, event, lambda *a: self.goto_state(state))

    def goto_state_timeout(self, ms: float, state: str) -> None:
        self.timeout(ms, lambda: self.goto_state(state))

    # -- teardown ------------------------------------------------------
    def _dispose(self) -> None:
        self._active = False
        ls = self._listeners
        if ls is not None:
            self._listeners = None
            for i in range(0, len(ls), 3):
                ls[i].remove_listener(ls[i + 1], ls[i + 2])
        ts = self._timers
        if ts is not None:
            self._timers = None
            for h in ts:
                h.cancel()


class FSM(EventEmitter):
    """Moore machine: outputs (entry actions) are a function of the state."""

    __slots__ = (
        "_loop",
        "_fsm_state",
        "_fsm_scope",
        "_fsm_valid",
        "_fsm_entering",
        "_fsm_pending",
        "_fsm_emit_queue",
        "_fsm_emit_scheduled",
        "_fsm_history",
    )

    #: ring-buffer length for state history (debugging aid; mooremachine
    #: keeps history visible in core dumps)
    HISTORY_LEN = 8

    #: per-class cache of state name -> entry method (unbound)
    _fsm_entry_cache: Dict[str, Callable] = {}

    def __init_subclass__(cls, **kw: Any) -> None:
        super().__init_subclass__(**kw)
        cls._fsm_entry_cache = {}

    def __init__(self, initial_state: str,
                 loop: Optional[asyncio.AbstractEventLoop] = None) -> None:
        super().__init__()
        self._loop = get_loop(loop)
        self._fsm_state: Optional[str] = None
        self._fsm_scope: Optional[StateScope] = None
        self._fsm_valid: Optional[Sequence[str]] = None
        self._fsm_entering = False
        self._fsm_pending: Optional[str] = None
        self._fsm_emit_queue: List[str] = []
        self._fsm_emit_scheduled = False
        self._fsm_history: List[str] = []
        self.goto_state(initial_state)

    # -- introspection -------------------------------------------------
    def get_state(self) -> str:
        if self._fsm_state is None:
            raise FSMError("FSM has no state yet")
        return self._fsm_state

    def is_in_state(self, state: str) -> bool:
        cur = self._fsm_state
        if cur is None:
            return False
        return cur == state or cur.startswith(state + ".")

    def get_state_history(self) -> List[str]:
        return list(self._fsm_history)

    # -- transitions ---------------------------------------------------
    def goto_state(self, state: str) -> None:
        valid = self._fsm_valid
        if valid is not None and state not in valid:
            raise FSMError(
                "%s: invalid transition %r -> %r (valid: %r)"
                % (type(self).__name__, self._fsm_state, state, valid)
            )
        if self._fsm_entering:
            # Requested while this FSM's entry function is still running:
            # defer until it returns (see module docstring, point 3).
            if self._fsm_pending is not None and self._fsm_pending != state:
                raise FSMError(
                    "%s: conflicting deferred transitions %r and %r from %r"
                    % (type(self).__name__, self._fsm_pending, state,
                       self._fsm_state)
                )
            self._fsm_pending = state
            return
        self._enter_loop(state)

    def _entry_for(self, state: str) -> Callable:
        cls = type(self)
        cache = cls._fsm_entry_cache
        entry = cache.get(state)
        if entry is None:
            entry = getattr(cls, "state_" + state.replace(".", "_"), None)
            if entry is None:
                raise FSMError(
                    "%s has no state-entry function for %r"
                    % (cls.__name__, state)
                )
            cache[state] = entry
        return entry

    def _enter_loop(self, state: str) -> None:
        next_state: Optional[str] = state
        hist = self._fsm_history
        while next_state is not None:
            target = next_state
            next_state = None
            if self._fsm_scope is not None:
                self._fsm_scope._dispose()
            self._fsm_valid = None
            self._fsm_state = target
            hist.append(target)
            if len(hist) > self.HISTORY_LEN:
                del hist[0]
            scope = StateScope(self)
            self._fsm_scope = scope
            if _TRACER is not None:
                _TRACER(self, target)
            entry = self._entry_for(target)
            self._fsm_entering = True
            try:
                entry(self, scope)
            finally:
                self._fsm_entering = False
                pend, self._fsm_pending = self._fsm_pending, None
            self._queue_state_changed(target)
            if pend is not None:
                valid = self._fsm_valid
                if valid is not None and pend not in valid:
                    raise FSMError(
                        "%s: invalid transition %r -> %r (valid: %r)"
                        % (type(self).__name__, target, pend, valid)
                    )
                next_state = pend
        # settled in a terminal state (declared by an empty
        # valid-transitions list): subclasses may break reference
        # cycles here (ClaimHandle does — see the native twin's
        # ch_terminal_cleanup)
        if self._fsm_valid is not None and len(self._fsm_valid) == 0:
            hook = getattr(self, "_fsm_terminal_settled", None)
            if hook is not None:
                hook()

    # -- async stateChanged delivery ------------------------------------
    def _queue_state_changed(self, state: str) -> None:
        self._fsm_emit_queue.append(state)
        if not self._fsm_emit_scheduled:
            self._fsm_emit_scheduled = True
            self._loop.call_soon(self._flush_state_changed)

    #: per-flush drain cap: a listener that keeps re-triggering this
    #: FSM's own transitions must not hog one loop callback forever
    #: (mirrors the native core's FSM_FLUSH_CAP)
    FLUSH_CAP = 64

    def _flush_state_changed(self) -> None:
        self._fsm_emit_scheduled = False
        q = self._fsm_emit_queue
        n = 0
        while q and n < self.FLUSH_CAP:
            st = q.pop(0)
            self.emit("stateChanged", st)
            n += 1
        if q and not self._fsm_emit_scheduled:
            self._fsm_emit_scheduled = True
            self._loop.call_soon(self._flush_state_changed)


# ---------------------------------------------------------------------------
# Native core: when cueball_amd._speed is built, its FSM/StateScope
# replace the pure-Python ones above (same semantics, validated by the
# same test suite; CUEBALL_PURE=1 forces the Python implementation).
import os as _os

PurePythonFSM = FSM
PurePythonStateScope = StateScope
NATIVE = False
if not _os.environ.get("CUEBALL_PURE"):
    try:
        from . import _speed as _speed_mod
        _speed_mod._set_helpers(get_loop, FSMError)
        FSM = _speed_mod.FSM  # noqa: F811
        StateScope = _speed_mod.StateScope  # noqa: F811
        NATIVE = True
    except ImportError:
        pass
