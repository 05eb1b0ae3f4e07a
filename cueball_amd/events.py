"""Node-style EventEmitter.

The whole framework is built on the observer pattern the reference gets
from node's ``events`` module (see /root/reference/lib/*.js, all FSMs are
EventEmitters).  We reimplement the exact delivery semantics we rely on:

- ``emit`` delivers synchronously, to a snapshot of the listener list
  taken at emit time (so a listener removed *by another listener during
  the same emit* is still called, and a listener added during the emit is
  not).
- ``once`` wrappers expose the original function via ``.listener`` (the
  claim-handle leak detector inspects this, reference
  lib/connection-fsm.js:751).
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List


class EventEmitter:
    __slots__ = ("_events",)

    def __init__(self) -> None:
        self._events: Dict[str, List[Callable]] = {}

    # -- registration -------------------------------------------------
    def on(self, event: str, listener: Callable) -> Callable:
        self._events.setdefault(event, []).append(listener)
        return listener

    add_listener = on

    def once(self, event: str, listener: Callable) -> Callable:
        def wrapper(*args: Any) -> None:
            self.remove_listener(event, wrapper)
            listener(*args)

        wrapper.listener = listener  # type: ignore[attr-defined]
        self.on(event, wrapper)
        return wrapper

    def remove_listener(self, event: str, listener: Callable) -> None:
        ls = self._events.get(event)
        if not ls:
            return
        try:
            ls.remove(listener)
        except ValueError:
            # also allow removing a once() registration by its inner fn
            # (== comparison: bound methods are fresh objects per access)
            for w in ls:
                inner = getattr(w, "listener", None)
                if inner is not None and inner == listener:
                    ls.remove(w)
                    break
        if not ls:
            self._events.pop(event, None)

    def remove_all_listeners(self, event: str | None = None) -> None:
        if event is None:
            self._events.clear()
        else:
            self._events.pop(event, None)

    # -- introspection ------------------------------------------------
    def listeners(self, event: str) -> List[Callable]:
        return list(self._events.get(event, ()))

    def listener_count(self, event: str) -> int:
        return len(self._events.get(event, ()))

    def event_names(self) -> List[str]:
        return list(self._events.keys())

    # -- delivery -----------------------------------------------------
    def emit(self, event: str, *args: Any) -> bool:
        ls = self._events.get(event)
        if not ls:
            return False
        if len(ls) == 1:
            # copy-free fast path; snapshot semantics still hold (a
            # listener added during this call is not invoked)
            ls[0](*args)
            return True
        for listener in tuple(ls):
            listener(*args)
        return True


# ---------------------------------------------------------------------------
# Native core: when the C++ extension is built (cueball_amd._speed), its
# EventEmitter replaces the pure-Python one above — identical semantics
# (the full test suite runs against either; set CUEBALL_PURE=1 to force
# the Python implementation).
import os as _os

PurePythonEventEmitter = EventEmitter
NATIVE = False
if not _os.environ.get("CUEBALL_PURE"):
    try:
        from ._speed import EventEmitter  # noqa: F811
        NATIVE = True
    except ImportError:
        pass
