"""Attach-on-demand debugging for live processes.

The reference enables claim/release stack capture not only through the
explicit ``enableStackTraces()`` API but automatically when an
operator attaches dtrace to its ``capture-stack`` USDT probe
(lib/utils.js:59-99, the ``fire()`` callback trick) — debugging a
*live, already-misbehaving* process without a restart.  Linux/Python
has no passive probe-attach detection, so the equivalents here are:

1. **Signal toggle** — ``install_attach_handler()`` installs a handler
   (SIGUSR2 by default) that toggles claim/release stack capture and,
   with ``trace_transitions=True``, a transition tracer that logs
   every FSM state change.  ``kill -USR2 <pid>`` on a live process
   turns capture on; a second signal turns it off.  Because signal
   dispositions are process-global, the handler is only installed
   explicitly or via the environment, never as an import side effect
   (unlike the dtrace probe, a stolen signal could break the host
   application).
2. **Environment** — ``CUEBALL_STACK_TRACES=1`` enables capture at
   import; ``CUEBALL_DEBUG_SIGNAL=USR2`` (or ``USR1``/a number)
   installs the toggle handler at import.  Both are applied by
   ``apply_env()``, which the package calls on import.
"""

from __future__ import annotations

import os
import signal
import sys
from typing import Any, Optional

from . import utils as mod_utils
from .fsm import set_transition_tracer

__all__ = ["install_attach_handler", "remove_attach_handler",
           "apply_env", "attach_state"]

_state = {
    "installed_for": None,     # signal number or None
    "prev_handler": None,
    "active": False,           # capture currently on (via the toggle)
    "trace_transitions": False,
    "toggles": 0,
}


def attach_state() -> dict:
    """Introspection for tests/operators."""
    return dict(_state)


def _trace(fsm: Any, new_state: str) -> None:
    sys.stderr.write("[cueball-trace] %s -> %s\n"
                     % (type(fsm).__name__, new_state))


def _toggle(signum: int, frame: Any) -> None:
    _state["toggles"] += 1
    if _state["active"]:
        _state["active"] = False
        mod_utils.disable_stack_traces()
        if _state["trace_transitions"]:
            set_transition_tracer(None)
        sys.stderr.write("[cueball-debug] stack capture DISABLED "
                         "(signal %d)\n" % signum)
    else:
        _state["active"] = True
        mod_utils.enable_stack_traces()
        if _state["trace_transitions"]:
            set_transition_tracer(_trace)
        sys.stderr.write("[cueball-debug] stack capture ENABLED "
                         "(signal %d)\n" % signum)


def install_attach_handler(sig: int = signal.SIGUSR2, *,
                           trace_transitions: bool = False) -> None:
    """Install the live-debug toggle on ``sig`` (default SIGUSR2).

    Idempotent per signal; the previous handler is remembered and
    restored by remove_attach_handler()."""
    if _state["installed_for"] == sig:
        _state["trace_transitions"] = trace_transitions
        return
    if _state["installed_for"] is not None:
        remove_attach_handler()
    _state["prev_handler"] = signal.getsignal(sig)
    _state["installed_for"] = sig
    _state["trace_transitions"] = trace_transitions
    signal.signal(sig, _toggle)


def remove_attach_handler() -> None:
    sig = _state["installed_for"]
    if sig is None:
        return
    signal.signal(sig, _state["prev_handler"] or signal.SIG_DFL)
    _state["installed_for"] = None
    _state["prev_handler"] = None
    if _state["active"]:
        _state["active"] = False
        mod_utils.disable_stack_traces()
        if _state["trace_transitions"]:
            set_transition_tracer(None)


def _parse_signal(spec: str) -> Optional[int]:
    spec = spec.strip().upper()
    if not spec or spec in ("0", "OFF", "NONE"):
        return None
    if spec.isdigit():
        return int(spec)
    name = spec if spec.startswith("SIG") else "SIG" + spec
    return getattr(signal, name, None)


def apply_env(environ=None) -> None:
    """Honor CUEBALL_STACK_TRACES / CUEBALL_DEBUG_SIGNAL (called on
    package import)."""
    env = environ if environ is not None else os.environ
    if env.get("CUEBALL_STACK_TRACES", "") not in ("", "0"):
        mod_utils.enable_stack_traces()
    spec = env.get("CUEBALL_DEBUG_SIGNAL", "")
    if spec:
        sig = _parse_signal(spec)
        if sig is not None:
            install_attach_handler(
                sig, trace_transitions=env.get(
                    "CUEBALL_DEBUG_TRACE", "") not in ("", "0"))
