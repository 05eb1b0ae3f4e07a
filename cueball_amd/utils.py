"""Shared infrastructure: the rebalance planner, recovery-spec
validation, jittered delays, stack-trace capture gating and error
metrics (reference lib/utils.js).

``plan_rebalance`` is the algorithmic core of pool/set sizing; its
behavior (dead-backend single-connection + replacement allocation under
the max cap) is pinned by the 21-case table in tests/test_utils.py, the
rebuild's port of the reference's spec suite (test/utils.test.js:13-267).
"""

from __future__ import annotations

import math
import os
import random
import socket
import time
import traceback
from typing import Any, Dict, List, Mapping, Optional, Sequence

from . import metrics as mod_metrics

__all__ = [
    "shuffle",
    "plan_rebalance",
    "assert_recovery",
    "assert_recovery_set",
    "assert_claim_delay",
    "current_millis",
    "gen_delay",
    "stack_traces_enabled",
    "enable_stack_traces",
    "disable_stack_traces",
    "maybe_capture_stack_trace",
    "create_error_metrics",
    "update_error_metrics",
    "METRIC_CUEBALL_EVENT_COUNTER",
    "TRACKED_ERROR_EVENTS",
]

METRIC_CUEBALL_EVENT_COUNTER = "cueball_events"

#: error-related events tracked in metrics (lib/utils.js:37-46)
TRACKED_ERROR_EVENTS = frozenset([
    "timeout-during-connect",
    "error-during-connect",
    "close-during-connect",
    "error-while-connected",
    "retries-exhausted",
    "claim-timeout",
    "error-while-claimed",
    "failed-state",
])

_STACK_TRACES_ENABLED = False


def stack_traces_enabled() -> bool:
    """Claim/release stack capture is off by default for performance
    (lib/utils.js:52-58); toggled via cueball_amd.enable_stack_traces()."""
    return _STACK_TRACES_ENABLED


def enable_stack_traces() -> None:
    global _STACK_TRACES_ENABLED
    _STACK_TRACES_ENABLED = True


def disable_stack_traces() -> None:
    global _STACK_TRACES_ENABLED
    _STACK_TRACES_ENABLED = False


_DISABLED_STACK = [
    "unknown (stack traces disabled)",
    "unknown (stack traces disabled)",
]


def maybe_capture_stack_trace() -> List[str]:
    """Real stack frames if enabled, else a shared 2-frame placeholder
    (lib/utils.js:106-115; off by default "for performance" — the
    shared list keeps the disabled path allocation-free)."""
    if not _STACK_TRACES_ENABLED:
        return _DISABLED_STACK
    frames = traceback.extract_stack()[:-1]
    return ["%s (%s:%d)" % (f.name, f.filename, f.lineno) for f in frames]


def current_millis() -> float:
    """Monotonic wall time in ms (lib/utils.js:198-204).  NOTE: pool/CoDel
    code paths use the event loop's clock instead so virtual-time tests
    work; this is for callers outside a loop."""
    return time.monotonic() * 1000.0


def shuffle(array: List[Any], rng: Optional[random.Random] = None) -> List[Any]:
    """In-place Fisher-Yates shuffle (lib/utils.js:207-217)."""
    rnd = rng.random if rng is not None else random.random
    i = len(array)
    while i > 0:
        j = int(rnd() * i)
        i -= 1
        array[i], array[j] = array[j], array[i]
    return array


def gen_delay(recov_or_delay: Any, spread: Optional[float] = None,
              rng: Optional[random.Random] = None) -> int:
    """Randomize a retry delay by +/- spread/2 around its base value, to
    de-synchronize retries across clients (lib/utils.js:446-461,
    docs/internals.adoc:388-404)."""
    base = recov_or_delay
    if isinstance(recov_or_delay, Mapping) and spread is None:
        base = recov_or_delay["delay"]
        spread = recov_or_delay.get("delay_spread",
                                    recov_or_delay.get("delaySpread"))
    if not isinstance(base, (int, float)):
        raise TypeError("base delay must be a number")
    if spread is None:
        spread = 0.2
    rnd = rng.random if rng is not None else random.random
    return int(round(base * (1 - spread / 2.0 + rnd() * spread)))


def _num(v: Any) -> bool:
    return isinstance(v, (int, float)) and not isinstance(v, bool)


def assert_recovery(obj: Any, name: str = "recovery") -> None:
    """Validate one recovery spec: {retries, timeout, maxTimeout?, delay,
    maxDelay?, delaySpread?} with the retries<32-needs-max rule
    (lib/utils.js:125-186, docs/api.adoc:680-749)."""
    if not isinstance(obj, Mapping):
        raise TypeError("%s must be a mapping" % name)
    allowed = {"retries", "timeout", "maxTimeout", "delay", "maxDelay",
               "delaySpread"}
    extra = set(obj.keys()) - allowed
    if extra:
        raise ValueError("%s has unknown keys: %r" % (name, sorted(extra)))

    retries = obj.get("retries")
    if not _num(retries) or not math.isfinite(retries) or retries < 0:
        raise ValueError("%s.retries must be a finite number >= 0" % name)
    timeout = obj.get("timeout")
    if not _num(timeout) or not math.isfinite(timeout) or timeout <= 0:
        raise ValueError("%s.timeout must be a finite number > 0" % name)
    max_timeout = obj.get("maxTimeout")
    if max_timeout is not None:
        if not _num(max_timeout) or timeout > max_timeout:
            raise ValueError("%s.maxTimeout must be >= timeout" % name)
    delay = obj.get("delay")
    if not _num(delay) or not math.isfinite(delay) or delay < 0:
        raise ValueError("%s.delay must be a finite number >= 0" % name)
    max_delay = obj.get("maxDelay")
    if max_delay is not None:
        if not _num(max_delay) or delay > max_delay:
            raise ValueError("%s.maxDelay must be >= delay" % name)
    spread = obj.get("delaySpread")
    if spread is not None:
        if not _num(spread) or not (0.0 <= spread <= 1.0):
            raise ValueError("%s.delaySpread must be between 0.0 and 1.0"
                             % name)

    day_ms = 1000 * 3600 * 24
    if max_delay is None:
        if retries >= 32:
            raise ValueError("%s.maxDelay is required when retries >= 32 "
                             "(exponential increase becomes unreasonably "
                             "large)" % name)
        if delay * (1 << int(retries)) >= day_ms:
            raise ValueError("%s.maxDelay is required with given values of "
                             "retries and delay (effective unspecified "
                             "maxDelay is > 1 day)" % name)
    if max_timeout is None:
        if retries >= 32:
            raise ValueError("%s.maxTimeout is required when retries >= 32 "
                             "(exponential increase becomes unreasonably "
                             "large)" % name)
        if timeout * (1 << int(retries)) >= day_ms:
            raise ValueError("%s.maxTimeout is required with given values "
                             "of retries and timeout (effective unspecified "
                             "maxTimeout is > 1 day)" % name)


def assert_recovery_set(obj: Any) -> None:
    """Validate a whole recovery-spec object: per-operation keys like
    default/dns/dns_srv/connect/initial (lib/utils.js:117-123)."""
    if not isinstance(obj, Mapping):
        raise TypeError("recovery must be a mapping")
    if "default" not in obj:
        raise ValueError("recovery.default is required")
    for k, v in obj.items():
        assert_recovery(v, "recovery.%s" % k)


def assert_claim_delay(delay: Any) -> None:
    if delay is None:
        return
    if not _num(delay) or not math.isfinite(delay):
        raise ValueError("targetClaimDelay must be a finite number")
    if delay <= 0 or delay != math.floor(delay):
        raise ValueError("targetClaimDelay must be a positive integer")


def plan_rebalance(connections: Mapping[str, Sequence[Any]],
                   dead: Mapping[str, bool],
                   target: int, maximum: int,
                   singleton: bool = False) -> Dict[str, List[Any]]:
    """Compute the add/remove plan that takes the pool to an even spread.

    Semantics (reference lib/utils.js:239-393): round-robin ``target``
    connections over the backend preference list (the iteration order of
    ``connections``); any dead backend gets exactly one slot (its
    monitor) plus a queued *replacement* allocation, replacements
    themselves round-robin with the documented cap/starvation rules so
    every backend is tried at least once even when the max cap prevents
    double-replacement.  With ``singleton`` (Sets), at most one
    connection per distinct backend.

    Returns {"add": [backend keys...], "remove": [connection objects...]}.
    """
    if target < 0:
        raise ValueError("target must be >= 0")
    if maximum < target:
        raise ValueError("max must be >= target")

    keys = list(connections.keys())
    wanted: Dict[str, int] = {}
    plan: Dict[str, List[Any]] = {"add": [], "remove": []}
    replacements = 0
    done = 0

    # Pass 1: spread `target` connections round-robin; dead backends get
    # exactly one (the monitor) and queue a replacement.
    for _ in range(target):
        if not keys:
            break
        k = keys.pop(0)
        keys.append(k)
        if k not in wanted:
            wanted[k] = 0
        if not dead.get(k, False):
            if singleton:
                if wanted[k] == 0:
                    wanted[k] = 1
                    done += 1
            else:
                wanted[k] += 1
                done += 1
            continue
        if wanted[k] == 0:
            wanted[k] = 1
            done += 1
        replacements += 1

    if done + replacements > maximum:
        replacements = maximum - done

    # Pass 2: allocate replacements round-robin, with starvation control
    # under the max cap (see reference comments at lib/utils.js:314-327).
    i = 0
    while i < replacements:
        k = keys.pop(0)
        keys.append(k)
        first_visit = k not in wanted
        if first_visit:
            wanted[k] = 0
        alive = not dead.get(k, False)
        if alive:
            if singleton:
                if wanted[k] == 0:
                    wanted[k] = 1
                    done += 1
                    i += 1
                    continue
            else:
                wanted[k] += 1
                done += 1
                i += 1
                continue

        count = done + replacements - i
        if singleton:
            empties = [kk for kk in keys
                       if not dead.get(kk, False) and kk not in wanted]
        else:
            empties = [kk for kk in keys
                       if not dead.get(kk, False) or kk not in wanted]

        if count + 1 <= maximum:
            # room for both this dead backend and a further replacement
            if wanted[k] == 0:
                wanted[k] = 1
                done += 1
            if empties:
                replacements += 1
        elif count <= maximum and empties:
            # only room for one, but live/untried candidates exist: skip
            replacements += 1
        elif count <= maximum:
            # only room for one and everything looks dead: use this one
            if wanted[k] == 0:
                wanted[k] = 1
                done += 1
        else:
            break
        i += 1

    # Diff what we have against what we want.  Removals prefer backends
    # at the tail of the preference list, and remove the oldest
    # connections first; adds go in preference order.
    fwd = list(connections.keys())
    for key in reversed(fwd):
        have = len(connections.get(key) or ())
        want = wanted.get(key, 0)
        lst = list(connections.get(key) or ())
        while have > want:
            plan["remove"].append(lst.pop(0))
            have -= 1
    for key in fwd:
        have = len(connections.get(key) or ())
        want = wanted.get(key, 0)
        while have < want:
            plan["add"].append(key)
            have += 1

    return plan


def create_error_metrics(options: Mapping[str, Any]) -> mod_metrics.Collector:
    """Get (or make) the shared collector and register the cueball_events
    counter (idempotent, lib/utils.js:395-418)."""
    collector = options.get("collector")
    if collector is None:
        collector = mod_metrics.create_collector(labels={
            "component": "cueball",
        })
    collector.counter(
        name=METRIC_CUEBALL_EVENT_COUNTER,
        help="Total number of cueball error events",
    )
    return collector


def update_error_metrics(collector: mod_metrics.Collector, uuid: str,
                         err_str: str) -> None:
    """Count one tracked error event (lib/utils.js:420-444)."""
    if err_str not in TRACKED_ERROR_EVENTS:
        return
    errors = collector.get_collector(METRIC_CUEBALL_EVENT_COUNTER)
    errors.increment({
        "hostname": socket.gethostname(),
        "uuid": uuid,
        "type": "error",
        "evt": err_str,
    })
