"""Controlled-Delay (CoDel) AQM on the claim waiter queue.

Re-design of reference lib/codel.js (same algorithm, from the published
CoDel pseudocode): when the sojourn time of dequeued claims stays above
``target_claim_delay`` for a full control interval, start timing out
claims at dequeue, at an interval that shrinks with ``1/sqrt(count)``.
``get_max_idle()`` supplies the adaptive upper bound used as the claim
timeout when CoDel is active (lib/codel.js:109-118, lib/pool.js:874-885).

Times are in milliseconds on the event loop's (possibly virtual) clock.
"""

from __future__ import annotations

import math
from .fsm import get_loop

__all__ = ["ControlledDelay", "CODEL_INTERVAL"]

#: CoDel control interval in ms (lib/codel.js:16)
CODEL_INTERVAL = 100.0


class ControlledDelay:
    __slots__ = ("cd_targdelay", "cd_first_above_time", "cd_drop_next",
                 "cd_count", "cd_dropping", "cd_last_empty", "_loop")

    def __init__(self, target_claim_delay: float, loop=None) -> None:
        if not (isinstance(target_claim_delay, (int, float))
                and math.isfinite(target_claim_delay)):
            raise ValueError("target_claim_delay must be finite")
        self.cd_targdelay = float(target_claim_delay)
        self.cd_first_above_time = 0.0
        self.cd_drop_next = 0.0
        self.cd_count = 0
        self.cd_dropping = False
        self._loop = get_loop(loop)
        # Treat the queue as having just been empty at construction so a
        # freshly started pool uses the healthy 10x-target bound from
        # get_max_idle(), matching the reference where the initial
        # undefined compare evaluates false (lib/codel.js:109-118).
        self.cd_last_empty = self._now()

    def _now(self) -> float:
        return self._loop.time() * 1000.0

    def can_drop(self, now: float, start: float) -> bool:
        sojourn = now - start
        if sojourn < self.cd_targdelay:
            self.cd_first_above_time = 0.0
        elif self.cd_first_above_time == 0.0:
            self.cd_first_above_time = now + CODEL_INTERVAL
        elif now >= self.cd_first_above_time:
            return True
        return False

    def get_drop_next(self, now: float) -> float:
        return now + CODEL_INTERVAL / math.sqrt(self.cd_count)

    def overloaded(self, start: float) -> bool:
        """Fed each claim's enqueue time at dequeue; True => time it out."""
        now = self._now()
        ok_to_drop = self.can_drop(now, start)
        drop_claim = False

        if self.cd_dropping:
            if not ok_to_drop:
                self.cd_dropping = False
            elif now >= self.cd_drop_next:
                drop_claim = True
                self.cd_count += 1
        elif ok_to_drop and (
                (now - self.cd_drop_next < CODEL_INTERVAL)
                or (now - self.cd_first_above_time >= CODEL_INTERVAL)):
            drop_claim = True
            self.cd_dropping = True
            if now - self.cd_drop_next < CODEL_INTERVAL:
                self.cd_count = self.cd_count - 2 if self.cd_count > 2 else 1
            else:
                self.cd_count = 1
            self.cd_drop_next = self.get_drop_next(now)

        return drop_claim

    def empty(self) -> None:
        """The waiter queue drained; reset the above-target tracking."""
        self.cd_last_empty = self._now()
        self.cd_first_above_time = 0.0

    def get_max_idle(self) -> float:
        """Adaptive claim-timeout bound: high when healthy, 3x target when
        the queue has not been empty for a while (overloaded)."""
        bound = self.cd_targdelay * 10.0
        now = self._now()
        if self.cd_last_empty < now - bound:
            return self.cd_targdelay * 3.0
        return bound


# ---------------------------------------------------------------------------
# Native core: the C ControlledDelay replaces the pure-Python one
# above when the extension is built — identical algorithm (the CoDel
# suites run against either; CUEBALL_PURE=1 forces the Python class,
# which also lets the pool's python feed path exercise CoDel).
import os as _os

PurePythonControlledDelay = ControlledDelay

if not _os.environ.get("CUEBALL_PURE"):
    try:
        from ._speed import ControlledDelay  # type: ignore # noqa: F811
    except ImportError:
        pass
