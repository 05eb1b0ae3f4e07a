"""Typed errors with cause chains (reference lib/errors.js).

The reference builds on VError for cause chaining; here each error keeps
a ``cause`` attribute and sets ``__cause__`` so Python tracebacks chain
naturally.  ``full_message`` renders the VError-style "a: b: c" string.
"""

from __future__ import annotations

from typing import Any, Optional

__all__ = [
    "CueballError",
    "ClaimHandleMisusedError",
    "ClaimTimeoutError",
    "NoBackendsError",
    "PoolFailedError",
    "PoolStoppingError",
    "ConnectionError_",
    "ConnectionTimeoutError",
    "ConnectionClosedError",
    "full_message",
]


class CueballError(Exception):
    """Base class: carries an optional cause, chained like VError."""

    def __init__(self, message: str, cause: Optional[BaseException] = None) -> None:
        super().__init__(message)
        self.cause = cause
        if cause is not None:
            self.__cause__ = cause

    @property
    def message(self) -> str:
        return self.args[0]

    def __str__(self) -> str:
        return full_message(self)


def full_message(err: BaseException) -> str:
    """VError.fullMessage() equivalent: 'msg: causemsg: ...'."""
    parts = []
    seen = set()
    cur: Optional[BaseException] = err
    while cur is not None and id(cur) not in seen:
        seen.add(id(cur))
        if isinstance(cur, CueballError):
            parts.append(cur.args[0] if cur.args else type(cur).__name__)
            cur = cur.cause
        else:
            parts.append("%s" % (cur.args[0] if cur.args else type(cur).__name__,))
            cur = cur.__cause__
    return ": ".join(str(p) for p in parts)


class ClaimHandleMisusedError(CueballError):
    """Claim handle treated as if it were a socket (lib/errors.js:25)."""

    def __init__(self) -> None:
        super().__init__(
            "Cueball claim handle used as if it was a socket "
            "(check the order and number of arguments in your claim callbacks)"
        )


class ClaimTimeoutError(CueballError):
    def __init__(self, pool: Any) -> None:
        self.pool = pool
        super().__init__(
            "Timed out while waiting for connection in pool %s (%s)"
            % (getattr(pool, "p_uuid", "?"), getattr(pool, "p_domain", "?"))
        )


class NoBackendsError(CueballError):
    def __init__(self, pool: Any, cause: Optional[BaseException] = None) -> None:
        self.pool = pool
        super().__init__(
            "No backends available in pool %s (%s)"
            % (getattr(pool, "p_uuid", "?"), getattr(pool, "p_domain", "?")),
            cause,
        )


class PoolFailedError(CueballError):
    def __init__(self, pool: Any, cause: Optional[BaseException] = None) -> None:
        self.pool = pool
        dead = len(getattr(pool, "p_dead", {}))
        avail = len(getattr(pool, "p_keys", []))
        uuid = str(getattr(pool, "p_uuid", "?")).split("-")[0]
        super().__init__(
            "Connections to backends of pool %s (%s) are persistently "
            "failing; request aborted (%d of %d declared dead, in state "
            '"failed")' % (uuid, getattr(pool, "p_domain", "?"), dead, avail),
            cause,
        )


class PoolStoppingError(CueballError):
    def __init__(self, pool: Any) -> None:
        self.pool = pool
        uuid = str(getattr(pool, "p_uuid", "?")).split("-")[0]
        super().__init__(
            "Pool %s (%s) is stopping and cannot take new requests"
            % (uuid, getattr(pool, "p_domain", "?"))
        )


class ConnectionError_(CueballError):
    """A connection emitted an error event (lib/errors.js:81).

    Trailing underscore avoids shadowing the Python builtin; exported from
    the package facade as ``ConnectionError_``.
    """

    def __init__(self, backend: dict, event: str, state: str,
                 cause: Optional[BaseException] = None) -> None:
        self.backend = backend
        super().__init__(
            'Connection to backend %s (%s:%s) emitted "%s" during %s'
            % (backend.get("name") or backend.get("key"),
               backend.get("address"), backend.get("port"), event, state),
            cause,
        )


class ConnectionTimeoutError(CueballError):
    def __init__(self, backend: dict) -> None:
        self.backend = backend
        super().__init__(
            "Connection timed out to backend %s (%s:%s)"
            % (backend.get("name") or backend.get("key"),
               backend.get("address"), backend.get("port"))
        )


class ConnectionClosedError(CueballError):
    def __init__(self, backend: dict) -> None:
        self.backend = backend
        super().__init__(
            "Connection closed unexpectedly to backend %s (%s:%s)"
            % (backend.get("name") or backend.get("key"),
               backend.get("address"), backend.get("port"))
        )
