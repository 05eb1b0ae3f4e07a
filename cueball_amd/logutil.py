"""Structured child loggers (bunyan-equivalent, minimal).

Every component creates a child logger with bound context fields
(backend key/address/port, component name...), mirroring the reference's
bunyan usage (lib/pool.js:149-157, lib/connection-fsm.js:149-154).
Backed by stdlib logging; fields are rendered into the message suffix.
"""

from __future__ import annotations

import logging
from typing import Any, Mapping, Optional

__all__ = ["CueballLogger", "default_logger"]

TRACE = 5
logging.addLevelName(TRACE, "TRACE")


class CueballLogger:
    __slots__ = ("_logger", "fields")

    def __init__(self, logger: Optional[logging.Logger] = None,
                 fields: Optional[Mapping[str, Any]] = None) -> None:
        self._logger = logger or logging.getLogger("cueball")
        self.fields = dict(fields or {})

    def child(self, fields: Optional[Mapping[str, Any]] = None,
              **kw: Any) -> "CueballLogger":
        merged = dict(self.fields)
        merged.update(fields or {})
        merged.update(kw)
        return CueballLogger(self._logger, merged)

    def _log(self, level: int, msg: str, *args: Any, **extra: Any) -> None:
        if not self._logger.isEnabledFor(level):
            return
        if args:
            msg = msg % args
        ctx = dict(self.fields)
        ctx.update(extra)
        if ctx:
            msg = "%s [%s]" % (msg, ", ".join(
                "%s=%s" % (k, v) for k, v in ctx.items()))
        self._logger.log(level, msg)

    def trace(self, msg: str, *args: Any, **extra: Any) -> None:
        self._log(TRACE, msg, *args, **extra)

    def debug(self, msg: str, *args: Any, **extra: Any) -> None:
        self._log(logging.DEBUG, msg, *args, **extra)

    def info(self, msg: str, *args: Any, **extra: Any) -> None:
        self._log(logging.INFO, msg, *args, **extra)

    def warn(self, msg: str, *args: Any, **extra: Any) -> None:
        self._log(logging.WARNING, msg, *args, **extra)

    warning = warn

    def error(self, msg: str, *args: Any, **extra: Any) -> None:
        self._log(logging.ERROR, msg, *args, **extra)


def default_logger() -> CueballLogger:
    return CueballLogger(logging.getLogger("cueball"))
