"""Prometheus-style metric collection (artedi equivalent).

The reference counts error events through the artedi collector
(lib/utils.js:395-444) with a fixed label set {hostname, uuid, type,
evt}; consumers may pass in a shared collector (README.adoc:113,137).
This is a small self-contained implementation with Prometheus text
exposition for scraping alongside the kang endpoint.
"""

from __future__ import annotations

import threading
from typing import Dict, Mapping, Optional, Tuple

__all__ = ["Collector", "Counter", "Gauge", "create_collector"]


def _labels_key(labels: Mapping[str, str]) -> Tuple[Tuple[str, str], ...]:
    return tuple(sorted((str(k), str(v)) for k, v in labels.items()))


class _Metric:
    kind = "untyped"

    def __init__(self, name: str, help_: str,
                 static_labels: Mapping[str, str]) -> None:
        self.name = name
        self.help = help_
        self.static_labels = dict(static_labels)
        self._values: Dict[Tuple[Tuple[str, str], ...], float] = {}
        self._lock = threading.Lock()

    def _bump(self, labels: Optional[Mapping[str, str]], delta: float,
              absolute: bool = False) -> None:
        key = _labels_key(labels or {})
        with self._lock:
            if absolute:
                self._values[key] = delta
            else:
                self._values[key] = self._values.get(key, 0.0) + delta

    def value(self, labels: Optional[Mapping[str, str]] = None) -> float:
        return self._values.get(_labels_key(labels or {}), 0.0)

    def expose(self) -> str:
        lines = [
            "# HELP %s %s" % (self.name, self.help),
            "# TYPE %s %s" % (self.name, self.kind),
        ]
        with self._lock:
            items = list(self._values.items())
        for key, val in items:
            labels = dict(self.static_labels)
            labels.update(dict(key))
            if labels:
                lstr = ",".join('%s="%s"' % (k, v)
                                for k, v in sorted(labels.items()))
                lines.append("%s{%s} %s" % (self.name, lstr, _fmt(val)))
            else:
                lines.append("%s %s" % (self.name, _fmt(val)))
        return "\n".join(lines) + "\n"


def _fmt(v: float) -> str:
    return str(int(v)) if float(v).is_integer() else repr(v)


class Counter(_Metric):
    kind = "counter"

    def increment(self, labels: Optional[Mapping[str, str]] = None,
                  delta: float = 1.0) -> None:
        if delta < 0:
            raise ValueError("counter cannot decrease")
        self._bump(labels, delta)

    add = increment


class Gauge(_Metric):
    kind = "gauge"

    def set(self, value: float,
            labels: Optional[Mapping[str, str]] = None) -> None:
        self._bump(labels, value, absolute=True)

    def add(self, delta: float,
            labels: Optional[Mapping[str, str]] = None) -> None:
        self._bump(labels, delta)


class Collector:
    """Registry of named metrics; ``counter()``/``gauge()`` are idempotent
    (lib/utils.js:407-411 relies on re-registration being harmless)."""

    def __init__(self, labels: Optional[Mapping[str, str]] = None) -> None:
        self.labels = dict(labels or {})
        self._metrics: Dict[str, _Metric] = {}
        self._lock = threading.Lock()

    def _register(self, cls, name: str, help_: str) -> _Metric:
        with self._lock:
            m = self._metrics.get(name)
            if m is not None:
                if not isinstance(m, cls):
                    raise ValueError("metric %r already registered with a "
                                     "different type" % name)
                return m
            m = cls(name, help_, self.labels)
            self._metrics[name] = m
            return m

    def counter(self, *, name: str, help: str = "") -> Counter:
        return self._register(Counter, name, help)  # type: ignore[return-value]

    def gauge(self, *, name: str, help: str = "") -> Gauge:
        return self._register(Gauge, name, help)  # type: ignore[return-value]

    def get_collector(self, name: str) -> _Metric:
        return self._metrics[name]

    def collect(self) -> str:
        """Prometheus text exposition of every metric."""
        return "".join(m.expose() for m in self._metrics.values())


def create_collector(labels: Optional[Mapping[str, str]] = None) -> Collector:
    return Collector(labels=labels)
