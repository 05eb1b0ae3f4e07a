"""Intrusive doubly-linked deque with O(1) removal by node handle.

Equivalent of reference lib/queue.js: the pool keeps its idle queue, init
queue and waiter queue in this structure so a slot that changes state can
unlink itself from the middle of a queue without a scan
(lib/pool.js:689, :756, :960).

Implemented with ``__slots__`` and no per-operation allocation beyond the
node itself; this sits on the claim hot path.
"""

from __future__ import annotations

from typing import Any, Callable, Iterator, Optional

__all__ = ["Queue", "QueueNode"]


class QueueNode:
    __slots__ = ("value", "_next", "_prev", "_queue")

    def __init__(self, queue: Optional["Queue"], value: Any) -> None:
        self.value = value
        self._next: Optional["QueueNode"] = None
        self._prev: Optional["QueueNode"] = None
        self._queue = queue

    @property
    def linked(self) -> bool:
        return self._next is not None

    def remove(self) -> None:
        nxt = self._next
        prv = self._prev
        if nxt is None or prv is None:
            raise ValueError("QueueNode.remove() on unlinked node")
        prv._next = nxt
        nxt._prev = prv
        self._next = None
        self._prev = None
        self._queue._len -= 1  # type: ignore[union-attr]


class Queue:
    __slots__ = ("_head", "_tail", "_len")

    def __init__(self) -> None:
        self._head = QueueNode(None, None)
        self._tail = QueueNode(None, None)
        self._head._next = self._tail
        self._tail._prev = self._head
        self._len = 0

    def is_empty(self) -> bool:
        return self._head._next is self._tail

    def peek(self) -> Any:
        n = self._head._next
        if n is self._tail:
            raise IndexError("peek from empty Queue")
        return n.value  # type: ignore[union-attr]

    def push(self, value: Any) -> QueueNode:
        """Append; returns the node handle for later O(1) .remove()."""
        n = QueueNode(self, value)
        prev = self._tail._prev
        n._prev = prev
        n._next = self._tail
        prev._next = n  # type: ignore[union-attr]
        self._tail._prev = n
        self._len += 1
        return n

    def shift(self) -> Any:
        n = self._head._next
        if n is self._tail:
            raise IndexError("shift from empty Queue")
        n.remove()  # type: ignore[union-attr]
        return n.value  # type: ignore[union-attr]

    def for_each(self, cb: Callable[[Any, QueueNode], None]) -> None:
        n = self._head._next
        while n is not self._tail:
            nxt = n._next  # type: ignore[union-attr]
            cb(n.value, n)  # type: ignore[union-attr]
            n = nxt

    def __iter__(self) -> Iterator[Any]:
        n = self._head._next
        while n is not self._tail:
            yield n.value  # type: ignore[union-attr]
            n = n._next  # type: ignore[union-attr]

    def __len__(self) -> int:
        return self._len

    @property
    def length(self) -> int:
        return self._len


# ---------------------------------------------------------------------------
# Native core: when the C++ extension is built, its intrusive deque
# replaces the pure-Python one above — identical semantics (the suite
# runs against either; CUEBALL_PURE=1 forces the Python classes).
import os as _os

PurePythonQueue = Queue
PurePythonQueueNode = QueueNode

if not _os.environ.get("CUEBALL_PURE"):
    try:
        from ._speed import Queue, QueueNode  # type: ignore # noqa: F811
    except ImportError:
        pass
