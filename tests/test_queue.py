"""Intrusive deque tests (reference lib/queue.js semantics)."""

import pytest

from cueball_amd.queue import Queue


def test_push_shift_fifo():
    q = Queue()
    q.push(1)
    q.push(2)
    q.push(3)
    assert len(q) == 3
    assert q.shift() == 1
    assert q.shift() == 2
    assert q.shift() == 3
    assert q.is_empty()


def test_peek():
    q = Queue()
    q.push("a")
    q.push("b")
    assert q.peek() == "a"
    assert len(q) == 2


def test_o1_removal_by_node():
    q = Queue()
    q.push(1)
    n2 = q.push(2)
    q.push(3)
    n2.remove()
    assert len(q) == 2
    assert list(q) == [1, 3]
    assert not n2.linked


def test_double_remove_raises():
    q = Queue()
    n = q.push(1)
    n.remove()
    with pytest.raises(ValueError):
        n.remove()


def test_shift_empty_raises():
    q = Queue()
    with pytest.raises(IndexError):
        q.shift()


def test_for_each_allows_removal_during_iteration():
    q = Queue()
    nodes = [q.push(i) for i in range(5)]
    seen = []

    def cb(v, node):
        seen.append(v)
        if v % 2 == 0:
            node.remove()

    q.for_each(cb)
    assert seen == [0, 1, 2, 3, 4]
    assert list(q) == [1, 3]
