"""Small-unit coverage: logutil, metrics exposition, monitor registry,
errors, facade exports."""

import logging

import pytest

import cueball_amd
from cueball_amd.errors import (ConnectionError_, CueballError,
                                full_message)
from cueball_amd.logutil import CueballLogger
from cueball_amd.metrics import create_collector
from cueball_amd.pool_monitor import PoolMonitor


def test_facade_exports():
    for name in cueball_amd.__all__:
        assert getattr(cueball_amd, name) is not None, name


def test_logger_child_fields(caplog):
    log = CueballLogger(logging.getLogger("t"))
    child = log.child(component="X", backend="b1")
    gchild = child.child(port=99)
    with caplog.at_level(logging.WARNING, logger="t"):
        gchild.warn("something %s", "bad", extra_field=7)
    assert "something bad" in caplog.text
    assert "component=X" in caplog.text
    assert "backend=b1" in caplog.text
    assert "port=99" in caplog.text
    assert "extra_field=7" in caplog.text


def test_metrics_counter_and_gauge_exposition():
    c = create_collector(labels={"component": "cueball"})
    cnt = c.counter(name="reqs", help="requests")
    cnt.increment({"code": "200"})
    cnt.increment({"code": "200"})
    cnt.increment({"code": "500"})
    g = c.gauge(name="depth", help="queue depth")
    g.set(7, {"q": "waiters"})
    text = c.collect()
    assert "# TYPE reqs counter" in text
    assert 'reqs{code="200",component="cueball"} 2' in text
    assert 'reqs{code="500",component="cueball"} 1' in text
    assert "# TYPE depth gauge" in text
    assert 'depth{component="cueball",q="waiters"} 7' in text
    # idempotent re-registration (artedi semantics)
    assert c.counter(name="reqs") is cnt
    with pytest.raises(ValueError):
        c.gauge(name="reqs")

    with pytest.raises(ValueError):
        cnt.increment(delta=-1)


def test_full_message_chain():
    inner = ValueError("inner boom")
    mid = CueballError("mid layer", inner)
    outer = CueballError("outer", mid)
    assert full_message(outer) == "outer: mid layer: inner boom"
    e = ConnectionError_({"key": "b1", "address": "1.2.3.4", "port": 80},
                         "error", "connect", inner)
    assert "b1" in str(e)
    assert "inner boom" in str(e)


def test_monitor_registry_errors():
    m = PoolMonitor()

    class FakePool:
        p_uuid = "u1"

    p = FakePool()
    m.register_pool(p)
    assert m.list_objects("pool") == ["u1"]
    m.unregister_pool(p)
    with pytest.raises(KeyError):
        m.unregister_pool(p)
    with pytest.raises(ValueError):
        m.list_objects("bogus")
