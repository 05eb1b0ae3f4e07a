"""Native-core specifics: the C components added in round 2 (intrusive
deque, claim ticket, claim_fast, SlotKit, SlotDispatch) and their
fallback seams.  These tests exercise the native objects directly when
the extension is loaded and are skipped under CUEBALL_PURE."""

import os

import pytest

from cueball_amd.events import NATIVE
from cueball_amd.pool import ConnectionPool
from cueball_amd.resolver import ResolverFSM
from cueball_amd.testing import DummyConnection, DummyResolver, settle
from conftest import run_vt

pytestmark = pytest.mark.skipif(
    not NATIVE or os.environ.get("CUEBALL_PURE"),
    reason="native core not loaded")

RECOVERY = {"default": {"timeout": 500, "retries": 1, "delay": 0}}


def _mk(loop, conns, checker=None, **opts):
    resolver = DummyResolver()
    rfsm = ResolverFSM(resolver, {"loop": loop})

    def constructor(backend):
        c = DummyConnection(backend)
        conns.append(c)
        return c

    pool_opts = {
        "domain": "native",
        "constructor": constructor,
        "recovery": RECOVERY,
        "spares": 2,
        "maximum": 4,
        "resolver": rfsm,
        "loop": loop,
    }
    if checker is not None:
        pool_opts["checker"] = checker
        pool_opts["checkTimeout"] = 30000
    pool_opts.update(opts)
    pool = ConnectionPool(pool_opts)
    rfsm.start()
    return pool, resolver


def test_slotkit_installed_only_without_checker():
    async def body(loop):
        conns = []
        pool, resolver = _mk(loop, conns)
        resolver.add("b1", {})
        await settle(loop)
        slots = [f for fl in pool.p_connections.values() for f in fl]
        assert slots and all(
            getattr(s, "_csf_kit", None) is not None for s in slots)
        pool.stop()
        await settle(loop)

        conns2 = []
        pool2, resolver2 = _mk(loop, conns2,
                               checker=lambda hdl, conn: hdl.release())
        resolver2.add("b1", {})
        await settle(loop)
        slots2 = [f for fl in pool2.p_connections.values() for f in fl]
        assert slots2 and all(
            getattr(s, "_csf_kit", None) is None for s in slots2)
        pool2.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_native_queue_iteration_snapshot():
    from cueball_amd.queue import Queue

    q = Queue()
    nodes = [q.push(i) for i in range(5)]
    assert list(q) == [0, 1, 2, 3, 4]
    nodes[2].remove()
    assert list(q) == [0, 1, 3, 4]
    assert len(q) == 4
    assert q._len == 4
    assert q.length == 4
    # for_each with removal of the current node
    seen = []

    def cb(v, n):
        seen.append(v)
        if v == 1:
            n.remove()

    q.for_each(cb)
    assert seen == [0, 1, 3, 4]
    assert list(q) == [0, 3, 4]


def test_native_ticket_claim_and_requeue():
    """The native ticket must retry after a reject: kill the claimed
    slot's socket mid-handshake and watch the claim land elsewhere."""

    async def body(loop):
        conns = []
        pool, resolver = _mk(loop, conns, spares=2, maximum=2)
        resolver.add("b1", {})
        await settle(loop)
        for c in conns:
            c.connect()
        await settle(loop)

        # both idle; claim both, then a third claim queues as waiter
        boxes = []

        def mkcb():
            b = {}
            boxes.append(b)
            return lambda e, h=None, c=None: b.update(err=e, hdl=h)

        pool.claim({}, mkcb())
        pool.claim({}, mkcb())
        pool.claim({}, mkcb())
        await settle(loop)
        assert boxes[0]["hdl"] is not None
        assert boxes[1]["hdl"] is not None
        assert "hdl" not in boxes[2] or boxes[2].get("hdl") is None
        assert pool.get_stats()["waiterCount"] == 1

        # release one: the waiter must be fed through the native
        # SlotDispatch idle path
        boxes[0]["hdl"].release()
        await settle(loop)
        assert boxes[2].get("hdl") is not None

        for b in (boxes[1], boxes[2]):
            b["hdl"].release()
        await settle(loop)
        assert pool.get_stats()["idleConnections"] == 2
        pool.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_native_dispatch_unwanted_backend_on_release():
    """Release a conn whose backend disappeared while it was claimed:
    the C idle path must hand off to set_unwanted."""

    async def body(loop):
        conns = []
        pool, resolver = _mk(loop, conns, spares=1, maximum=1)
        resolver.add("b1", {})
        await settle(loop)
        for c in conns:
            c.connect()
        await settle(loop)

        box = {}
        pool.claim({}, lambda e, h=None, c=None: box.update(hdl=h))
        await settle(loop)
        hdl = box["hdl"]

        resolver.remove("b1")
        await settle(loop)
        # claimed conn survives removal until release
        hdl.release()
        await settle(loop)
        # the slot was told it is unwanted and wound down
        assert all(c.seen_unwanted or c.dead for c in conns)
        pool.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_native_and_pure_tickets_interchangeable():
    """A handle claimed through claim_fast still supports the full
    public handle API (release/close/cancel paths are shared C)."""

    async def body(loop):
        conns = []
        pool, resolver = _mk(loop, conns)
        resolver.add("b1", {})
        await settle(loop)
        for c in conns:
            c.connect()
        await settle(loop)

        box = {}
        ret = pool.claim({}, lambda e, h=None, c=None: box.update(h=h))
        await settle(loop)
        assert ret is box["h"]
        assert ret.is_in_state("claimed")
        with pytest.raises(Exception):
            ret.try_(None)  # misuse: only valid while waiting
        ret.release()
        await settle(loop)
        assert ret.is_in_state("released")
        pool.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))
