"""Listener/resource hygiene regression tests (round-2 review items).

Covers:
- stop-path stateChanged listeners are scoped and removed once the
  pool/set finishes stopping (no growth across stop/start churn);
- DnsClient per-loop semaphores do not accumulate as event loops are
  created and destroyed;
- CoDel's get_max_idle() uses the healthy 10x bound at cold start;
- an explicit delaySpread of 0.0 is preserved (not replaced by the
  0.2 default).
"""

import gc
import math

from cueball_amd.codel import ControlledDelay
from cueball_amd.connection_set import ConnectionSet
from cueball_amd.dns_client import DnsClient
from cueball_amd.logutil import default_logger
from cueball_amd.pool import ConnectionPool
from cueball_amd.pool_monitor import monitor as global_monitor
from cueball_amd.resolver import ResolverFSM
from cueball_amd.testing import (DummyConnection, DummyResolver, VirtualLoop,
                                 advance, settle)
from conftest import run_vt

RECOVERY = {"default": {"timeout": 500, "retries": 1, "delay": 0}}


def _mk_pool(loop, resolver_fsm, connections, **opts):
    def constructor(backend):
        c = DummyConnection(backend)
        c.backend = backend.get("key")
        connections.append(c)
        return c

    pool_opts = {
        "domain": "hygiene",
        "constructor": constructor,
        "recovery": RECOVERY,
        "spares": 2,
        "maximum": 4,
        "resolver": resolver_fsm,
        "loop": loop,
    }
    pool_opts.update(opts)
    return ConnectionPool(pool_opts)


def test_stop_path_listener_is_scoped():
    """A slot still busy when the pool stops gets a stop-path listener;
    it must be gone again once the pool reaches 'stopped'."""

    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []
        pool = _mk_pool(loop, rfsm, conns)
        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        for c in list(conns):
            c.connect()
        await settle(loop)

        box = {}

        def cb(err, hdl=None, conn=None):
            box["hdl"] = hdl

        pool.claim({}, cb)
        await settle(loop)
        assert box.get("hdl") is not None

        slots = [f for fl in pool.p_connections.values() for f in fl]
        base = {id(f): f.listener_count("stateChanged") for f in slots}

        pool.stop()
        await settle(loop)
        # pool is waiting on the busy slot: the stop path registered
        # exactly one extra listener on it
        busy = [f for f in slots if f.is_in_state("busy")]
        assert busy, "expected the claimed slot to still be busy"
        assert not pool.is_in_state("stopped")
        for f in busy:
            assert f.listener_count("stateChanged") == base[id(f)] + 1

        box["hdl"].release()
        await settle(loop)
        assert pool.is_in_state("stopped")
        # scoped stop listeners were removed when the state was exited
        for f in slots:
            assert f.listener_count("stateChanged") <= base[id(f)]

    run_vt(lambda loop: body(loop))


def test_pool_stop_start_churn_no_listener_growth():
    """1k pool lifecycles against one shared resolver: the resolver's
    listener table must return to baseline every time."""

    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        rfsm.start()
        resolver.add("b1", {})
        resolver.add("b2", {})
        await settle(loop)

        def rcount():
            return sum(rfsm.listener_count(e)
                       for e in ("added", "removed", "stateChanged"))

        base = rcount()
        for _ in range(1000):
            conns = []
            pool = _mk_pool(loop, rfsm, conns)
            await settle(loop)
            for c in list(conns):
                c.connect()
            await settle(loop)
            pool.stop()
            await settle(loop)
            assert pool.is_in_state("stopped")
            assert rcount() == base
        # and the monitor registry is clean
        assert global_monitor.list_objects("pool") == []

    run_vt(lambda loop: body(loop))


def test_cset_stop_listener_scoping():
    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []

        def constructor(backend):
            c = DummyConnection(backend)
            conns.append(c)
            return c

        def rcount():
            return sum(rfsm.listener_count(e)
                       for e in ("added", "removed", "stateChanged"))

        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        base = None
        for _ in range(50):
            cs = ConnectionSet({
                "constructor": constructor,
                "recovery": RECOVERY,
                "target": 1,
                "maximum": 2,
                "resolver": rfsm,
                "loop": loop,
            })
            held = {}
            cs.on("added", lambda ck, conn, hdl: held.setdefault(ck, hdl))
            cs.on("removed", lambda ck, conn, hdl:
                  held.pop(ck).release() if ck in held else None)
            await settle(loop)
            for c in list(conns):
                if not c.connected and not c.dead:
                    c.connect()
            await settle(loop)
            if base is None:
                base = rcount()
            cs.stop()
            await settle(loop)
            assert cs.is_in_state("stopped")
            assert rcount() <= base
            conns.clear()

    run_vt(lambda loop: body(loop))


def test_dns_client_sems_do_not_accumulate():
    client = DnsClient(concurrency=3)
    for _ in range(200):
        loop = VirtualLoop()
        try:
            sem = client._sem(loop)
            assert sem is client._sem(loop)
        finally:
            loop.close()
        del loop, sem
    gc.collect()
    assert len(client._sems) <= 1


def test_codel_cold_start_uses_healthy_bound():
    async def body(loop):
        # jump the virtual clock well past 10x target before creating
        # the CoDel instance: a fresh instance must still report the
        # healthy bound, not the overloaded 3x clamp
        await advance(loop, 100.0)
        cd = ControlledDelay(500.0, loop=loop)
        assert cd.get_max_idle() == 5000.0
        # ...and it degrades to 3x once the queue has not been empty
        # for longer than the bound
        await advance(loop, 6.0)
        assert cd.get_max_idle() == 1500.0
        cd.empty()
        assert cd.get_max_idle() == 5000.0

    run_vt(lambda loop: body(loop))


def test_explicit_zero_delay_spread_preserved():
    from cueball_amd.connection_fsm import SocketMgrFSM

    async def body(loop):
        recov = {"default": {"timeout": 100, "retries": 2, "delay": 50,
                             "maxDelay": 1000, "delaySpread": 0.0}}
        made = []

        def constructor(backend):
            c = DummyConnection(backend)
            made.append(c)
            return c

        class FakePool:
            def _incr_counter(self, name):
                pass

        sm = SocketMgrFSM({
            "backend": {"key": "k", "name": "h", "address": "::1",
                        "port": 1},
            "constructor": constructor,
            "recovery": recov,
            "pool": FakePool(),
            "slot": None,
            "monitor": False,
            "log": default_logger(),
            "loop": loop,
        })
        assert sm.sm_delay_spread == 0.0
        assert math.isclose(sm.sm_max_delay, 1000)

    run_vt(lambda loop: body(loop))
