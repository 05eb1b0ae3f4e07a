"""The claim path must produce ZERO cyclic garbage (both runtimes).

CPython's cyclic collector was measured to dominate claim p99 (4x)
before the terminal-state cycle breaking; this pins the property so it
cannot regress.  Runs on a real event loop with real sockets, like the
benchmark."""

import asyncio
import gc


def test_claim_cycle_leaves_no_cyclic_garbage():
    import bench

    async def main():
        backends = await bench.start_backends(4)
        loop = asyncio.get_running_loop()
        pool = bench.make_pool(backends, spares=4, maximum=8, loop=loop)
        await bench.wait_for_idle(pool, 4)
        lat = []
        drv = bench.ClaimDriver(pool, loop, 8, lat)
        await drv.run_step(2000)   # warm (pool growth allocates)

        gc.collect()
        gc.disable()
        try:
            await drv.run_step(5000)
            collected = gc.collect()
        finally:
            gc.enable()
        assert collected == 0, (
            "%d cyclic objects left by 5000 claims" % collected)

        pool.stop()
        for s, _ in backends:
            s.close()
        await asyncio.sleep(0.1)

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
    finally:
        loop.close()


def test_timed_out_claims_leave_no_cyclic_garbage():
    """The shed/timeout path (codel-style) ends in the 'failed'
    terminal state; it must be cycle-free too."""
    from cueball_amd.pool import ConnectionPool
    from cueball_amd.resolver import ResolverFSM
    from cueball_amd.testing import DummyResolver, settle
    from conftest import run_vt

    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        pool = ConnectionPool({
            "domain": "gc.test",
            "constructor": lambda b: (_ for _ in ()).throw(
                RuntimeError("never called")),
            "recovery": {"default": {"timeout": 500, "retries": 1,
                                     "delay": 0}},
            "spares": 1,
            "maximum": 1,
            "resolver": rfsm,
            "loop": loop,
        })
        rfsm.start()
        await settle(loop)

        fired = []
        gc.collect()
        gc.disable()
        try:
            for i in range(500):
                pool.claim({"timeout": 5},
                           lambda e, h=None, c=None: fired.append(e))
            from cueball_amd.testing import advance
            await advance(loop, 0.1)
            collected = gc.collect()
        finally:
            gc.enable()
        assert len(fired) == 500
        assert all(e is not None for e in fired)
        assert collected == 0, (
            "%d cyclic objects left by 500 timed-out claims"
            % collected)
        pool.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_timed_out_waiters_unlink_eagerly():
    """Claims that time out while queued must leave the waiter queue
    immediately (not on the next feed): a sustained overload cannot
    accumulate dead entries.  (Deliberate divergence from the
    reference, which leaves them linked until a dequeue walks past,
    lib/pool.js:934-951.)"""
    from cueball_amd.pool import ConnectionPool
    from cueball_amd.resolver import ResolverFSM
    from cueball_amd.testing import (DummyConnection, DummyResolver,
                                     advance, settle)
    from conftest import run_vt

    async def body(loop):
        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []

        def ctor(backend):
            c = DummyConnection(backend)
            conns.append(c)
            return c

        pool = ConnectionPool({
            "domain": "unlink.test",
            "constructor": ctor,
            "recovery": {"default": {"timeout": 500, "retries": 1,
                                     "delay": 0}},
            "spares": 1,
            "maximum": 1,
            "resolver": rfsm,
            "loop": loop,
        })
        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        for c in conns:
            c.connect()
        await settle(loop)

        # occupy the only connection, then queue 200 claims that will
        # time out with no feed ever happening
        box = {}
        pool.claim({}, lambda e, h=None, c=None: box.update(h=h))
        await settle(loop)
        fired = []
        for _ in range(200):
            pool.claim({"timeout": 50},
                       lambda e, h=None, c=None: fired.append(e))
        await settle(loop)
        assert pool.get_stats()["waiterCount"] == 200
        await advance(loop, 0.2)
        assert len(fired) == 200
        # every dead entry unlinked without any release/feed
        assert pool.get_stats()["waiterCount"] == 0

        box["h"].release()
        pool.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))
