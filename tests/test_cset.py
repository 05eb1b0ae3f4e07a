"""ConnectionSet behavior tests (port of reference test/cset.test.js)."""

import pytest

from cueball_amd.connection_set import ConnectionSet
from cueball_amd.testing import (DummyConnection, DummyResolver, advance,
                                 settle)
from conftest import run_vt

RECOVERY = {"default": {"timeout": 500, "retries": 1, "delay": 0}}


class Ctx:
    def __init__(self, loop, target=2, maximum=4, recovery=None, **opts):
        self.loop = loop
        self.connections = []
        self.resolver = DummyResolver()
        self.added = []    # (ckey, conn, hdl)
        self.removed = []  # (ckey, conn, hdl)

        def constructor(backend):
            c = DummyConnection(backend)
            c.backend = backend.get("key")
            c.seen = False
            self.connections.append(c)
            orig_destroy = c.destroy

            def destroy():
                if c in self.connections:
                    self.connections.remove(c)
                orig_destroy()

            c.destroy = destroy
            return c

        cset_opts = {
            "constructor": constructor,
            "recovery": recovery or RECOVERY,
            "target": target,
            "maximum": maximum,
            "resolver": self.resolver,
            "loop": loop,
        }
        cset_opts.update(opts)
        self.cset = ConnectionSet(cset_opts)

        self.cset.on("added", self._on_added)
        self.cset.on("removed", self._on_removed)

    def _on_added(self, ckey, conn, hdl):
        self.added.append((ckey, conn, hdl))

    def _on_removed(self, ckey, conn, hdl):
        conn.seen = True
        self.removed.append((ckey, conn, hdl))
        hdl.release()

    def counts(self):
        out = {}
        for c in self.connections:
            out[c.backend] = out.get(c.backend, 0) + 1
        return out

    def by_backend(self, key):
        return [c for c in self.connections if c.backend == key]

    def in_set(self):
        return [conn for (ck, conn, h) in self.added
                if not any(conn is c for (ck2, c, h2) in self.removed)]


def test_cset_with_one_backend():
    async def body(loop):
        ctx = Ctx(loop, target=2, maximum=4)
        ctx.resolver.start()
        assert len(ctx.connections) == 0
        ctx.resolver.add("b1", {})
        await advance(loop, 0.1)
        # singleton: one connection per backend regardless of target
        assert len(ctx.connections) == 1
        ctx.connections[0].connect()
        await settle(loop)
        assert len(ctx.added) == 1
        ckey, conn, hdl = ctx.added[0]
        assert ckey.startswith("b1.")
        assert conn is ctx.connections[0]

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")
        # 'removed' was emitted during stop and we released
        assert len(ctx.removed) == 1

    run_vt(lambda loop: body(loop))


def test_cset_with_two_backends():
    async def body(loop):
        ctx = Ctx(loop, target=2, maximum=4)
        ctx.resolver.start()
        ctx.resolver.add("b1", {})
        ctx.resolver.add("b2", {})
        await settle(loop)
        assert len(ctx.connections) == 2
        for c in list(ctx.connections):
            c.connect()
        await settle(loop)
        assert len(ctx.added) == 2
        backends = sorted(c.backend for c in ctx.connections)
        assert backends == ["b1", "b2"]

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_cset_swapping():
    """target=1/max=1: a better-placed backend appears; the old conn is
    only drained after the new one is up (never drop the last working
    connection, lib/set.js:417-429)."""
    async def body(loop):
        ctx = Ctx(loop, target=1, maximum=1)
        ctx.resolver.start()
        ctx.resolver.add("b1", {})
        await settle(loop)
        assert ctx.counts() == {"b1": 1}
        ctx.by_backend("b1")[0].connect()
        await advance(loop, 0.5)
        assert len(ctx.added) == 1
        conn = ctx.by_backend("b1")[0]

        ctx.resolver.add("b0", {})
        ctx.cset.cs_keys.sort()
        assert ctx.cset.cs_keys[0] == "b0"
        await advance(loop, 0.5)
        # b0 slot created, b1 still advertised and alive
        assert len(ctx.connections) == 2
        assert not conn.dead
        assert not conn.seen
        assert len(ctx.in_set()) == 1

        ctx.by_backend("b0")[0].connect()
        await advance(loop, 1.0)
        # b1 drained (removed emitted + released) and torn down
        assert ctx.counts() == {"b0": 1}
        assert conn.dead
        assert conn.seen
        ins = ctx.in_set()
        assert len(ins) == 1
        assert ins[0] is ctx.by_backend("b0")[0]

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_removing_a_backend():
    async def body(loop):
        ctx = Ctx(loop, target=3, maximum=5)
        ctx.resolver.start()
        for b in ("b1", "b2", "b3"):
            ctx.resolver.add(b, {})
        await settle(loop)
        assert ctx.counts() == {"b1": 1, "b2": 1, "b3": 1}
        ctx.by_backend("b1")[0].connect()
        ctx.by_backend("b2")[0].connect()
        # stay under the 500ms connect timeout: in the reference's
        # real-clock test the removal fires just before b3's timeout
        await advance(loop, 0.4)

        conn = ctx.by_backend("b2")[0]
        conn2 = ctx.by_backend("b3")[0]
        assert not conn2.seen_unwanted
        unwanted_during_connect = []
        orig = conn2.set_unwanted

        def watch():
            if not conn2.connected and not conn2.dead:
                unwanted_during_connect.append(True)
            orig()

        conn2.set_unwanted = watch

        ctx.resolver.remove("b2")
        ctx.resolver.remove("b3")
        await advance(loop, 1.0)
        assert conn.dead
        assert conn2.dead
        assert conn.seen           # advertised conn got 'removed'
        assert not conn2.seen      # never advertised: no 'removed'
        assert unwanted_during_connect
        assert ctx.counts() == {"b1": 1}

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_removing_unused_backend_cueball_47():
    async def body(loop):
        ctx = Ctx(loop, target=2, maximum=5)
        ctx.resolver.start()
        bkeys = ["b1", "b2", "b3"]
        for b in bkeys:
            ctx.resolver.add(b, {})
        await settle(loop)
        # target=2 singleton: only 2 of the 3 backends get slots
        assert len(ctx.connections) == 2
        counts = ctx.counts()
        bs = [k for k in bkeys if counts.get(k, 0) > 0]
        nbs = [k for k in bkeys if counts.get(k, 0) == 0]
        assert len(bs) == 2
        ctx.by_backend(bs[0])[0].connect()
        ctx.by_backend(bs[1])[0].connect()

        ctx.resolver.remove(nbs[0])
        await advance(loop, 1.0)
        assert len(ctx.connections) == 2
        counts = ctx.counts()
        assert counts[bs[0]] == 1
        assert counts[bs[1]] == 1
        assert nbs[0] not in counts

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_cset_with_error():
    """A backend that always errors gets declared dead; the cset keeps
    a monitor on it; the other backend still gets advertised."""
    async def body(loop):
        ctx = Ctx(loop, target=2, maximum=4)
        ctx.resolver.start()
        ctx.resolver.add("b1", {})
        ctx.resolver.add("b2", {})
        await settle(loop)
        assert len(ctx.connections) == 2

        ctx.by_backend("b1")[0].connect()
        ctx.by_backend("b2")[0].emit("error", RuntimeError("nope"))
        await advance(loop, 0.5)

        assert ctx.cset.cs_dead.get("b2") is True
        assert len(ctx.added) == 1
        assert ctx.added[0][1].backend == "b1"

        # monitor reconnects b2 -> dead flag cleared, conn advertised
        mons = ctx.by_backend("b2")
        assert len(mons) == 1
        mons[0].connect()
        await advance(loop, 0.5)
        assert "b2" not in ctx.cset.cs_dead
        assert len(ctx.added) == 2

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_cset_connect_reject_92():
    """Slot loses its socket between idle and claim: the logical
    connection must retry rather than wedging (reference #92)."""
    async def body(loop):
        ctx = Ctx(loop, target=1, maximum=2)
        ctx.resolver.start()
        ctx.resolver.add("b1", {})
        await settle(loop)
        conn = ctx.by_backend("b1")[0]
        # connect, then immediately close before the lconn claim settles
        conn.connect()
        conn.emit("close")
        await advance(loop, 0.5)
        # the replacement connects normally and is advertised
        fresh = ctx.by_backend("b1")
        assert len(fresh) == 1
        fresh[0].connect()
        await advance(loop, 0.5)
        assert len(ctx.added) >= 1
        assert ctx.added[-1][1] is fresh[0]

        ctx.cset.stop()
        await advance(loop, 1.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_added_must_be_handled():
    async def body(loop):
        connections = []

        def constructor(backend):
            c = DummyConnection(backend)
            connections.append(c)
            return c

        resolver = DummyResolver()
        cset = ConnectionSet({
            "constructor": constructor,
            "recovery": RECOVERY,
            "target": 1,
            "maximum": 1,
            "resolver": resolver,
            "loop": loop,
        })
        resolver.start()
        resolver.add("b1", {})
        await settle(loop)
        # the failure surfaces from the async claim cascade: it reaches
        # the loop's exception handler (the node equivalent crashes the
        # process with an uncaught exception)
        caught = []
        loop.set_exception_handler(
            lambda lp, ctx_: caught.append(ctx_.get("exception")))
        connections[0].connect()
        await settle(loop)
        assert any(isinstance(e, RuntimeError)
                   and "must be handled" in str(e) for e in caught)

    run_vt(lambda loop: body(loop))


def test_set_target_grows_set():
    async def body(loop):
        ctx = Ctx(loop, target=1, maximum=5)
        ctx.resolver.start()
        for b in ("b1", "b2", "b3"):
            ctx.resolver.add(b, {})
        await settle(loop)
        assert len(ctx.connections) == 1

        ctx.cset.set_target(3)
        await settle(loop)
        assert len(ctx.connections) == 3

    run_vt(lambda loop: body(loop))


def test_removing_last_backends_resolver():
    """All advertised backends removed at once: everything drains, and
    the remaining backend takes over (test/cset.test.js:578)."""
    async def body(loop):
        ctx = Ctx(loop, target=3, maximum=5)
        ctx.resolver.start()
        for b in ("b1", "b2", "b3", "b4"):
            ctx.resolver.add(b, {})
        ctx.cset.cs_keys.sort()
        assert ctx.cset.cs_keys == ["b1", "b2", "b3", "b4"]
        await settle(loop)
        assert ctx.counts() == {"b1": 1, "b2": 1, "b3": 1}
        c1 = ctx.by_backend("b1")[0]
        c2 = ctx.by_backend("b2")[0]
        c3 = ctx.by_backend("b3")[0]
        for c in (c1, c2, c3):
            c.connect()
        await advance(loop, 0.4)
        assert len(ctx.in_set()) == 3

        ctx.resolver.remove("b1")
        ctx.resolver.remove("b2")
        ctx.resolver.remove("b3")
        await advance(loop, 0.4)
        assert c1.dead and c2.dead and c3.dead
        assert c1.seen and c2.seen and c3.seen
        assert len(ctx.in_set()) == 0
        assert ctx.counts() == {"b4": 1}
        ctx.by_backend("b4")[0].connect()
        await advance(loop, 1.0)
        assert ctx.counts() == {"b4": 1}
        assert len(ctx.in_set()) == 1

        ctx.cset.stop()
        await advance(loop, 2.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_removing_last_backend_rebal():
    """A rebalance away from an advertised backend drains it only after
    replacements connect (test/cset.test.js:669)."""
    async def body(loop):
        ctx = Ctx(loop, target=2, maximum=5)
        ctx.resolver.start()
        for b in ("b1", "b2", "b3", "b4"):
            ctx.resolver.add(b, {})
        ctx.cset.cs_keys.sort()
        assert ctx.cset.cs_keys == ["b1", "b2", "b3", "b4"]
        await settle(loop)
        assert ctx.counts() == {"b1": 1, "b2": 1}
        c1 = ctx.by_backend("b1")[0]
        c2 = ctx.by_backend("b2")[0]
        c1.connect()
        c2.connect()
        await advance(loop, 0.4)
        assert len(ctx.in_set()) == 2

        ctx.cset.cs_keys.reverse()  # now prefers b4, b3
        ctx.cset.rebalance()
        await advance(loop, 0.45)
        # b3/b4 slots created; their conns connect -> b1/b2 drain
        for c in list(ctx.connections):
            if not c.connected and not c.dead:
                c.connect()
        await advance(loop, 1.0)
        counts = ctx.counts()
        assert counts.get("b4") == 1 and counts.get("b3") == 1
        assert "b1" not in counts
        ins = ctx.in_set()
        assert len(ins) == 2

        ctx.cset.stop()
        await advance(loop, 2.0)
        assert ctx.cset.is_in_state("stopped")

    run_vt(lambda loop: body(loop))
