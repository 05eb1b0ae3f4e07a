"""The flush batcher's drain cap: a self-sustaining stateChanged chain
must keep yielding to the event loop so other callbacks and IO are not
starved (speed.cpp FLUSH_DRAIN_CAP)."""

from cueball_amd.fsm import FSM
from conftest import run_vt


class PingPong(FSM):
    """Transitions a<->b; each stateChanged re-triggers the next
    transition from its listener, forming a chain that would run
    forever in one callback without the cap."""

    def __init__(self, loop, limit):
        self.hops = 0
        self.limit = limit
        super().__init__("a", loop=loop)

    def state_a(self, S):
        pass

    def state_b(self, S):
        pass


def test_chain_yields_to_other_callbacks():
    async def body(loop):
        fsm = PingPong(loop, limit=40000)

        def on_change(st):
            fsm.hops += 1
            if fsm.hops < fsm.limit:
                fsm.goto_state("b" if st == "a" else "a")

        fsm.on("stateChanged", on_change)

        # schedule a callback AFTER the chain starts; with the drain
        # cap it must run long before the chain completes
        progress = {}

        def probe():
            progress.setdefault("hops_at_probe", fsm.hops)

        loop.call_soon(probe)

        # drive until the chain finishes
        import asyncio
        for _ in range(40000):
            if fsm.hops >= fsm.limit:
                break
            await asyncio.sleep(0)
        assert fsm.hops >= fsm.limit
        # the probe observed the chain mid-flight (not only at the end)
        assert progress["hops_at_probe"] < fsm.limit

    run_vt(lambda loop: body(loop))


def test_chain_completes_in_bounded_loop_turns():
    async def body(loop):
        fsm = PingPong(loop, limit=20000)

        def on_change(st):
            fsm.hops += 1
            if fsm.hops < fsm.limit:
                fsm.goto_state("b" if st == "a" else "a")

        fsm.on("stateChanged", on_change)

        import asyncio
        turns = 0
        while fsm.hops < fsm.limit and turns < 20000:
            await asyncio.sleep(0)
            turns += 1
        assert fsm.hops >= fsm.limit
        # with batching, far fewer loop turns than transitions (the
        # pure runtime flushes per-FSM per turn and is exempt)
        from cueball_amd.events import NATIVE
        if NATIVE:
            assert turns < fsm.limit / 4

    run_vt(lambda loop: body(loop))
