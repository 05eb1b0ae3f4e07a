"""FSM runtime semantics tests.

These pin the mooremachine-equivalent behaviors the rest of the
framework depends on (survey §7 hard-part #1): scoped listener/timer
teardown, synchronous chained transitions with in-entry deferral, async
stateChanged delivery to listeners present at delivery time, dotted
sub-states, and valid-transition enforcement.
"""

import pytest

from cueball_amd.events import EventEmitter
from cueball_amd.fsm import FSM, FSMError
from cueball_amd.testing import VirtualLoop, advance, settle
from conftest import run_vt


class Light(FSM):
    def __init__(self, loop):
        self.entries = []
        super().__init__("red", loop=loop)

    def state_red(self, S):
        self.entries.append("red")
        S.valid_transitions(["green"])
        S.on(self, "go", lambda: S.goto_state("green"))

    def state_green(self, S):
        self.entries.append("green")
        S.valid_transitions(["red"])
        S.on(self, "stop", lambda: S.goto_state("red"))


def test_initial_state_entered():
    def body(loop):
        fsm = Light(loop)
        assert fsm.get_state() == "red"
        assert fsm.entries == ["red"]
        assert fsm.is_in_state("red")
        assert not fsm.is_in_state("green")
        return _noop()

    run_vt(body)


async def _noop():
    return None


def test_transition_is_synchronous():
    async def body(loop):
        fsm = Light(loop)
        fsm.emit("go")
        assert fsm.get_state() == "green"
        fsm.emit("stop")
        assert fsm.get_state() == "red"

    run_vt(lambda loop: body(loop))


def test_invalid_transition_raises():
    async def body(loop):
        fsm = Light(loop)

        with pytest.raises(FSMError):
            fsm.goto_state("red")  # red -> red not declared valid

    run_vt(lambda loop: body(loop))


def test_scoped_listeners_removed_on_exit():
    async def body(loop):
        fsm = Light(loop)
        src = EventEmitter()
        hits = []

        class Watcher(FSM):
            def __init__(self):
                super().__init__("a", loop=loop)

            def state_a(self, S):
                S.on(src, "ping", lambda: hits.append("a"))
                S.on(self, "next", lambda: S.goto_state("b"))

            def state_b(self, S):
                S.on(src, "ping", lambda: hits.append("b"))

        w = Watcher()
        src.emit("ping")
        assert hits == ["a"]
        w.emit("next")
        src.emit("ping")
        assert hits == ["a", "b"]

    run_vt(lambda loop: body(loop))


def test_scoped_timeout_cancelled_on_exit():
    async def body(loop):
        fired = []

        class T(FSM):
            def __init__(self):
                super().__init__("a", loop=loop)

            def state_a(self, S):
                S.timeout(100, lambda: fired.append("a-timer"))
                S.on(self, "next", lambda: S.goto_state("b"))

            def state_b(self, S):
                S.timeout(50, lambda: fired.append("b-timer"))

        t = T()
        t.emit("next")
        await advance(loop, 1.0)
        assert fired == ["b-timer"]

    run_vt(lambda loop: body(loop))


def test_state_changed_emitted_async():
    async def body(loop):
        fsm = Light(loop)
        seen = []
        fsm.on("stateChanged", seen.append)
        fsm.emit("go")
        assert seen == []  # not delivered synchronously
        await settle(loop)
        # the initial state's own emission is also delivered async — the
        # pool claim path depends on catching the handle's initial
        # 'waiting' to run its first tryNext (lib/pool.js:922-927)
        assert seen == ["red", "green"]

    run_vt(lambda loop: body(loop))


def test_state_changed_delivered_to_late_listener():
    """A listener registered after the transition but before the flush
    still receives the pending event (lib/connection-fsm.js:1209-1216
    depends on this)."""
    async def body(loop):
        fsm = Light(loop)
        fsm.emit("go")  # transition done, emission pending
        seen = []
        fsm.on("stateChanged", seen.append)
        await settle(loop)
        assert seen == ["red", "green"]

    run_vt(lambda loop: body(loop))


def test_goto_during_entry_is_deferred_but_chains():
    order = []

    class Chain(FSM):
        def __init__(self, loop):
            super().__init__("a", loop=loop)

        def state_a(self, S):
            order.append("enter-a")
            S.goto_state("b")
            order.append("after-goto")

        def state_b(self, S):
            order.append("enter-b")

    async def body(loop):
        fsm = Chain(loop)
        assert fsm.get_state() == "b"
        assert order == ["enter-a", "after-goto", "enter-b"]

    run_vt(lambda loop: body(loop))


def test_multiple_transitions_all_emitted_in_order():
    async def body(loop):
        fsm = Light(loop)
        seen = []
        fsm.on("stateChanged", seen.append)
        fsm.emit("go")
        fsm.emit("stop")
        fsm.emit("go")
        await settle(loop)
        assert seen == ["red", "green", "red", "green"]

    run_vt(lambda loop: body(loop))


def test_dotted_substate():
    class Sub(FSM):
        def __init__(self, loop):
            super().__init__("running", loop=loop)

        def state_running(self, S):
            S.valid_transitions(["stopping"])
            S.on(self, "stop", lambda: S.goto_state("stopping"))

        def state_stopping(self, S):
            S.valid_transitions(["stopping.cleanup"])
            S.goto_state("stopping.cleanup")

        def state_stopping_cleanup(self, S):
            S.valid_transitions(["stopped"])
            S.on(self, "done", lambda: S.goto_state("stopped"))

        def state_stopped(self, S):
            S.valid_transitions([])

    async def body(loop):
        fsm = Sub(loop)
        fsm.emit("stop")
        assert fsm.get_state() == "stopping.cleanup"
        assert fsm.is_in_state("stopping")  # prefix semantics
        assert fsm.is_in_state("stopping.cleanup")
        fsm.emit("done")
        assert fsm.is_in_state("stopped")

    run_vt(lambda loop: body(loop))


def test_stale_scope_goto_ignored():
    """A handler whose state has already been exited during the same
    cascade must not cause a transition."""
    src = EventEmitter()

    class R(FSM):
        def __init__(self, loop):
            super().__init__("a", loop=loop)

        def state_a(self, S):
            S.valid_transitions(["b"])
            # two listeners for the same event: the first transitions
            # away; the second (still in the emit snapshot) must be a
            # no-op rather than a double transition.
            S.on(src, "evt", lambda: S.goto_state("b"))
            S.on(src, "evt", lambda: S.goto_state("b"))

        def state_b(self, S):
            S.valid_transitions([])

    async def body(loop):
        fsm = R(loop)
        src.emit("evt")
        assert fsm.get_state() == "b"
        assert fsm.get_state_history() == ["a", "b"]

    run_vt(lambda loop: body(loop))


def test_immediate_cancelled_on_exit():
    async def body(loop):
        fired = []

        class T(FSM):
            def __init__(self):
                super().__init__("a", loop=loop)

            def state_a(self, S):
                S.immediate(lambda: fired.append("a"))
                S.on(self, "next", lambda: S.goto_state("b"))

            def state_b(self, S):
                pass

        t = T()
        t.emit("next")  # leaves a before the immediate runs
        await settle(loop)
        assert fired == []

    run_vt(lambda loop: body(loop))


def test_scallback_noop_after_exit():
    async def body(loop):
        calls = []

        class T(FSM):
            def __init__(self):
                super().__init__("a", loop=loop)

            def state_a(self, S):
                self.cb = S.callback(lambda v: calls.append(v))
                S.on(self, "next", lambda: S.goto_state("b"))

            def state_b(self, S):
                pass

        t = T()
        t.cb(1)
        t.emit("next")
        t.cb(2)  # state a exited: must be ignored
        assert calls == [1]

    run_vt(lambda loop: body(loop))


def test_transition_tracer_hook():
    """set_transition_tracer: the mooremachine-DTrace-probe analog —
    fires synchronously on every transition of every FSM."""
    import cueball_amd.fsm as mod_fsm

    seen = []

    def tracer(fsm, state):
        seen.append((type(fsm).__name__, state))

    async def body(loop):
        mod_fsm.set_transition_tracer(tracer)
        try:
            fsm = Light(loop)
            fsm.emit("go")
            fsm.emit("stop")
        finally:
            mod_fsm.set_transition_tracer(None)
        assert [s for (n, s) in seen if n == "Light"] == \
            ["red", "green", "red"]
        # cleared: no more events
        n0 = len(seen)
        fsm.emit("go")
        assert len(seen) == n0

    run_vt(lambda loop: body(loop))


def test_scope_dispose_respects_remove_listener_override():
    """A user subclass overriding remove_listener must see the removals
    the state scope performs at exit (the native core's fast path only
    applies when the MRO resolves to the built-in method)."""
    removed = []

    class Audited(EventEmitter):
        def remove_listener(self, event, listener):
            removed.append(event)
            super().remove_listener(event, listener)

    async def body(loop):
        src = Audited()

        class W(FSM):
            def __init__(self):
                super().__init__("a", loop=loop)

            def state_a(self, S):
                S.on(src, "ping", lambda: None)
                S.on(self, "next", lambda: S.goto_state("b"))

            def state_b(self, S):
                pass

        w = W()
        w.emit("next")
        assert "ping" in removed
        assert src.listener_count("ping") == 0

    run_vt(lambda loop: body(loop))
