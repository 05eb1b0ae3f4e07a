"""Attach-on-demand debugging (cueball_amd.debug): the analog of the
reference's dtrace-attach stack-capture enablement (lib/utils.js:59-99).
"""

import os
import signal

import pytest

import cueball_amd
from cueball_amd import debug as mod_debug
from cueball_amd import utils as mod_utils
from cueball_amd.pool import ConnectionPool
from cueball_amd.resolver import ResolverFSM
from cueball_amd.testing import DummyConnection, DummyResolver, settle
from conftest import run_vt

RECOVERY = {"default": {"timeout": 500, "retries": 1, "delay": 0}}


@pytest.fixture(autouse=True)
def _cleanup():
    yield
    mod_debug.remove_attach_handler()
    mod_utils.disable_stack_traces()


def test_signal_toggles_capture_and_stacks_appear():
    async def body(loop):
        mod_debug.install_attach_handler(signal.SIGUSR2)
        assert mod_utils.stack_traces_enabled() is False

        resolver = DummyResolver()
        rfsm = ResolverFSM(resolver, {"loop": loop})
        conns = []

        def constructor(backend):
            c = DummyConnection(backend)
            conns.append(c)
            return c

        pool = ConnectionPool({
            "domain": "dbg",
            "constructor": constructor,
            "recovery": RECOVERY,
            "spares": 1,
            "maximum": 2,
            "resolver": rfsm,
            "loop": loop,
        })
        rfsm.start()
        resolver.add("b1", {})
        await settle(loop)
        for c in conns:
            c.connect()
        await settle(loop)

        # signal the live process: capture turns ON
        os.kill(os.getpid(), signal.SIGUSR2)
        await settle(loop)  # let the handler run
        assert mod_utils.stack_traces_enabled() is True

        box = {}
        pool.claim({}, lambda e, h=None, c=None: box.update(h=h))
        await settle(loop)
        hdl = box["h"]
        hdl.release()
        with pytest.raises(Exception) as ei:
            hdl.release()
        msg = str(ei.value)
        # a real frame from this test file must appear, not the
        # disabled-capture placeholder
        assert "stack traces disabled" not in msg
        assert "test_debug_attach" in msg

        # second signal: capture OFF again
        os.kill(os.getpid(), signal.SIGUSR2)
        await settle(loop)
        assert mod_utils.stack_traces_enabled() is False
        assert mod_debug.attach_state()["toggles"] == 2

        pool.stop()
        await settle(loop)

    run_vt(lambda loop: body(loop))


def test_env_enables_capture(monkeypatch):
    assert mod_utils.stack_traces_enabled() is False
    mod_debug.apply_env({"CUEBALL_STACK_TRACES": "1"})
    assert mod_utils.stack_traces_enabled() is True
    mod_utils.disable_stack_traces()
    # "0" and empty are no-ops
    mod_debug.apply_env({"CUEBALL_STACK_TRACES": "0"})
    assert mod_utils.stack_traces_enabled() is False


def test_env_installs_signal_handler():
    mod_debug.apply_env({"CUEBALL_DEBUG_SIGNAL": "USR1"})
    assert mod_debug.attach_state()["installed_for"] == signal.SIGUSR1
    os.kill(os.getpid(), signal.SIGUSR1)
    # handler runs synchronously for a non-asyncio process once the
    # interpreter checks signals
    import time
    time.sleep(0.01)
    assert mod_utils.stack_traces_enabled() is True
    mod_debug.remove_attach_handler()
    # removal restores the previous disposition and disables capture
    assert mod_utils.stack_traces_enabled() is False
    assert mod_debug.attach_state()["installed_for"] is None


def test_parse_signal():
    assert mod_debug._parse_signal("USR2") == signal.SIGUSR2
    assert mod_debug._parse_signal("SIGUSR1") == signal.SIGUSR1
    assert mod_debug._parse_signal("10") == 10
    assert mod_debug._parse_signal("") is None
    assert mod_debug._parse_signal("OFF") is None
    assert mod_debug._parse_signal("NOSUCHSIG") is None


def test_package_exports():
    assert cueball_amd.install_attach_handler is \
        mod_debug.install_attach_handler
