"""Socket-event protocol corner cases of the Agent's request ticket
(lib/agent.js:296-396): agentRemove+abort and abort-after-free
interleavings, driven deterministically with scripted fakes."""

from cueball_amd.agent import _RequestTicket
from cueball_amd.events import EventEmitter
from conftest import run_vt


class Emitter(EventEmitter):
    """EventEmitter with attribute storage (the pure-Python emitter
    uses __slots__)."""

    def __init__(self):
        super().__init__()
        self.sock = None

    def on_socket(self, s):
        self.sock = s


class FakeHandle:
    def __init__(self):
        self.released = 0
        self.closed = 0
        self.leak_check_disabled = 0

    def release(self):
        self.released += 1

    def close(self):
        self.closed += 1

    def disable_release_leak_check(self):
        self.leak_check_disabled += 1


class FakeWaiter:
    def __init__(self):
        self.cancelled = 0

    def cancel(self):
        self.cancelled += 1


class FakePool:
    def __init__(self):
        self.claim_cb = None
        self.waiter = FakeWaiter()

    def claim(self, opts, cb):
        self.claim_cb = cb
        return self.waiter


class FakeAgent:
    cba_err_on_empty = False

    def __init__(self, loop):
        self._loop = loop


def mk(loop):
    agent = FakeAgent(loop)
    pool = FakePool()
    req = Emitter()
    ticket = _RequestTicket(agent, pool, req)
    hdl = FakeHandle()
    sock = Emitter()
    pool.claim_cb(None, hdl, sock)
    return ticket, req, sock, hdl, pool


def test_agent_remove_then_abort_then_close():
    async def body(loop):
        ticket, req, sock, hdl, pool = mk(loop)

        sock.emit("agentRemove")
        # after agentRemove the abort listener is gone: an abort must
        # NOT close the held lease (the socket belongs to the upgrade
        # user now; reference lib/agent.js:386-395)
        req.emit("abort")
        assert hdl.closed == 0
        assert hdl.released == 0

        # a stray 'free' must also be ignored (listener removed)
        sock.emit("free")
        assert hdl.released == 0

        # 'close' ends the lease: released with leak check disabled
        sock.emit("close")
        assert hdl.released == 1
        assert hdl.leak_check_disabled == 1
        assert hdl.closed == 0
        # idempotence: nothing left to fire
        sock.emit("close")
        sock.emit("free")
        assert hdl.released == 1

    run_vt(lambda loop: body(loop))


def test_abort_after_free_is_noop():
    async def body(loop):
        ticket, req, sock, hdl, pool = mk(loop)

        sock.emit("free")
        assert hdl.released == 1
        assert hdl.closed == 0

        # the request aborts after its socket was already freed: the
        # abort listener was removed by onFree; no double release, no
        # close (reference onFree removes 'abort',
        # lib/agent.js:377-385)
        req.emit("abort")
        assert hdl.released == 1
        assert hdl.closed == 0

        # stray close after free: listener was removed by onFree
        sock.emit("close")
        assert hdl.released == 1
        assert hdl.leak_check_disabled == 0

    run_vt(lambda loop: body(loop))


def test_abort_while_claimed_closes_and_mutes_events():
    async def body(loop):
        ticket, req, sock, hdl, pool = mk(loop)

        req.emit("abort")
        assert hdl.closed == 1
        assert hdl.released == 0

        # all socket listeners were removed on abort: free/close/
        # agentRemove must do nothing further
        sock.emit("free")
        sock.emit("close")
        sock.emit("agentRemove")
        assert hdl.closed == 1
        assert hdl.released == 0

    run_vt(lambda loop: body(loop))


def test_abort_before_claim_cancels_waiter():
    async def body(loop):
        agent = FakeAgent(loop)
        pool = FakePool()
        req = Emitter()
        _RequestTicket(agent, pool, req)

        req.emit("abort")
        assert pool.waiter.cancelled == 1
        # a second abort is harmless
        req.emit("abort")
        assert pool.waiter.cancelled == 1

    run_vt(lambda loop: body(loop))


def test_free_then_agent_remove_is_noop():
    async def body(loop):
        ticket, req, sock, hdl, pool = mk(loop)
        sock.emit("free")
        assert hdl.released == 1
        sock.emit("agentRemove")
        sock.emit("close")
        assert hdl.released == 1
        assert hdl.closed == 0

    run_vt(lambda loop: body(loop))
