"""Soak: sustained claims under backend churn, failures and recovery.

Real sockets, real clock.  The CPU version runs a few seconds; the
gpu-marked variant runs longer on the box.  Asserts liveness (claims
keep completing), no stuck waiters, and clean shutdown.
"""

import asyncio
import random

import pytest

from cueball_amd.connection import tcp_constructor
from cueball_amd.pool import ConnectionPool
from cueball_amd.testing import DummyResolver


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


async def start_backend(handlers):
    async def echo(reader, writer):
        try:
            while data := await reader.read(4096):
                writer.write(data)
                await writer.drain()
        except (ConnectionResetError, BrokenPipeError):
            pass
        finally:
            writer.close()

    srv = await asyncio.start_server(echo, "127.0.0.1", 0)
    handlers.append(srv)
    return srv, srv.sockets[0].getsockname()[1]


async def soak(duration, n_backends=6, churn_every=0.7, seed=42):
    rng = random.Random(seed)
    loop = asyncio.get_running_loop()
    servers = []
    backends = {}
    for i in range(n_backends):
        srv, port = await start_backend(servers)
        backends["b%d" % i] = (srv, port)

    resolver = DummyResolver()
    pool = ConnectionPool({
        "domain": "soak.test",
        "constructor": tcp_constructor(loop=loop),
        "resolver": resolver,
        "recovery": {"default": {"timeout": 1000, "retries": 2,
                                 "delay": 50, "maxDelay": 500}},
        "spares": 4,
        "maximum": 12,
        "loop": loop,
    })
    resolver.start()
    for k, (_, port) in backends.items():
        resolver.add(k, {"address": "127.0.0.1", "port": port})

    stats = {"ok": 0, "err": 0}
    stop = loop.time() + duration

    async def claimer():
        while loop.time() < stop:
            try:
                hdl, conn = await asyncio.wait_for(
                    pool.claim_async(), timeout=10)
            except Exception:
                stats["err"] += 1
                await asyncio.sleep(0.01)
                continue
            try:
                fut = loop.create_future()
                listener = conn.on(
                    "data", lambda d: fut.done() or fut.set_result(d))
                try:
                    conn.write(b"ping")
                    await asyncio.wait_for(fut, timeout=5)
                    stats["ok"] += 1
                    conn.remove_listener("data", listener)
                    hdl.release()
                except (asyncio.TimeoutError, ConnectionResetError,
                        OSError):
                    stats["err"] += 1
                    conn.remove_listener("data", listener)
                    hdl.close()
            except Exception:
                stats["err"] += 1
        return None

    async def churner():
        removed = {}
        while loop.time() < stop:
            await asyncio.sleep(churn_every)
            # remove a live backend or re-add a removed one
            if removed and (len(removed) >= 2 or rng.random() < 0.5):
                k, port = removed.popitem()
                resolver.add(k, {"address": "127.0.0.1", "port": port})
            else:
                k = rng.choice([k for k in backends if k not in removed])
                removed[k] = backends[k][1]
                resolver.remove(k)
        for k, port in removed.items():
            resolver.add(k, {"address": "127.0.0.1", "port": port})

    tasks = [asyncio.ensure_future(claimer()) for _ in range(8)]
    tasks.append(asyncio.ensure_future(churner()))
    await asyncio.gather(*tasks)

    # liveness: plenty of successful round trips, pool still healthy
    assert stats["ok"] > duration * 50, stats
    assert pool.is_in_state("running"), pool.get_state()
    hdl, conn = await asyncio.wait_for(pool.claim_async(), timeout=10)
    hdl.release()

    pool.stop()
    t0 = loop.time()
    while not pool.is_in_state("stopped") and loop.time() - t0 < 15:
        await asyncio.sleep(0.05)
    assert pool.is_in_state("stopped")
    for srv in servers:
        srv.close()
    await asyncio.sleep(0.1)
    return stats


def test_soak_short():
    stats = run(soak(3.0))
    assert stats["ok"] > 150


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_soak_long_box():
    stats = run(soak(20.0))
    assert stats["ok"] > 2000
