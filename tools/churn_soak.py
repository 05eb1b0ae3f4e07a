"""Churn + failure soak: long-run stability of the recovery machinery.

Unlike tools/soak.py (steady claims against healthy backends), this
drives the paths that only matter over time: backends appear and
disappear, connections get killed mid-life, claims keep flowing, and
every cycle the pool must re-plan, declare/recover dead backends, and
keep its bookkeeping exact.  Prints a mark every 30 s with rates, pool
stats and RSS; exits non-zero if claims stop completing or the pool
wedges.

Run: python tools/churn_soak.py [--minutes M]
"""

import argparse
import asyncio
import os
import random
import resource
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cueball_amd.connection import tcp_constructor
from cueball_amd.pool import ConnectionPool
from cueball_amd.resolver import ResolverFSM
from cueball_amd.testing import DummyResolver


async def main(minutes: float) -> int:
    loop = asyncio.get_running_loop()

    # six possible backends; 3-6 alive at any time
    servers = []

    async def start_server():
        proto_factory = lambda: _Echo()  # noqa: E731
        srv = await loop.create_server(proto_factory, "127.0.0.1", 0)
        return srv, srv.sockets[0].getsockname()[1]

    class _Echo(asyncio.Protocol):
        def connection_made(self, tr):
            self.tr = tr

        def data_received(self, data):
            self.tr.write(data)

        def connection_lost(self, exc):
            pass

    for _ in range(6):
        servers.append(await start_server())

    resolver = DummyResolver()
    rfsm = ResolverFSM(resolver, {})
    pool = ConnectionPool({
        "domain": "churn.soak",
        "constructor": tcp_constructor(loop=loop),
        "resolver": rfsm,
        "recovery": {"default": {"timeout": 500, "retries": 2,
                                 "delay": 50, "maxDelay": 500}},
        "spares": 4,
        "maximum": 8,
        "loop": loop,
    })
    rfsm.start()
    alive = set()
    for i in range(4):
        resolver.add("b%d" % i, {"address": "127.0.0.1",
                                 "port": servers[i][1]})
        alive.add(i)

    counters = {"ok": 0, "err": 0, "stall": 0}

    stop_flag = {"stop": False}

    async def claimer():
        while not stop_flag["stop"]:
            try:
                hdl, conn = await asyncio.wait_for(
                    pool.claim_async({"timeout": 2000}), timeout=5)
                counters["ok"] += 1
                if random.random() < 0.01:
                    # simulate user killing the connection
                    hdl.close()
                else:
                    hdl.release()
            except asyncio.TimeoutError:
                counters["stall"] += 1
            except Exception:
                counters["err"] += 1
                await asyncio.sleep(0.01)

    async def churner():
        i = 0
        while not stop_flag["stop"]:
            await asyncio.sleep(random.uniform(1.0, 3.0))
            i = (i + 1) % 6
            if i in alive and len(alive) > 3:
                resolver.remove("b%d" % i)
                alive.discard(i)
            elif i not in alive:
                # maybe restart the listener on a fresh port first
                if random.random() < 0.5:
                    servers[i][0].close()
                    servers[i] = await start_server()
                resolver.add("b%d" % i, {"address": "127.0.0.1",
                                         "port": servers[i][1]})
                alive.add(i)

    async def killer():
        """Kill a live server socket occasionally: conns die mid-life,
        the slot retries, maybe the backend goes dead + monitored."""
        while not stop_flag["stop"]:
            await asyncio.sleep(random.uniform(2.0, 5.0))
            i = random.choice(sorted(alive)) if alive else None
            if i is not None and random.random() < 0.5:
                servers[i][0].close()
                servers[i] = await start_server()
                # note: resolver still points at the OLD port => the
                # backend looks dead until churner re-adds it

    tasks = [asyncio.ensure_future(claimer()) for _ in range(8)]
    tasks.append(asyncio.ensure_future(churner()))
    tasks.append(asyncio.ensure_future(killer()))

    t0 = time.monotonic()
    last_ok = 0
    rc = 0
    while time.monotonic() - t0 < minutes * 60:
        await asyncio.sleep(30)
        stats = pool.get_stats()
        rss = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024.0
        d_ok = counters["ok"] - last_ok
        last_ok = counters["ok"]
        print("mark %4.0fs: ok %d (+%d), err %d, stall %d, "
              "conns %d (idle %d), dead %d, rss %.1f MB"
              % (time.monotonic() - t0, counters["ok"], d_ok,
                 counters["err"], counters["stall"],
                 stats["totalConnections"], stats["idleConnections"],
                 len(pool.p_dead), rss), flush=True)
        if d_ok == 0:
            print("FAIL: no claims completed in the last 30s", flush=True)
            rc = 1
            break

    stop_flag["stop"] = True
    for t in tasks:
        t.cancel()
    await asyncio.sleep(0.2)
    pool.stop()
    for srv, _ in servers:
        srv.close()
    await asyncio.sleep(0.2)
    print("FINAL: ok %d err %d stall %d in %.0f s; pool %s"
          % (counters["ok"], counters["err"], counters["stall"],
             time.monotonic() - t0, pool.get_state()), flush=True)
    return rc


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=5.0)
    args = ap.parse_args()
    sys.exit(asyncio.run(main(args.minutes)))
