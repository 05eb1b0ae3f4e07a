"""Endurance soak: sustained claim/release cycles over real sockets.

Run: python tools/soak.py [--minutes M] [--concurrency C] [--backends B]

Claims and releases continuously against a local MockHttpServer through
the full pool path (resolver -> slots -> TCP connections), printing a
rate + RSS mark every 10M claims so rate or memory drift over time is
visible.  Used for the endurance numbers recorded in BASELINE.md.
"""

import argparse
import asyncio
import gc
import os
import resource
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cueball_amd.connection import tcp_constructor
from cueball_amd.pool import ConnectionPool
from cueball_amd.resolver import StaticIpResolver
from cueball_amd.testing import MockHttpServer


async def main(minutes: float, concurrency: int, backends: int) -> None:
    srv = MockHttpServer()
    await srv.start()
    res = StaticIpResolver({"backends": [
        {"address": "127.0.0.1", "port": srv.port}
        for _ in range(backends)]})
    pool = ConnectionPool({
        "domain": "soak.test", "resolver": res,
        "constructor": tcp_constructor(),
        "spares": backends, "maximum": backends * 2,
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 100, "maxDelay": 1000}},
    })
    res.start()
    loop = asyncio.get_running_loop()
    t0 = loop.time()
    while pool.get_state() != "running" and loop.time() - t0 < 30:
        await asyncio.sleep(0.05)
    assert pool.get_state() == "running", pool.get_state()

    stop_at = time.monotonic() + minutes * 60.0
    count = 0
    mark = 10_000_000
    t_start = time.monotonic()

    async def worker():
        nonlocal count, mark
        while time.monotonic() < stop_at:
            fut = loop.create_future()

            def cb(err, hdl, conn, fut=fut):
                if err is not None:
                    fut.set_exception(err)
                else:
                    fut.set_result(hdl)

            pool.claim({"timeout": 5000}, cb)
            hdl = await fut
            hdl.release()
            count += 1
            if count >= mark:
                el = time.monotonic() - t_start
                rss = (resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
                       / 1024.0)
                print("mark %dM: %.0f claims/s, rss %.1f MB, gc %s"
                      % (count // 1_000_000, count / el, rss,
                         gc.get_count()), flush=True)
                mark += 10_000_000

    await asyncio.gather(*[worker() for _ in range(concurrency)])
    el = time.monotonic() - t_start
    rss = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024.0
    print("FINAL: %d claims in %.1f s = %.0f claims/s, rss %.1f MB"
          % (count, el, count / el, rss), flush=True)
    pool.stop()
    srv.stop()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=5.0)
    ap.add_argument("--concurrency", type=int, default=32)
    ap.add_argument("--backends", type=int, default=8)
    a = ap.parse_args()
    asyncio.run(main(a.minutes, a.concurrency, a.backends))
