"""Observability tour: kang snapshot + Prometheus metrics over HTTP.

Runs self-contained: starts a mock HTTP backend, an HttpAgent with a
shared metrics collector, fires a few requests, then scrapes the
KangServer's /kang/snapshot and /metrics endpoints the way an operator
(or Prometheus) would.
"""

import asyncio
import json
import os
import sys
import urllib.request

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cueball_amd
from cueball_amd.kang import KangServer
from cueball_amd.metrics import create_collector
from cueball_amd.testing import MockHttpServer

RECOVERY = {"default": {"timeout": 2000, "retries": 3, "delay": 200,
                        "maxDelay": 2000}}


async def main():
    srv = MockHttpServer()
    await srv.start()

    collector = create_collector()
    agent = cueball_amd.HttpAgent({
        "defaultPort": srv.port,
        "recovery": RECOVERY,
        "spares": 2,
        "maximum": 4,
        "collector": collector,
    })

    for i in range(5):
        resp = await agent.request_async("127.0.0.1", "GET", "/item/%d" % i)
        assert resp.status_code == 200

    kang = KangServer(collector=collector)
    await kang.start()

    def fetch(path):
        with urllib.request.urlopen(
                "http://127.0.0.1:%d%s" % (kang.port, path), timeout=5) as r:
            return r.read().decode()

    loop = asyncio.get_running_loop()
    snapshot = json.loads(await loop.run_in_executor(
        None, fetch, "/kang/snapshot"))
    print("kang service:", snapshot["service"]["name"])
    for uuid, pool in snapshot["pool"].items():
        print("pool %s state=%s backends=%d counters=%s"
              % (uuid[:8], pool["state"], len(pool["backends"]),
                 {k: v for k, v in sorted(pool["counters"].items())[:3]}))

    await asyncio.sleep(0.5)  # let the 5 Hz gauge tick publish
    metrics = await loop.run_in_executor(None, fetch, "/metrics")
    shown = 0
    for line in metrics.splitlines():
        if line.startswith("cueball_") and not line.startswith("#"):
            print("metric:", line)
            shown += 1
            if shown >= 5:
                break

    kang.stop()
    fut = loop.create_future()
    agent.stop(lambda e: fut.set_result(None))
    await asyncio.wait_for(fut, 10)
    srv.stop()
    print("OK")


if __name__ == "__main__":
    asyncio.run(main())
