"""Stock aiohttp code with cueball pooling underneath.

The only change to an existing aiohttp codebase is the connector
argument — requests, sessions, timeouts and streaming work unchanged,
while connections come from a cueball ConnectionPool (spares/maximum
sizing, recovery backoff, DNS service discovery for hostnames, CoDel
shedding if target_claim_delay is set, Kang/Prometheus introspection).

Run: python examples/aiohttp_example.py
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import aiohttp

from cueball_amd.aiohttp_connector import CueballConnector
from cueball_amd.testing import MockHttpServer


async def main():
    # stand-in backend; point at any real service instead
    srv = MockHttpServer()
    await srv.start()
    base = "http://127.0.0.1:%d" % srv.port

    connector = CueballConnector(
        recovery={"default": {"timeout": 2000, "retries": 3,
                              "delay": 100, "maxDelay": 2000}},
        spares=2, maximum=8)

    async with aiohttp.ClientSession(connector=connector) as sess:
        # ordinary aiohttp calls — nothing cueball-specific here
        for i in range(5):
            async with sess.get("%s/item/%d" % (base, i)) as resp:
                data = await resp.json(content_type=None)
                print("GET /item/%d -> %d %r"
                      % (i, resp.status, data))

        print("cueball pool stats:", connector.pool_stats())
        print("server saw %d requests over %d connection(s) "
              "(keep-alive reuse)"
              % (srv.request_count, srv.conn_count))

    srv.stop()


if __name__ == "__main__":
    asyncio.run(main())
