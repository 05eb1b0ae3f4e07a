"""Stock httpx code with cueball pooling underneath.

The only change to an existing httpx codebase is the transport
argument; requests, sessions and streaming work unchanged while
connections come from cueball HttpAgent pools (pool-per-host sizing,
recovery backoff, DNS service discovery for hostnames, Kang and
Prometheus introspection).

Run: python examples/httpx_example.py
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import httpx

from cueball_amd.httpx_transport import CueballTransport
from cueball_amd.testing import MockHttpServer


async def main():
    srv = MockHttpServer()
    await srv.start()
    base = "http://127.0.0.1:%d" % srv.port

    transport = CueballTransport(
        recovery={"default": {"timeout": 2000, "retries": 3,
                              "delay": 100, "maxDelay": 2000}},
        spares=2, maximum=8)

    async with httpx.AsyncClient(transport=transport) as client:
        for i in range(5):
            r = await client.get("%s/item/%d" % (base, i))
            print("GET /item/%d -> %d %r" % (i, r.status_code, r.json()))

        print("cueball pool stats:", transport.pool_stats())
        print("server saw %d requests over %d connection(s) "
              "(keep-alive reuse)"
              % (srv.request_count, srv.conn_count))

    srv.stop()


if __name__ == "__main__":
    asyncio.run(main())
