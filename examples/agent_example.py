"""Keep-alive HTTP through HttpAgent against a local server."""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cueball_amd
from cueball_amd.testing import MockHttpServer


async def main():
    srv = MockHttpServer()
    await srv.start()

    agent = cueball_amd.HttpAgent({
        "defaultPort": srv.port,
        "spares": 2,
        "maximum": 4,
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 200, "maxDelay": 2000}},
        "ping": "/ping",
        "pingInterval": 5000,
    })

    for i in range(3):
        resp = await agent.request_async("127.0.0.1", "GET", "/item/%d" % i)
        print(resp.status_code, resp.body.decode())

    print("server connections used:", srv.conn_count, "(keep-alive reuse)")

    done = asyncio.get_running_loop().create_future()
    agent.stop(lambda err: done.set_result(None))
    await done
    srv.stop()


if __name__ == "__main__":
    asyncio.run(main())
