"""ConnectionSet: every connection is advertised to the consumer
(added/removed contract) — the shape used by multiplexing protocols."""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cueball_amd
from cueball_amd.connection import tcp_constructor


async def main():
    async def echo(reader, writer):
        while data := await reader.read(4096):
            writer.write(data)
            await writer.drain()
        writer.close()

    servers = [await asyncio.start_server(echo, "127.0.0.1", 0)
               for _ in range(3)]
    ports = [s.sockets[0].getsockname()[1] for s in servers]

    resolver = cueball_amd.StaticIpResolver({
        "backends": [{"address": "127.0.0.1", "port": p} for p in ports],
    })
    cset = cueball_amd.ConnectionSet({
        "resolver": resolver,
        "constructor": tcp_constructor(),
        "target": 3,
        "maximum": 5,
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 200, "maxDelay": 2000}},
    })

    live = {}

    def on_added(ckey, conn, hdl):
        print("added:", ckey)
        live[ckey] = (conn, hdl)

    def on_removed(ckey, conn, hdl):
        print("removed:", ckey)
        live.pop(ckey, None)
        hdl.release()  # we have no in-flight work to drain

    cset.on("added", on_added)
    cset.on("removed", on_removed)
    resolver.start()

    while len(live) < 3:
        await asyncio.sleep(0.05)
    print("set established with", len(live), "connections")

    cset.stop()
    await asyncio.sleep(0.5)
    for s in servers:
        s.close()


if __name__ == "__main__":
    asyncio.run(main())
