"""Claim/release a TCP connection pool against local echo backends.

Runs self-contained: starts two local echo servers, discovers them via
StaticIpResolver, claims a connection, echoes a payload, releases, and
shows pool stats + a kang snapshot.
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cueball_amd
from cueball_amd.connection import tcp_constructor
from cueball_amd.kang import KangServer

RECOVERY = {"default": {"timeout": 2000, "retries": 3, "delay": 200,
                        "maxDelay": 2000}}


async def main():
    loop = asyncio.get_running_loop()

    async def echo(reader, writer):
        while data := await reader.read(4096):
            writer.write(data)
            await writer.drain()
        writer.close()

    servers = [await asyncio.start_server(echo, "127.0.0.1", 0)
               for _ in range(2)]
    ports = [s.sockets[0].getsockname()[1] for s in servers]

    resolver = cueball_amd.StaticIpResolver({
        "backends": [{"address": "127.0.0.1", "port": p} for p in ports],
    })
    pool = cueball_amd.ConnectionPool({
        "domain": "echo.local",
        "resolver": resolver,
        "constructor": tcp_constructor(),
        "spares": 2,
        "maximum": 4,
        "recovery": RECOVERY,
    })
    resolver.start()

    handle, conn = await pool.claim_async()
    reply = loop.create_future()
    conn.on("data", lambda d: reply.done() or reply.set_result(d))
    conn.write(b"hello, cueball")
    print("echoed:", await reply)
    handle.release()

    print("stats:", pool.get_stats())

    kang = KangServer()
    await kang.start()
    print("kang snapshot types:", kang.snapshot()["types"])
    kang.stop()

    pool.stop()
    for s in servers:
        s.close()
    await asyncio.sleep(0.2)


if __name__ == "__main__":
    asyncio.run(main())
