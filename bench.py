#!/usr/bin/env python3
"""Benchmark harness for the cueball_amd framework.

Measures the BASELINE.json headline metric: **pool claims/sec and
p50/p99 claim latency on an 8-backend synthetic set**, at 1..N worker
processes (one process per --gpus "worker"; the workload is host-CPU
event-loop work — the reference is a connection-pool library with no
GPU compute).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`.
For N>1 the driver launches via torch.distributed.run with one rank per
GPU; ranks coordinate with gloo barriers (the payload is CPU work) and
rank 0 prints ONE JSON line with the whole-job aggregate.

A "step" = --claims-per-step claim/release operations through the pool
(default 20000), driven by --concurrency concurrent claimers over 8
local TCP backends.

Scenario configs (BASELINE.json "configs"): headline (default),
static1, dns, cset, codel, agent — select with --config.
"""

import argparse
import asyncio
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from cueball_amd.connection import tcp_constructor  # noqa: E402
from cueball_amd.pool import ConnectionPool  # noqa: E402
from cueball_amd.resolver import StaticIpResolver  # noqa: E402


# ---------------------------------------------------------------------------
# infrastructure

class _EchoProtocol(asyncio.Protocol):
    """Raw-protocol echo: the asyncio equivalent of node's
    ``socket.pipe(socket)`` (no streams layer on the hot path)."""

    def connection_made(self, transport):
        self.transport = transport

    def data_received(self, data):
        self.transport.write(data)

    def connection_lost(self, exc):
        pass


async def start_backends(n, port0=0):
    """n local TCP echo servers; returns list of (server, port)."""
    servers = []
    loop = asyncio.get_running_loop()
    for i in range(n):
        srv = await loop.create_server(_EchoProtocol, "127.0.0.1", 0)
        port = srv.sockets[0].getsockname()[1]
        servers.append((srv, port))
    return servers


def make_pool(backends, spares, maximum, loop, target_claim_delay=None):
    opts = {
        "domain": "bench.local",
        "constructor": tcp_constructor(loop=loop),
        "resolver": StaticIpResolver({
            "backends": [{"address": "127.0.0.1", "port": p}
                         for (_, p) in backends],
            "loop": loop,
        }),
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 100, "maxDelay": 2000}},
        "spares": spares,
        "maximum": maximum,
        "loop": loop,
    }
    if target_claim_delay is not None:
        opts["targetClaimDelay"] = target_claim_delay
    pool = ConnectionPool(opts)
    opts["resolver"].start()
    return pool


async def wait_for_idle(pool, want, timeout=15.0):
    t0 = time.monotonic()
    while time.monotonic() - t0 < timeout:
        if pool.get_stats()["idleConnections"] >= want:
            return
        await asyncio.sleep(0.01)
    raise RuntimeError("pool never reached %d idle connections "
                       "(stats: %r)" % (want, pool.get_stats()))


class ClaimDriver:
    """C concurrent claim/release chains; counts ops + latencies."""

    def __init__(self, pool, loop, concurrency, latencies):
        self.pool = pool
        self.loop = loop
        self.concurrency = concurrency
        self.latencies = latencies
        self.ops = 0
        self.target = 0
        self.done_fut = None
        self.active = 0
        self.record_lat = True

    def run_step(self, n_ops):
        """Returns a future resolved when n_ops claims completed."""
        self.target = n_ops
        self.ops = 0
        self.done_fut = self.loop.create_future()
        for _ in range(self.concurrency):
            self.active += 1
            self._claim_once()
        return self.done_fut

    def _claim_once(self):
        t0 = self.loop.time()

        def cb(err, hdl=None, conn=None):
            if err is not None:
                # overload shed / timeout: count as completed op
                pass
            else:
                hdl.release()
            if self.record_lat:
                self.latencies.append(self.loop.time() - t0)
            self.ops += 1
            if self.ops + self.active - 1 < self.target:
                self._claim_once()
            else:
                self.active -= 1
                if self.active == 0 and not self.done_fut.done():
                    self.done_fut.set_result(None)

        self.pool.claim({}, cb)


# ---------------------------------------------------------------------------
# scenarios

async def scenario_headline(args, results):
    """8-backend synthetic set, claim/release loop (BASELINE headline)."""
    backends = await start_backends(8)
    loop = asyncio.get_running_loop()
    pool = make_pool(backends, spares=args.spares, maximum=args.maximum,
                     loop=loop)
    await wait_for_idle(pool, min(args.spares, 8))

    lat = []
    driver = ClaimDriver(pool, loop, args.concurrency, lat)

    for _ in range(args.warmup):
        await driver.run_step(args.claims_per_step)

    lat.clear()
    barrier()
    cuda_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        await driver.run_step(args.claims_per_step)
    cuda_sync()
    t1 = time.perf_counter()
    barrier()

    results["elapsed"] = t1 - t0
    results["ops"] = args.steps * args.claims_per_step
    results["lat_p50"] = statistics.median(lat) * 1000 if lat else None
    results["lat_p99"] = (statistics.quantiles(lat, n=100)[98] * 1000
                          if len(lat) >= 100 else None)
    pool.stop()
    for srv, _ in backends:
        srv.close()
    await asyncio.sleep(0.1)


async def scenario_static1(args, results):
    """StaticResolver 1 backend, spares=1/max=2 (BASELINE config #1)."""
    backends = await start_backends(1)
    loop = asyncio.get_running_loop()
    pool = make_pool(backends, spares=1, maximum=2, loop=loop)
    await wait_for_idle(pool, 1)
    lat = []
    driver = ClaimDriver(pool, loop, 1, lat)
    for _ in range(args.warmup):
        await driver.run_step(args.claims_per_step)
    lat.clear()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        await driver.run_step(args.claims_per_step)
    t1 = time.perf_counter()
    results["elapsed"] = t1 - t0
    results["ops"] = args.steps * args.claims_per_step
    results["lat_p50"] = statistics.median(lat) * 1000 if lat else None
    results["lat_p99"] = (statistics.quantiles(lat, n=100)[98] * 1000
                          if len(lat) >= 100 else None)
    pool.stop()
    for srv, _ in backends:
        srv.close()
    await asyncio.sleep(0.1)


async def scenario_dns(args, results):
    """DNS-SRV resolver against local mock DNS, 4 backends, pool
    spares=2/max=8 (BASELINE config #2)."""
    from cueball_amd.resolver import DNSResolver, DNSResolverFSM
    from cueball_amd.testing import MockDnsServer

    loop = asyncio.get_running_loop()
    backends = await start_backends(4)
    dns = MockDnsServer()
    await dns.start()
    for i, (_, port) in enumerate(backends):
        dns.add_srv("_bench._tcp.svc.bench", "b%d.svc.bench" % i, port,
                    ttl=60)
        dns.add_a("b%d.svc.bench" % i, "127.0.0.1", ttl=60)

    DNSResolverFSM._nic_cache = {"lo": [{"family": "IPv4",
                                         "address": "127.0.0.1"}]}
    DNSResolverFSM._nic_cache_updated = loop.time() * 1000.0

    resolver = DNSResolver({
        "domain": "svc.bench",
        "service": "_bench._tcp",
        "resolvers": [dns.resolver_address],
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 100, "maxDelay": 2000}},
    })
    pool = ConnectionPool({
        "domain": "svc.bench",
        "constructor": tcp_constructor(loop=loop),
        "resolver": resolver,
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 100, "maxDelay": 2000}},
        "spares": 2,
        "maximum": 8,
        "loop": loop,
    })
    resolver.start()
    await wait_for_idle(pool, 2)
    lat = []
    driver = ClaimDriver(pool, loop, args.concurrency, lat)
    for _ in range(args.warmup):
        await driver.run_step(args.claims_per_step)
    lat.clear()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        await driver.run_step(args.claims_per_step)
    t1 = time.perf_counter()
    results["elapsed"] = t1 - t0
    results["ops"] = args.steps * args.claims_per_step
    results["lat_p50"] = statistics.median(lat) * 1000 if lat else None
    results["lat_p99"] = (statistics.quantiles(lat, n=100)[98] * 1000
                          if len(lat) >= 100 else None)
    pool.stop()
    resolver.stop()
    dns.stop()
    for srv, _ in backends:
        srv.close()
    await asyncio.sleep(0.1)


async def scenario_cset(args, results):
    """ConnectionSet under backend churn (BASELINE config #4): 6
    backends, remove/re-add 2 every churn interval under load."""
    from cueball_amd.connection_set import ConnectionSet
    from cueball_amd.testing import DummyResolver

    loop = asyncio.get_running_loop()
    backends = await start_backends(6)
    resolver = DummyResolver()
    cset = ConnectionSet({
        "constructor": tcp_constructor(loop=loop),
        "resolver": resolver,
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 100, "maxDelay": 2000}},
        "target": 6,
        "maximum": 8,
        "loop": loop,
    })
    live = {}

    def on_added(ck, conn, hdl):
        live[ck] = (conn, hdl)

    def on_removed(ck, conn, hdl):
        live.pop(ck, None)
        hdl.release()

    cset.on("added", on_added)
    cset.on("removed", on_removed)
    resolver.start()
    for i, (_, port) in enumerate(backends):
        resolver.add("b%d" % i, {"address": "127.0.0.1", "port": port})

    t_deadline = time.monotonic() + 10
    while len(live) < 6 and time.monotonic() < t_deadline:
        await asyncio.sleep(0.01)

    # sustained load: write/echo on every live conn; churn 2 backends
    ops = {"n": 0}

    async def load():
        while True:
            for ck, (conn, hdl) in list(live.items()):
                if conn.connected:
                    try:
                        conn.write(b"x" * 64)
                    except ConnectionResetError:
                        pass
                    ops["n"] += 1
            await asyncio.sleep(0)

    async def churn():
        i = 0
        while True:
            await asyncio.sleep(args.churn_interval)
            for j in (0, 1):
                k = "b%d" % ((i + j) % 6)
                resolver.remove(k)
            await asyncio.sleep(args.churn_interval)
            for j in (0, 1):
                k = "b%d" % ((i + j) % 6)
                _, port = backends[(i + j) % 6]
                resolver.add(k, {"address": "127.0.0.1", "port": port})
            i = (i + 2) % 6

    load_task = asyncio.ensure_future(load())
    churn_task = asyncio.ensure_future(churn())
    t0 = time.perf_counter()
    await asyncio.sleep(args.steps * 1.0)  # steps == seconds here
    t1 = time.perf_counter()
    load_task.cancel()
    churn_task.cancel()
    results["elapsed"] = t1 - t0
    results["ops"] = ops["n"]
    results["lat_p50"] = None
    results["lat_p99"] = None
    cset.stop()
    for srv, _ in backends:
        srv.close()
    await asyncio.sleep(0.2)


async def scenario_codel(args, results):
    """CoDel overload shed (BASELINE config #5): pool max=4, 10k queued
    claims, targetDelay=5ms."""
    backends = await start_backends(4)
    loop = asyncio.get_running_loop()
    pool = make_pool(backends, spares=4, maximum=4, loop=loop,
                     target_claim_delay=5)
    await wait_for_idle(pool, 4)

    n = args.claims_per_step
    outcomes = {"ok": 0, "shed": 0}
    lat = []
    done = loop.create_future()

    def make_cb(t0):
        def cb(err, hdl=None, conn=None):
            lat.append(loop.time() - t0)
            if err is None:
                outcomes["ok"] += 1
                # hold briefly to force queueing
                loop.call_later(0.0005, hdl.release)
            else:
                outcomes["shed"] += 1
            if outcomes["ok"] + outcomes["shed"] == n and not done.done():
                done.set_result(None)
        return cb

    t0 = time.perf_counter()
    for _ in range(n):
        pool.claim({}, make_cb(loop.time()))
    await done
    t1 = time.perf_counter()
    results["elapsed"] = t1 - t0
    results["ops"] = n
    results["lat_p50"] = statistics.median(lat) * 1000 if lat else None
    results["lat_p99"] = (statistics.quantiles(lat, n=100)[98] * 1000
                          if len(lat) >= 100 else None)
    results["shed"] = outcomes["shed"]
    pool.stop()
    for srv, _ in backends:
        srv.close()
    await asyncio.sleep(0.1)


class _FastHttpProtocol(asyncio.Protocol):
    """GET-only keep-alive HTTP/1.1 backend on a raw protocol (the
    Python analog of the bare node http.createServer the reference
    bench uses; the streams-based MockHttpServer stays for tests)."""

    RESP = (b"HTTP/1.1 200 OK\r\nContent-Length: 2\r\n"
            b"Connection: keep-alive\r\n\r\nok")

    def connection_made(self, transport):
        self.transport = transport
        self.buf = b""

    def data_received(self, data):
        buf = self.buf + data
        n = buf.count(b"\r\n\r\n")
        if n:
            self.transport.write(self.RESP * n)
            self.buf = buf[buf.rfind(b"\r\n\r\n") + 4:]
        else:
            self.buf = buf

    def connection_lost(self, exc):
        pass


class _FastHttpServer:
    def __init__(self, server, port):
        self._server = server
        self.port = port

    def stop(self):
        self._server.close()


async def scenario_agent(args, results):
    """HttpAgent keep-alive over 8 local HTTP backends, 1000 concurrent
    GETs per step (BASELINE config #3)."""
    from cueball_amd.agent import HttpAgent
    from cueball_amd.testing import DummyResolver

    loop = asyncio.get_running_loop()
    servers = []
    for _ in range(8):
        srv = await loop.create_server(_FastHttpProtocol, "127.0.0.1", 0)
        servers.append(_FastHttpServer(
            srv, srv.sockets[0].getsockname()[1]))

    resolver = DummyResolver()
    agent = HttpAgent({
        "defaultPort": servers[0].port,
        "recovery": {"default": {"timeout": 2000, "retries": 3,
                                 "delay": 100, "maxDelay": 2000}},
        "spares": 8,
        "maximum": 32,
        "loop": loop,
    })
    agent.create_pool("svc.bench", {"resolver": resolver})
    resolver.start()
    for i, s in enumerate(servers):
        resolver.add("b%d" % i, {"address": "127.0.0.1", "port": s.port,
                                 "name": "b%d" % i})
    pool = agent.get_pool("svc.bench")
    t_deadline = time.monotonic() + 15
    while pool.get_stats()["idleConnections"] < 8 and \
            time.monotonic() < t_deadline:
        await asyncio.sleep(0.01)

    concurrent = args.agent_concurrency
    lat = []

    async def one_get(i):
        t0 = loop.time()
        resp = await agent.request_async("svc.bench", "GET", "/%d" % i)
        lat.append(loop.time() - t0)
        assert resp.status_code == 200

    async def step():
        await asyncio.gather(*[one_get(i) for i in range(concurrent)])

    for _ in range(args.warmup):
        await step()
    lat.clear()
    barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        await step()
    t1 = time.perf_counter()
    barrier()
    results["elapsed"] = t1 - t0
    results["ops"] = args.steps * concurrent
    results["lat_p50"] = statistics.median(lat) * 1000 if lat else None
    results["lat_p99"] = (statistics.quantiles(lat, n=100)[98] * 1000
                          if len(lat) >= 100 else None)
    fut = loop.create_future()
    agent.stop(lambda e: fut.set_result(None))
    await fut
    for s in servers:
        s.stop()
    await asyncio.sleep(0.1)


SCENARIOS = {
    "headline": scenario_headline,
    "static1": scenario_static1,
    "dns": scenario_dns,
    "cset": scenario_cset,
    "codel": scenario_codel,
    "agent": scenario_agent,
}


# ---------------------------------------------------------------------------
# distributed plumbing

_DIST = {"on": False}


def dist_init():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(backend="gloo")
    _DIST["on"] = True
    return dist.get_rank(), world


def barrier():
    if _DIST["on"]:
        import torch.distributed as dist
        dist.barrier()


def cuda_sync():
    # only meaningful when torch is already in play (distributed runs /
    # GPU boxes); never pay the torch import just for a no-op sync
    import sys as _sys
    torch = _sys.modules.get("torch")
    if torch is None and (_DIST["on"] or os.environ.get("WORLD_SIZE")):
        import torch  # noqa: F811
    if torch is not None and torch.cuda.is_available():
        torch.cuda.synchronize()


def reduce_results(results, rank, world):
    """All-ranks aggregate: sum ops, max elapsed."""
    if not _DIST["on"]:
        return results
    import torch
    import torch.distributed as dist
    t = torch.tensor([results["elapsed"]], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    ops = torch.tensor([float(results["ops"])], dtype=torch.float64)
    dist.all_reduce(ops, op=dist.ReduceOp.SUM)
    out = dict(results)
    out["elapsed"] = float(t.item())
    out["ops"] = float(ops.item())
    return out


# The reference publishes no numbers, so the baseline is
# node-cueball itself measured head-to-head on the MI355X benchmark box
# via tools/noderef (profiles/headtohead_mi355x_v4.json).  Keyed by
# (config, workers); vs_baseline is null for unmeasured combinations.
_REFERENCE_MEASURED = {
    ("headline", 1): 137252.0,
    ("headline", 2): 282753.0,
    ("headline", 4): 546788.0,
    ("headline", 8): 1058642.0,
    ("static1", 1): 135854.0,
    ("dns", 1): 142658.0,
    ("codel", 1): 54712.0,
    ("agent", 1): 23478.0,
    ("cset", 1): 225797.0,
}


def _vs_baseline(config, world, value):
    ref = _REFERENCE_MEASURED.get((config, world))
    if ref is None:
        return None
    return round(value / ref, 3)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1,
                    help="worker count (driver: one rank per GPU)")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--config", default="headline",
                    choices=sorted(SCENARIOS.keys()))
    ap.add_argument("--claims-per-step", type=int, default=20000)
    ap.add_argument("--concurrency", type=int, default=16)
    ap.add_argument("--spares", type=int, default=8)
    ap.add_argument("--maximum", type=int, default=16)
    ap.add_argument("--churn-interval", type=float, default=5.0)
    ap.add_argument("--agent-concurrency", type=int, default=1000,
                    help="concurrent GETs in the agent scenario "
                         "(BASELINE config #3 says 1000)")
    args = ap.parse_args()

    rank, world = dist_init()

    results = {}
    asyncio.run(SCENARIOS[args.config](args, results))

    agg = reduce_results(results, rank, world)

    if rank == 0:
        value = agg["ops"] / agg["elapsed"]
        ms_per_step = (agg["elapsed"] / args.steps) * 1000.0
        out = {
            "metric": "pool claims/sec (8-backend synthetic TCP set)",
            "value": round(value, 1),
            "unit": "claims/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": _vs_baseline(args.config, world, value),
            "dtype": "n/a",
            "data": "synthetic",
            "config": {
                "model": "ConnectionPool claim/release (config: %s)"
                         % args.config,
                "global_batch": args.claims_per_step * world,
                "seq_len": 1,
                "parallelism": "%d worker processes (event loops)" % world,
                "backends": 8,
                "spares": args.spares,
                "maximum": args.maximum,
                "concurrency": args.concurrency,
                "claim_latency_p50_ms": (round(agg["lat_p50"], 4)
                                         if agg.get("lat_p50") else None),
                "claim_latency_p99_ms": (round(agg["lat_p99"], 4)
                                         if agg.get("lat_p99") else None),
            },
        }
        if "shed" in agg:
            out["config"]["shed_claims"] = agg["shed"]
        print(json.dumps(out))

    if _DIST["on"]:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
