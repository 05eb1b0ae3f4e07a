"""Claim-latency distribution: deep percentiles for the headline path.

Runs the bench claim/release workload and prints
p50/p90/p99/p99.9/p99.99/max plus a coarse histogram — the evidence
artifact for tail-latency claims (the GC story in BASELINE.md).

Run: python tools/latency_hist.py [--claims N] [--concurrency C]
"""

import argparse
import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench  # noqa: E402


async def main(claims: int, concurrency: int) -> None:
    backends = await bench.start_backends(8)
    loop = asyncio.get_running_loop()
    pool = bench.make_pool(backends, spares=8, maximum=16, loop=loop)
    await bench.wait_for_idle(pool, 8)

    lat = []
    drv = bench.ClaimDriver(pool, loop, concurrency, lat)
    await drv.run_step(min(claims, 20000))  # warmup
    lat.clear()

    import time
    t0 = time.perf_counter()
    remaining = claims
    while remaining > 0:
        n = min(remaining, 50000)
        await drv.run_step(n)
        remaining -= n
    elapsed = time.perf_counter() - t0

    lat.sort()
    n = len(lat)

    def pct(p):
        return lat[min(n - 1, int(n * p))] * 1e6  # µs

    print("claims: %d at concurrency %d -> %.0f claims/s"
          % (n, concurrency, n / elapsed))
    print("p50    %8.1f us" % pct(0.50))
    print("p90    %8.1f us" % pct(0.90))
    print("p99    %8.1f us" % pct(0.99))
    print("p99.9  %8.1f us" % pct(0.999))
    print("p99.99 %8.1f us" % pct(0.9999))
    print("max    %8.1f us" % (lat[-1] * 1e6))

    # coarse log-ish histogram
    bounds_us = [25, 50, 100, 200, 500, 1000, 5000, float("inf")]
    counts = [0] * len(bounds_us)
    for v in lat:
        us = v * 1e6
        for i, b in enumerate(bounds_us):
            if us <= b:
                counts[i] += 1
                break
    lo = 0
    for b, c in zip(bounds_us, counts):
        label = ("<=%g us" % b) if b != float("inf") else (">%g us" % lo)
        print("%10s %9d (%.3f%%)" % (label, c, 100.0 * c / n))
        lo = b

    pool.stop()
    for s, _ in backends:
        s.close()
    await asyncio.sleep(0.1)


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--claims", type=int, default=500000)
    ap.add_argument("--concurrency", type=int, default=16)
    args = ap.parse_args()
    asyncio.run(main(args.claims, args.concurrency))
