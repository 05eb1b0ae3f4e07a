"""GPU-box validation tests (run with -m gpu on a real MI355X box).

The framework is host-side (the reference is a pure connection-pool
library), so these verify that the full stack works on
the deployment box: native extension loads, end-to-end claim path over
real sockets, a short throughput sanity run, and — since the box has a
GPU — that tensors round-trip on cuda:0 alongside the event loop.
"""

import asyncio
import json
import subprocess
import sys
import os

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_native_extension_loads():
    """The native runtime core must be loaded on the box — no silent
    pure-Python fallback (the .so is built in-tree and ships with the
    snapshot)."""
    from cueball_amd import _speed  # noqa: F401
    import cueball_amd.fsm as fsm
    import cueball_amd.events as events
    assert fsm.NATIVE, "FSM core fell back to pure Python"
    assert events.NATIVE, "EventEmitter fell back to pure Python"
    assert fsm.FSM is _speed.FSM


def test_smoke_entrypoint():
    sys.path.insert(0, ROOT)
    import __graft_entry__
    __graft_entry__.smoke()


def test_torch_cuda_alongside_event_loop():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU visible")

    async def body():
        t = torch.randn(1 << 20, device="cuda:0")
        s = float(t.float().abs().sum().item())
        assert s > 0
        torch.cuda.synchronize()

    asyncio.run(body())


def test_bench_short_run():
    """bench.py must produce a valid JSON line quickly on the box."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--steps", "3", "--warmup", "1", "--claims-per-step", "5000"],
        capture_output=True, text=True, timeout=300, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    data = json.loads(line)
    assert data["metric"].startswith("pool claims/sec")
    # perf floor: the box measures ~100k claims/s with the native core
    # (profiles/bench_final_headline.json); 30k guards against silent
    # regressions (e.g. falling back to pure Python) with headroom for
    # noise
    assert data["value"] > 30000, data
    assert data["config"]["claim_latency_p50_ms"] is not None
