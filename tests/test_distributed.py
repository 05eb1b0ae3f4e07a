"""Multi-process (gloo, world_size=2) benchmark-path test.

The driver runs bench.py under torch.distributed.run with one rank per
GPU; this covers the same rendezvous + barrier + all-reduce path here
on CPU with two processes (gloo backend, 127.0.0.1 rendezvous).
"""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.timeout(300)
def test_bench_two_workers_gloo():
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--claims-per-step", "2000"],
        capture_output=True, text=True, timeout=280, cwd=ROOT, env=env)
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    data = json.loads(lines[0])
    assert data["n_gpus"] == 2
    # whole-job aggregate: 2 ranks x K steps x claims-per-step
    assert data["value"] > 0
    assert data["config"]["parallelism"].startswith("2 worker")
