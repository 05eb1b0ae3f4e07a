#!/usr/bin/env python3
"""Head-to-head runner: node-cueball reference vs cueball_amd rebuild.

Runs the same scenarios with the same parameters through
tools/noderef/bench_ref.js (reference, via offline dependency shims)
and bench.py (rebuild) on the same host, sequentially, and writes a
JSON summary + markdown table.

Usage:
    python tools/noderef/compare.py [--quick] [--out profiles/headtohead]

--quick uses short step counts for smoke runs; the full run is sized
to give multi-second timed regions per scenario.
"""

import argparse
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
SHIMS = os.path.join(REPO, "tools", "noderef", "shims")
BENCH_REF = os.path.join(REPO, "tools", "noderef", "bench_ref.js")
BENCH_REF_MULTI = os.path.join(REPO, "tools", "noderef",
                               "bench_ref_multi.js")
BENCH_PY = os.path.join(REPO, "bench.py")


def run_json(cmd, env=None, timeout=900):
    """Run cmd, return the last JSON line of stdout."""
    e = dict(os.environ)
    if env:
        e.update(env)
    proc = subprocess.run(cmd, capture_output=True, text=True,
                          timeout=timeout, env=e, cwd=REPO)
    line = None
    for ln in proc.stdout.strip().splitlines():
        ln = ln.strip()
        if ln.startswith("{") and ln.endswith("}"):
            line = ln
    if proc.returncode != 0 or line is None:
        sys.stderr.write("FAILED: %s\nstdout: %s\nstderr: %s\n"
                         % (" ".join(cmd), proc.stdout[-2000:],
                            proc.stderr[-2000:]))
        return None
    return json.loads(line)


def ref_cmd(config, steps, warmup, claims, workers=1, extra=()):
    script = BENCH_REF_MULTI if workers > 1 else BENCH_REF
    cmd = ["node", script]
    if workers > 1:
        cmd += ["--workers", str(workers)]
    cmd += ["--config", config, "--steps", str(steps),
            "--warmup", str(warmup), "--claims-per-step", str(claims)]
    cmd += list(extra)
    return cmd


def py_cmd(config, steps, warmup, claims, workers=1, extra=()):
    if workers > 1:
        port = 29640 + workers
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", "--nproc-per-node", str(workers),
               "--master-addr", "127.0.0.1",
               "--master-port", str(port), BENCH_PY]
    else:
        cmd = [sys.executable, BENCH_PY]
    cmd += ["--gpus", str(workers), "--config", config,
            "--steps", str(steps), "--warmup", str(warmup),
            "--claims-per-step", str(claims)]
    cmd += list(extra)
    return cmd


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--out", default="profiles/headtohead")
    ap.add_argument("--skip-scale", action="store_true")
    ap.add_argument("--max-workers", type=int, default=8)
    args = ap.parse_args()

    if args.quick:
        steps, warmup, claims = 3, 1, 10000
        agent_steps, agent_conc = 3, 200
        cset_steps = 4
        codel_claims = 10000
    else:
        steps, warmup, claims = 10, 2, 20000
        agent_steps, agent_conc = 10, 1000
        cset_steps = 12
        codel_claims = 20000

    # Hosts without /root/reference (e.g. the benchmark box) get the
    # vendored copy of the reference sources (see README.md here).
    if not os.environ.get("CUEBALL_REF") and \
            not os.path.isdir("/root/reference/lib"):
        import tarfile
        import tempfile
        tgz = os.path.join(REPO, "tools", "noderef",
                           "reference-cueball-2.10.3.tgz")
        tmp = tempfile.mkdtemp(prefix="noderef-")
        with tarfile.open(tgz) as tf:
            tf.extractall(tmp)
        os.environ["CUEBALL_REF"] = os.path.join(tmp, "reference")
        print("extracted reference to %s" % os.environ["CUEBALL_REF"],
              flush=True)

    results = {"host": os.uname().nodename,
               "node": None, "python": sys.version.split()[0],
               "scenarios": {}, "scaling": {}}

    scenarios = [
        ("headline", steps, warmup, claims, ()),
        ("static1", steps, warmup, claims, ()),
        ("dns", steps, warmup, claims, ()),
        ("codel", 1, 0, codel_claims, ()),
        ("agent", agent_steps, 1, claims,
         ("--agent-concurrency", str(agent_conc))),
        ("cset", cset_steps, 0, claims,
         ("--churn-interval", "1.5")),
    ]

    for (cfg, st, wu, cl, extra) in scenarios:
        print("== scenario %s ==" % cfg, flush=True)
        ref = run_json(ref_cmd(cfg, st, wu, cl, extra=extra),
                       env={"NODE_PATH": SHIMS})
        reb = run_json(py_cmd(cfg, st, wu, cl, extra=extra))
        if ref is not None:
            results["node"] = ref.get("node")
        entry = {
            "reference": ref and {
                "claims_per_s": ref["value"],
                "p50_ms": ref["config"].get("claim_latency_p50_ms"),
                "p99_ms": ref["config"].get("claim_latency_p99_ms"),
                "shed": ref["config"].get("shed_claims"),
            },
            "rebuild": reb and {
                "claims_per_s": reb["value"],
                "p50_ms": reb["config"].get("claim_latency_p50_ms"),
                "p99_ms": reb["config"].get("claim_latency_p99_ms"),
                "shed": reb["config"].get("shed_claims"),
            },
        }
        if ref and reb:
            entry["speedup"] = round(reb["value"] / ref["value"], 3)
        results["scenarios"][cfg] = entry
        print(json.dumps(entry), flush=True)

    if not args.skip_scale:
        ws = [w for w in (1, 2, 4, 8) if w <= args.max_workers]
        for w in ws:
            print("== scaling headline x%d ==" % w, flush=True)
            ref = run_json(
                ref_cmd("headline", steps, warmup, claims, workers=w),
                env={"NODE_PATH": SHIMS})
            reb = run_json(
                py_cmd("headline", steps, warmup, claims, workers=w))
            entry = {
                "reference": ref and {"claims_per_s": ref["value"]},
                "rebuild": reb and {"claims_per_s": reb["value"]},
            }
            if ref and reb:
                entry["speedup"] = round(reb["value"] / ref["value"], 3)
            results["scaling"]["%dw" % w] = entry
            print(json.dumps(entry), flush=True)

    out_json = os.path.join(REPO, args.out + ".json")
    os.makedirs(os.path.dirname(out_json), exist_ok=True)
    with open(out_json, "w") as f:
        json.dump(results, f, indent=2)

    # markdown table
    lines = ["# Head-to-head: node-cueball %s vs cueball_amd (same host)"
             % (results["node"] or "?"),
             "",
             "| scenario | reference claims/s | rebuild claims/s | "
             "speedup | ref p50/p99 ms | rebuild p50/p99 ms |",
             "|---|---|---|---|---|---|"]
    for cfg, e in results["scenarios"].items():
        r, b = e.get("reference"), e.get("rebuild")

        def fmt(x):
            return "%.0f" % x["claims_per_s"] if x else "FAIL"

        def lat(x):
            if not x or x.get("p50_ms") is None:
                return "-"
            return "%.3f / %.3f" % (x["p50_ms"], x["p99_ms"] or -1)

        lines.append("| %s | %s | %s | %s | %s | %s |"
                     % (cfg, fmt(r), fmt(b),
                        ("%.2fx" % e["speedup"])
                        if "speedup" in e else "-",
                        lat(r), lat(b)))
    if results["scaling"]:
        lines += ["", "| workers | reference claims/s | "
                  "rebuild claims/s | speedup |", "|---|---|---|---|"]
        for w, e in results["scaling"].items():
            r, b = e.get("reference"), e.get("rebuild")
            lines.append("| %s | %s | %s | %s |"
                         % (w,
                            "%.0f" % r["claims_per_s"] if r else "FAIL",
                            "%.0f" % b["claims_per_s"] if b else "FAIL",
                            ("%.2fx" % e["speedup"])
                            if "speedup" in e else "-"))
    with open(os.path.join(REPO, args.out + ".md"), "w") as f:
        f.write("\n".join(lines) + "\n")
    print("\n".join(lines))


if __name__ == "__main__":
    main()
