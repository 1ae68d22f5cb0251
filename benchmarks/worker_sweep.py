#!/usr/bin/env python3
"""BASELINE.md harness: sweep MaxConcurrentReconciles over 1/2/4/8 workers.

Runs bench.py once per worker count (the reference's `max-workers` flag /
controller-runtime MaxConcurrentReconciles — healthcheck_controller.go:298)
and writes a JSONL + markdown summary. This is the measured substitute for the
reference's nonexistent published numbers ("the reference must
be measured side-by-side ... to establish the bar").

Usage: python benchmarks/worker_sweep.py [--crs 1000] [--steps 5] [--out DIR]
"""
import argparse
import json
import pathlib
import subprocess
import sys

REPO = pathlib.Path(__file__).resolve().parents[1]


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--crs", type=int, default=1000)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--workers", default="1,2,4,8")
    ap.add_argument("--out", default=str(REPO / "benchmarks"))
    args = ap.parse_args()

    out_dir = pathlib.Path(args.out)
    out_dir.mkdir(parents=True, exist_ok=True)
    rows = []
    for w in [int(x) for x in args.workers.split(",")]:
        proc = subprocess.run(
            [sys.executable, str(REPO / "bench.py"), "--crs", str(args.crs),
             "--steps", str(args.steps), "--warmup", str(args.warmup),
             "--workers", str(w)],
            capture_output=True, text=True, cwd=REPO,
        )
        if proc.returncode != 0:
            print(f"workers={w}: FAILED\n{proc.stderr[-1000:]}", file=sys.stderr)
            return 1
        data = json.loads([l for l in proc.stdout.splitlines() if l.startswith("{")][-1])
        rows.append(data)
        cfg = data["config"]
        print(f"workers={w}: {data['value']} cycles/s, "
              f"p50={cfg['p50_reconcile_latency_ms']}ms "
              f"p99={cfg['p99_reconcile_latency_ms']}ms")

    jsonl = out_dir / "worker_sweep.jsonl"
    jsonl.write_text("".join(json.dumps(r) + "\n" for r in rows))

    md = out_dir / "worker_sweep.md"
    lines = [
        "# MaxConcurrentReconciles sweep (BASELINE.md 1/2/4/8 axis)",
        "",
        f"{args.crs} concurrent HealthCheck CRs per run, mixed interval/cron/remedy fleet.",
        "",
        "| workers | cycles/s | p50 reconcile (ms) | p99 reconcile (ms) | ms/wave |",
        "|---|---|---|---|---|",
    ]
    for r in rows:
        c = r["config"]
        lines.append(
            f"| {c['workers_per_rank']} | {r['value']} | "
            f"{c['p50_reconcile_latency_ms']} | {c['p99_reconcile_latency_ms']} | "
            f"{r['ms_per_step']} |"
        )
    md.write_text("\n".join(lines) + "\n")
    print(f"wrote {jsonl} and {md}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
