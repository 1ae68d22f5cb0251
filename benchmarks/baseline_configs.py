#!/usr/bin/env python3
"""Measure the five BASELINE.json configurations explicitly (wire regime).

BASELINE.md names five shapes; bench.py's default covers config 5 (the
scaling-curve point). This harness runs all five through bench.py so the
committed evidence maps 1:1 onto the baseline list:

1. single inline-hello CR, 1 worker            (repeatAfterSec regime)
2. 10 cron CRs, 2 workers, cluster level
3. failing checks + remedy, 4 workers
4. 100 CRs (mixed cluster/namespace level), 8 workers
5. 1000 mixed CRs (49/30/20 repeat/cron/remedy), 8 workers — the default

Notes: the remedy-limits state machine (config 3's RemedyRunsLimit /
RemedyResetInterval) is exercised by the test suite; the bench's remedy CRs
run the always-remedy branch. Prometheus scrape (config 4) is served by the
controller CLI, not the bench harness. Everything else matches the named
shapes; each line is bench.py's standard JSON with p50/p99 included.
"""
import json
import pathlib
import subprocess
import sys

REPO = pathlib.Path(__file__).resolve().parents[1]

CONFIGS = [
    ("config1_single_cr_1w",
     ["--crs", "1", "--workers", "1", "--cron-frac", "0", "--remedy-frac", "0"]),
    ("config2_10cron_2w",
     ["--crs", "10", "--workers", "2", "--cron-frac", "1.0", "--remedy-frac", "0"]),
    ("config3_fail_remedy_4w",
     ["--crs", "10", "--workers", "4", "--cron-frac", "0", "--remedy-frac", "1.0"]),
    ("config4_100crs_8w",
     ["--crs", "100", "--workers", "8", "--cron-frac", "0.3", "--remedy-frac", "0"]),
    ("config5_1000mixed_8w",
     ["--crs", "1000", "--workers", "8"]),  # bench defaults = the named mix
]


def main() -> int:
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    lines = []
    for name, flags in CONFIGS:
        cmd = [sys.executable, str(REPO / "bench.py"),
               "--steps", str(args.steps), "--warmup", str(args.warmup)] + flags
        out = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                             cwd=REPO)
        if out.returncode != 0:
            print(f"{name}: FAILED\n{out.stderr[-1000:]}", file=sys.stderr)
            return 1
        data = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
        data["baseline_config"] = name
        lines.append(data)
        c = data["config"]
        print(f"{name}: {data['value']:.1f} cycles/s  "
              f"p50 {c['p50_reconcile_latency_ms']:.3f} ms  "
              f"p99 {c['p99_reconcile_latency_ms']:.2f} ms", flush=True)
    if args.out:
        with open(args.out, "w") as f:
            for d in lines:
                f.write(json.dumps(d) + "\n")
    return 0


if __name__ == "__main__":
    sys.exit(main())
