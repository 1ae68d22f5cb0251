#!/usr/bin/env python3
"""Steady-state soak: a real timer-driven fleet (no bench waves).

Unlike bench.py — which drives explicit waves for deterministic step timing —
this runs the controller exactly as production would: CRs with short
``repeatAfterSec``, repeats fired by the controller's own timers, for a fixed
wall-clock duration. Reports sustained cycles/s, completion-latency
percentiles, and RSS growth (leak canary).

Usage: python benchmarks/soak.py [--crs 1000] [--repeat 5] [--duration 120]
"""
import argparse
import asyncio
import json
import os
import resource
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def rss_mb() -> float:
    """Current (not peak) resident set size."""
    try:
        with open("/proc/self/status") as f:
            for line in f:
                if line.startswith("VmRSS:"):
                    return int(line.split()[1]) / 1024.0
    except OSError:
        pass
    return resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024.0


async def main_async(args) -> dict:
    from active_monitor_amd.engine import Manager
    from active_monitor_amd.kube import MemoryApiServer, MemoryClient
    from active_monitor_amd.workflow import ScriptedWorkflowEngine

    engine = None
    apiserver_proc = None
    if args.apiserver == "wire":
        # real-wire regime (bench.py's default): separate apiserver process,
        # controller over 127.0.0.1 HTTP, the engine in the apiserver process
        from active_monitor_amd.kube.http import HttpClient

        apiserver_proc = await asyncio.create_subprocess_exec(
            sys.executable, "-m", "active_monitor_amd.kube.standalone",
            "--engine", "scripted-bench", "--remedy-frac", str(args.remedy_frac),
            "--engine-ttl", str(args.ttl),
            stdout=asyncio.subprocess.PIPE, stderr=asyncio.subprocess.DEVNULL,
        )
        line = await asyncio.wait_for(apiserver_proc.stdout.readline(), 60)
        url = json.loads(line[len(b"READY "):])["url"]
        server = None
        client = HttpClient(url, qps=0)
        await client.start()
    else:
        from active_monitor_amd.kube.standalone import bench_policy

        server = MemoryApiServer()
        client = MemoryClient(server)
        engine = ScriptedWorkflowEngine(client, policy=bench_policy(args.remedy_frac),
                                        ttl_seconds=args.ttl)
        await engine.start()
    manager = Manager(client, max_workers=args.workers)
    await manager.start()

    inline = (
        "apiVersion: argoproj.io/v1alpha1\nkind: Workflow\nspec:\n"
        "  entrypoint: e\n  templates:\n    - name: e\n      container:\n"
        "        command: [echo, ok]\n"
    )
    for i in range(args.crs):
        # names follow the bench convention (hc-NNNNN) so the scripted
        # engine's policy can fail the remedy-carrying fraction by name
        is_remedy = (i % 100) < args.remedy_frac * 100
        is_cron = (
            not is_remedy
            and (i % 100) < (args.remedy_frac + args.cron_frac) * 100
        )
        spec = {
            "level": "cluster",
            "workflow": {
                "generateName": f"hc-{i:05d}-wf-",
                "workflowtimeout": max(args.repeat, 5),
                "resource": {
                    "namespace": "health",
                    "serviceAccount": f"soak-sa-{i % 8}",
                    "source": {"inline": inline},
                },
            },
        }
        if is_cron:
            # the cron scheduler path: interval recomputed at every
            # completion (reference :251-263 semantics)
            spec["schedule"] = {"cron": f"@every {args.repeat}s"}
        else:
            spec["repeatAfterSec"] = args.repeat
        if is_remedy:
            spec["remedyworkflow"] = {
                "generateName": f"hc-{i:05d}-remedy-wf-",
                "workflowtimeout": max(args.repeat, 5),
                "resource": {
                    "namespace": "health",
                    "serviceAccount": f"soak-remedy-sa-{i}",
                    "source": {"inline": inline},
                },
            }
        await client.create({
            "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
            "kind": "HealthCheck",
            "metadata": {"name": f"hc-{i:05d}", "namespace": "health"},
            "spec": spec,
        })

    rec = manager.reconciler
    # settle: every CR has run once
    while rec.completed_runs < args.crs:
        await asyncio.sleep(0.05)
    rss_start = rss_mb()
    runs_start = rec.completed_runs
    t0 = time.monotonic()
    samples = []
    while time.monotonic() - t0 < args.duration:
        await asyncio.sleep(min(5.0, args.duration / 10))
        samples.append({
            "t": round(time.monotonic() - t0, 1),
            "completed": rec.completed_runs - runs_start,
            "rss_mb": round(rss_mb(), 1),
            "active_watches": rec.active_watches(),
            "queue": len(manager.queue),
            "objects": len(server) if server is not None else None,
            # leak canaries: bounded-structure sizes
            "hub_last": len(getattr(manager.wf_hub, "_last", ()) or ()),
            "hub_seq": len(getattr(manager.wf_hub, "_seq", ()) or ()),
            "agg": len(getattr(manager.recorder, "_agg", ()) or ()),
            "timers": len(rec.repeat_timers_by_name),
            "cache": len(manager.hc_cache),
        })
    elapsed = time.monotonic() - t0
    total = rec.completed_runs - runs_start
    lat = manager.drain_latencies()
    remedy_runs = None
    if args.remedy_frac > 0:
        objs = await client.list(
            "activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health")
        remedy_runs = sum(
            (o.get("status") or {}).get("remedyTotalRuns", 0) for o in objs)
    result = {
        "crs": args.crs,
        "remedy_frac": args.remedy_frac,
        "remedy_total_runs": remedy_runs,
        "repeat_after_sec": args.repeat,
        "duration_s": round(elapsed, 1),
        "cycles": total,
        "cycles_per_sec": round(total / elapsed, 1),
        "expected_cycles_per_sec": round(args.crs / args.repeat, 1),
        "p50_reconcile_ms": round(statistics.median(lat) * 1000, 4) if lat else None,
        "rss_start_mb": round(rss_start, 1),
        "rss_end_mb": round(rss_mb(), 1),
        "store_objects_end": len(server) if server is not None else None,
        "apiserver": args.apiserver,
        "apiserver_requests_per_cycle": (
            round(client.request_count / max(1, rec.completed_runs), 2)
            if apiserver_proc is not None else None
        ),
        "samples": samples,
    }
    await manager.stop()
    if engine is not None:
        await engine.stop()
    if apiserver_proc is not None:
        await client.close()
        apiserver_proc.terminate()
        await apiserver_proc.wait()
    return result


def main() -> int:
    import gc

    gc.set_threshold(50000, 50, 50)
    ap = argparse.ArgumentParser()
    ap.add_argument("--crs", type=int, default=1000)
    ap.add_argument("--repeat", type=int, default=5)
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--duration", type=float, default=120.0)
    ap.add_argument("--ttl", type=float, default=30.0,
                    help="completed-workflow TTL (Argo ttlStrategy equivalent)")
    ap.add_argument("--cron-frac", type=float, default=0.0,
                    help="fraction of CRs scheduled by cron (@every) instead "
                         "of repeatAfterSec")
    ap.add_argument("--remedy-frac", type=float, default=0.0,
                    help="fraction of CRs whose checks fail and carry a "
                         "remedy workflow (steady-state remedy machinery)")
    ap.add_argument("--apiserver", choices=["wire", "memory"], default="wire",
                    help="wire: separate apiserver process over 127.0.0.1 "
                         "HTTP (the bench.py headline regime); memory: "
                         "in-process store")
    args = ap.parse_args()
    print(json.dumps(asyncio.run(main_async(args))))
    return 0


if __name__ == "__main__":
    sys.exit(main())
