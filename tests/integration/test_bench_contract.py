"""bench.py contract tests — including the multi-process distributed path the
driver uses (torch.distributed.run, gloo on CPU, world_size 2)."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _parse_last_json(stdout: str) -> dict:
    lines = [l for l in stdout.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output: {stdout[-2000:]}"
    return json.loads(lines[-1])


def test_bench_single_process_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--crs", "60", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    data = _parse_last_json(out.stdout)
    assert data["metric"] == "sustained_healthcheck_cycles_per_sec"
    assert data["value"] > 0
    assert data["n_gpus"] == 1
    assert data["steps"] == 2 and data["warmup"] == 1
    assert data["higher_is_better"] is True
    assert data["scaling"] == "weak"
    assert data["data"] == "synthetic"
    assert data["ms_per_step"] > 0
    cfg = data["config"]
    assert cfg["crs_per_rank"] == 60
    assert cfg["p50_reconcile_latency_ms"] > 0


def test_bench_two_ranks_over_gloo():
    """The driver launches N>1 via torch.distributed.run; verify the gloo/CPU
    path end-to-end with world_size 2 and whole-job aggregation."""
    torch = pytest.importorskip("torch")
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29771",
         os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--crs", "40", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env,
    )
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    data = _parse_last_json(out.stdout)
    assert data["n_gpus"] == 2
    # whole-job aggregate: 2 ranks × 40 CRs
    assert data["config"]["max_concurrent_crs"] == 80
    assert data["value"] > 0
