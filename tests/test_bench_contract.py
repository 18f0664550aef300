"""bench.py driver-contract tests: single-process JSON line, and the exact
multi-process launch shape the driver uses (torch.distributed.run with
--master-addr 127.0.0.1, gloo rendezvous) at world_size 2 on CPU."""

import json
import os
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_FIELDS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def last_json_line(output: str) -> dict:
    lines = [l for l in output.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output:\n{output}"
    return json.loads(lines[-1])


def test_bench_single_process_contract():
    result = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1", "--objects", "8"],
        cwd=REPO_ROOT,
        capture_output=True,
        text=True,
        timeout=300,
    )
    assert result.returncode == 0, result.stderr[-2000:]
    payload = last_json_line(result.stdout)
    assert REQUIRED_FIELDS <= set(payload)
    assert payload["n_gpus"] == 1
    assert payload["value"] > 0
    assert payload["higher_is_better"] is True
    assert payload["scaling"] == "weak"
    assert payload["data"] == "synthetic"


def test_bench_world_size_2_gloo():
    """The driver launches bench.py via torch.distributed.run with one rank
    per GPU; on CPU the same path must work over gloo at world_size 2."""
    pytest.importorskip("torch")
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    result = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29377",
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--objects", "8",
        ],
        cwd=REPO_ROOT,
        capture_output=True,
        text=True,
        timeout=600,
        env=env,
    )
    assert result.returncode == 0, (result.stderr[-3000:] or result.stdout[-3000:])
    payload = last_json_line(result.stdout)
    assert payload["n_gpus"] == 2
    # whole-job aggregate: 8 objects × 2 steps × 2 ranks / elapsed
    assert payload["value"] > 0
    assert "x2" in payload["config"]["parallelism"]


def test_bench_http_boundary_mode():
    """--api http runs the same workload through the HTTP apiserver +
    REST client (smaller config: each op is a real network round trip)."""
    result = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--objects", "4", "--api", "http"],
        cwd=REPO_ROOT, capture_output=True, text=True, timeout=300,
    )
    assert result.returncode == 0, result.stderr[-2000:]
    payload = last_json_line(result.stdout)
    assert payload["config"]["api"] == "http"
    assert payload["value"] > 0
