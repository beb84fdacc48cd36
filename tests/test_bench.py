"""Benchmark harness smoke tests (CPU-only path)."""

import json
import subprocess
import sys
import os


def test_rolling_upgrade_benchmark_function():
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    import bench

    result = bench.run_rolling_upgrade_benchmark(
        n_nodes=2, steps=1, warmup=0, max_parallel=1, gpu_validate=False,
    )
    assert result["upgrades_completed"] == 1
    assert result["mean_wall_s"] > 0
    assert result["reconcile_p50_ms"] > 0


def test_bench_cli_prints_contract_json():
    repo = os.path.join(os.path.dirname(__file__), "..")
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "0", "--nodes", "2"],
        capture_output=True, text=True, cwd=repo, timeout=300,
    )
    assert out.returncode == 0, out.stderr
    line = out.stdout.strip().splitlines()[-1]
    data = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in data
    assert data["higher_is_better"] is False
    assert data["scaling"] == "weak"
    assert data["config"]["model"] == "amdgpu-driver-rolling-upgrade"


def test_bench_requestor_mode():
    import bench

    result = bench.run_rolling_upgrade_benchmark(
        n_nodes=2, steps=1, warmup=0, max_parallel=1, gpu_validate=False,
        mode="requestor",
    )
    assert result["upgrades_completed"] == 1


def test_performance_regression_guard():
    """Generous ceiling to catch pathological slowdowns in CI: a full 8-node
    rolling upgrade (live reconcile, no GPU) must finish well under 2s
    (measured ~5-25ms on healthy machines)."""
    import bench

    result = bench.run_rolling_upgrade_benchmark(
        n_nodes=8, steps=3, warmup=1, gpu_validate=False,
    )
    assert result["mean_wall_s"] < 2.0, result
    assert result["reconcile_p50_ms"] < 200.0, result
