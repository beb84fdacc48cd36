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


def test_bench_substrates_all_converge():
    """All three substrates complete an upgrade and
    report the substrate they measured."""
    import bench

    for substrate in ("inproc", "http", "cached"):
        result = bench.run_rolling_upgrade_benchmark(
            n_nodes=2, steps=1, warmup=0, max_parallel=1, gpu_validate=False,
            substrate=substrate,
        )
        assert result["upgrades_completed"] == 1, substrate
        assert result["substrate"] == substrate


def test_bench_anic_mode():
    """BASELINE config #4: NIC/xGMI driver path with wait-for-jobs hooks."""
    import bench
    from k8s_operator_libs_amd.upgrade import util

    result = bench.run_rolling_upgrade_benchmark(
        n_nodes=2, steps=1, warmup=0, max_parallel=2, gpu_validate=False,
        mode="anic", substrate="inproc",
    )
    assert result["upgrades_completed"] == 1
    # driver-name global restored after the run
    assert util.get_driver_name() == "amdgpu"


def test_bench_multirank_gloo_world4():
    """the distributed bench path executes under
    WORLD_SIZE=4 with the gloo backend — one simulated cluster per rank,
    MAX-reduce across ranks, exactly one contract JSON line from rank 0 —
    so the first 8-GPU lease runs the already-proven path."""
    repo = os.path.join(os.path.dirname(__file__), "..")
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node", "4",
         "--master-addr", "127.0.0.1", "--master-port", "29517",
         "bench.py", "--gpus", "4", "--steps", "1", "--warmup", "0",
         "--nodes", "2", "--substrate", "inproc",
         "--no-secondary-substrates"],
        capture_output=True, text=True, cwd=repo, timeout=600, env=env,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    json_lines = [l for l in out.stdout.strip().splitlines()
                  if l.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line: {json_lines}"
    data = json.loads(json_lines[0])
    assert data["n_gpus"] == 4
    assert data["config"]["parallelism"].startswith("dp4")
    assert data["value"] > 0
    # MAX-reduce: the reported wall is >= any single rank could have reported
    assert data["ms_per_step"] > 0
