"""GPU health-check implementation (see package docstring)."""

from __future__ import annotations

import importlib
import json
import os
import shutil
import subprocess
from typing import Any, Dict

# Acceptance thresholds for an MI355X node.
MFMA_F32_MAX_ERR = 1e-6       # exact fmaf-chain numerics: effectively zero
MFMA_BF16_MAX_ERR = 5e-2      # bf16 inputs, f32 accumulate, K=32
MFMA_FP8_MAX_ERR = 5e-2       # OCP e4m3 inputs, f32 accumulate
HBM_MIN_GBPS = 1000.0         # far below the ~6000 GB/s measured copy BW,
                              # catches catastrophically degraded memory
MFMA_MIN_TFLOPS = 1000.0      # deep mode: half the bf16 issue-rate floor,
                              # catches down-clocked / power-capped parts


class GpuHealthError(RuntimeError):
    pass


def _gpu_present() -> bool:
    if shutil.which("rocm-smi") or shutil.which("amd-smi"):
        return os.path.exists("/dev/kfd")
    return os.path.exists("/dev/kfd")


def load_native_validator():
    """Import the native extension, building it on the fly if necessary.

    On a machine with an AMD GPU a missing/unbuildable extension raises
    GpuHealthError — validation must never silently fall back.
    On CPU-only machines returns None.
    """
    try:
        return importlib.import_module(
            "k8s_operator_libs_amd.native._gpu_validator"
        )
    except ImportError as exc:
        try:
            from ..native import build as native_build

            native_build.build()
            return importlib.import_module(
                "k8s_operator_libs_amd.native._gpu_validator"
            )
        except Exception as build_exc:
            if _gpu_present():
                raise GpuHealthError(
                    f"native GPU validator unavailable on a GPU machine: "
                    f"import error={exc}; build error={build_exc}"
                ) from build_exc
            return None


def smi_probe() -> Dict[str, Any]:
    """amd-smi / rocm-smi process probe: is the driver loaded and are
    devices enumerated?  Returns {'tool': ..., 'ok': bool, 'raw': ...}."""
    for tool, args in (("amd-smi", ["list", "--json"]), ("rocm-smi", ["--json"])):
        path = shutil.which(tool)
        if not path:
            continue
        try:
            out = subprocess.run(
                [path, *args], capture_output=True, text=True, timeout=30
            )
            ok = out.returncode == 0
            raw: Any = out.stdout.strip()
            try:
                raw = json.loads(raw)
            except (ValueError, TypeError):
                pass
            return {"tool": tool, "ok": ok, "raw": raw}
        except (subprocess.TimeoutExpired, OSError) as exc:
            return {"tool": tool, "ok": False, "raw": str(exc)}
    return {"tool": None, "ok": False, "raw": "no amd-smi/rocm-smi found"}


def gpu_health_check(
    device: int = 0,
    *,
    bw_buf_mib: float = 512.0,
    bw_iters: int = 5,
    require_gpu: bool = False,
    deep: bool = False,
) -> Dict[str, Any]:
    """Full node GPU health check.  Returns a report dict with a top-level
    ``healthy`` verdict; raises GpuHealthError when ``require_gpu`` and no
    usable GPU/native validator is present."""
    report: Dict[str, Any] = {"healthy": False, "checks": {}}
    report["checks"]["smi"] = smi_probe()

    native = load_native_validator()
    if native is None:
        if require_gpu:
            raise GpuHealthError("no native GPU validator available")
        report["reason"] = "no GPU present; native checks skipped"
        return report

    probe = native.device_probe(device)
    report["checks"]["device"] = probe
    arch_ok = "gfx950" in probe.get("gcn_arch", "")
    report["checks"]["arch_ok"] = arch_ok

    mfma_f32_err = native.mfma_f32_check(device)
    mfma_bf16_err = native.mfma_bf16_check(device)
    report["checks"]["mfma_f32_max_err"] = mfma_f32_err
    report["checks"]["mfma_bf16_max_err"] = mfma_bf16_err

    bw = native.hbm_bandwidth_gbps(device, bw_buf_mib, bw_iters)
    report["checks"]["hbm_bandwidth_gbps"] = bw

    lds_ok = native.lds_roundtrip_check(device)
    report["checks"]["lds_ok"] = lds_ok

    deep_ok = True
    if deep:
        # deep mode: fp8 matrix path + sustained matrix-core burn-in
        fp8_err = native.mfma_fp8_check(device)
        tflops = native.mfma_throughput_tflops(device, 100000)
        report["checks"]["mfma_fp8_max_err"] = fp8_err
        report["checks"]["mfma_throughput_tflops"] = tflops
        deep_ok = fp8_err <= MFMA_FP8_MAX_ERR and tflops >= MFMA_MIN_TFLOPS

    # multi-GPU nodes: verify every xGMI peer link is up
    xgmi_ok = True
    if probe.get("device_count", 1) > 1:
        links = native.xgmi_p2p_probe(device, 64.0, 3)
        report["checks"]["xgmi_links"] = [dict(l) for l in links]
        xgmi_ok = all(l.get("accessible") for l in links)
    report["checks"]["xgmi_ok"] = xgmi_ok

    report["healthy"] = bool(
        arch_ok
        and mfma_f32_err <= MFMA_F32_MAX_ERR
        and mfma_bf16_err <= MFMA_BF16_MAX_ERR
        and bw >= HBM_MIN_GBPS
        and lds_ok
        and xgmi_ok
        and deep_ok
    )
    return report


def main() -> int:
    """CLI entry point for use inside a validation pod.  Pass ``--deep`` for
    the fp8 + burn-in checks."""
    import sys

    report = gpu_health_check(require_gpu=True, deep="--deep" in sys.argv)
    print(json.dumps(report, indent=2, default=str))  # noqa: T201 (CLI/build output)
    return 0 if report["healthy"] else 1


if __name__ == "__main__":
    raise SystemExit(main())
