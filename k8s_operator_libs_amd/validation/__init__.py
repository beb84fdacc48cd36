"""AMD GPU node health validation.

The AMD-native replacement for the NVML-based validator pods the reference
orchestrates: after a driver bump, a validation pod on each MI355X node runs
:func:`gpu_health_check` and reports Ready only when the GPU stack is sane.
The upgrade state machine's ValidationManager gates uncordon on that
readiness (BASELINE.json north_star).

Layers:
- :func:`smi_probe` — amd-smi / rocm-smi process-level probe (driver loaded,
  devices enumerated);
- the native ``_gpu_validator`` HIP extension — device probe, MFMA matrix
  core smoke tests (f32-exact and bf16), HBM streaming bandwidth, LDS
  integrity.  **Fails loudly if the extension is missing on a GPU machine**:
  a silent fallback would let a broken driver pass validation.
"""

from .gpu_health import (  # noqa: F401
    GpuHealthError,
    gpu_health_check,
    load_native_validator,
    smi_probe,
)
