"""amd-k8s-operator-libs: AMD/MI355X-native Kubernetes operator utility library.

A from-scratch library of reusable components for writing Kubernetes operators
that manage AMD Instinct GPU / NIC driver lifecycles:

- ``k8s_operator_libs_amd.api.upgrade.v1alpha1``: driver-upgrade policy API
  types embedded in consumer CRDs (capability parity with the reference's
  ``api/upgrade/v1alpha1/upgrade_spec.go``).
- ``k8s_operator_libs_amd.upgrade``: the rolling driver-upgrade state machine
  (upgrade-required -> cordon -> wait-for-jobs -> pod-deletion -> drain ->
  pod-restart -> validate -> uncordon -> done), persisted as
  ``amd.com/<driver>-driver-upgrade-state`` node labels (reference:
  ``pkg/upgrade/``).
- ``k8s_operator_libs_amd.crdutil``: CustomResourceDefinition lifecycle
  helpers (reference: ``pkg/crdutil/crdutil.go``).
- ``k8s_operator_libs_amd.core``: the Kubernetes client substrate - a typed
  ``Client`` interface with an in-memory apiserver (``FakeCluster``, the
  envtest equivalent), an httpx REST client for real clusters, and a FastAPI
  mini-apiserver for wire-level testing.
- ``k8s_operator_libs_amd.validation``: AMD GPU node health validation
  (amd-smi / rocm-smi / native HIP gfx950 health-check kernels) - the AMD
  replacement for NVML-based validator pods.
"""

__version__ = "0.1.0"
