"""Upgrade state names and AMD node label/annotation key formats.

Capability parity with the reference's ``pkg/upgrade/consts.go:19-93`` — the
same 13-state machine, with every key moved from the ``nvidia.com/`` domain to
``amd.com/``.  All ``*_FMT`` strings take the driver name (``amdgpu``,
``rocm``, ``anic``, ...) via ``str.format``-style ``{}`` substitution; use the
getters in :mod:`k8s_operator_libs_amd.upgrade.util` rather than formatting
these directly.
"""

# ---------------------------------------------------------------------------
# Node label / annotation key formats (reference consts.go:20-47, amd.com/)
# ---------------------------------------------------------------------------

# Node label holding the upgrade state of the node's driver.
UPGRADE_STATE_LABEL_KEY_FMT = "amd.com/{}-driver-upgrade-state"
# Node label (boolean) telling the state machine to skip this node entirely.
UPGRADE_SKIP_NODE_LABEL_KEY_FMT = "amd.com/{}-driver-upgrade.skip"
# Pod label selector key marking pods the drain should skip.
UPGRADE_SKIP_DRAIN_POD_SELECTOR_FMT = "amd.com/{}-driver-upgrade-drain.skip"
# Node annotation set by the driver pod's init container while it blocks
# waiting for a safe (workload-free) driver load.
UPGRADE_WAIT_FOR_SAFE_DRIVER_LOAD_ANNOTATION_KEY_FMT = (
    "amd.com/{}-driver-upgrade.driver-wait-for-safe-load"
)
# Node annotation remembering that the node was already unschedulable when the
# upgrade began (such nodes are never uncordoned by the state machine).
UPGRADE_INITIAL_STATE_ANNOTATION_KEY_FMT = (
    "amd.com/{}-driver-upgrade.node-initial-state.unschedulable"
)
# Node annotation stamping when the wait-for-pod-completion phase started.
UPGRADE_WAIT_FOR_POD_COMPLETION_START_TIME_ANNOTATION_KEY_FMT = (
    "amd.com/{}-driver-upgrade-wait-for-pod-completion-start-time"
)
# Node annotation stamping when the validation-required phase started.
UPGRADE_VALIDATION_START_TIME_ANNOTATION_KEY_FMT = (
    "amd.com/{}-driver-upgrade-validation-start-time"
)
# Node annotation requesting an upgrade explicitly (used for orphaned driver
# pods that have no owning DaemonSet to compare revisions against).
UPGRADE_REQUESTED_ANNOTATION_KEY_FMT = "amd.com/{}-driver-upgrade-requested"
# Node annotation marking that the node is being upgraded in requestor
# (maintenance-operator) mode rather than in-place mode.
UPGRADE_REQUESTOR_MODE_ANNOTATION_KEY_FMT = (
    "amd.com/{}-driver-upgrade-requestor-mode"
)

# ---------------------------------------------------------------------------
# Upgrade states (reference consts.go:48-83) — persisted as the value of the
# UPGRADE_STATE label on each node, which makes the whole machine stateless
# and idempotent per reconcile.
# ---------------------------------------------------------------------------

# The upgrade flow is disabled or the node hasn't been processed yet.
UPGRADE_STATE_UNKNOWN = ""
# Driver pod on the node is out of date; no actions performed yet.
UPGRADE_STATE_UPGRADE_REQUIRED = "upgrade-required"
# Node must be made unschedulable in preparation for the driver upgrade.
UPGRADE_STATE_CORDON_REQUIRED = "cordon-required"
# Wait (up to a timeout) for workload pods matching the policy selector to
# finish before deleting them.
UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED = "wait-for-jobs-required"
# Deletion of selected workload pods is required before the upgrade proceeds.
UPGRADE_STATE_POD_DELETION_REQUIRED = "pod-deletion-required"
# Full node drain scheduled; moves to pod-restart-required or upgrade-failed.
UPGRADE_STATE_DRAIN_REQUIRED = "drain-required"
# Node handed to an external maintenance operator (requestor mode only).
UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED = "node-maintenance-required"
# External maintenance finished; requestor must do post-maintenance work.
UPGRADE_STATE_POST_MAINTENANCE_REQUIRED = "post-maintenance-required"
# Driver pod on the node scheduled for restart (or safe-load unblock needed).
UPGRADE_STATE_POD_RESTART_REQUIRED = "pod-restart-required"
# New driver must be validated before uncordoning.
UPGRADE_STATE_VALIDATION_REQUIRED = "validation-required"
# Driver pod is up to date and Ready; node can be made schedulable again.
UPGRADE_STATE_UNCORDON_REQUIRED = "uncordon-required"
# Driver pod up to date and running, node schedulable.
UPGRADE_STATE_DONE = "upgrade-done"
# Any failure during the upgrade lands here (recoverable, see common manager).
UPGRADE_STATE_FAILED = "upgrade-failed"

#: Every known state, in rough pipeline order.
ALL_STATES = (
    UPGRADE_STATE_UNKNOWN,
    UPGRADE_STATE_UPGRADE_REQUIRED,
    UPGRADE_STATE_CORDON_REQUIRED,
    UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED,
    UPGRADE_STATE_POD_DELETION_REQUIRED,
    UPGRADE_STATE_DRAIN_REQUIRED,
    UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED,
    UPGRADE_STATE_POST_MAINTENANCE_REQUIRED,
    UPGRADE_STATE_POD_RESTART_REQUIRED,
    UPGRADE_STATE_VALIDATION_REQUIRED,
    UPGRADE_STATE_UNCORDON_REQUIRED,
    UPGRADE_STATE_DONE,
    UPGRADE_STATE_FAILED,
)

# Field selector template for listing pods by node (consts.go:87-88).
NODE_NAME_FIELD_SELECTOR_FMT = "spec.nodeName={}"
# Annotation value that requests deletion of the annotation key when patched.
NULL_STRING = "null"
TRUE_STRING = "true"
