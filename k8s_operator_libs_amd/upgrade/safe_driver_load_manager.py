"""Safe driver load handshake manager.

Capability parity with the reference's
``pkg/upgrade/safe_driver_load_manager.go`` and the protocol described in
``docs/automatic-ofed-upgrade.md:43-66``, retargeted at the AMD two-step
driver load: the amdgpu-dkms / ROCm driver pod's init container sets the
``amd.com/<driver>-driver-upgrade.driver-wait-for-safe-load`` annotation on
its node and blocks before (re)loading the kernel driver.  The state machine
treats such a node as upgrade-required, runs the normal cordon/drain pipeline
to clear workloads, and then removes the annotation
(:meth:`SafeDriverLoadManager.unblock_loading`) exactly in the pod-restart and
validation phases — the init container observes the removal, exits, and the
driver loads on a quiesced node.
"""

from __future__ import annotations

import logging
from ..core import meta
from ..core.meta import K8sObject
from . import consts, util
from .node_state_provider import NodeUpgradeStateProvider

logger = logging.getLogger(__name__)


class SafeDriverLoadManager:
    def __init__(self, node_state_provider: NodeUpgradeStateProvider) -> None:
        self._provider = node_state_provider

    def is_waiting_for_safe_driver_load(self, node: K8sObject) -> bool:
        """(safe_driver_load_manager.go:51-53)"""
        key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        return key in (node.get("metadata", {}).get("annotations") or {})

    def unblock_loading(self, node: K8sObject) -> None:
        """Remove the wait annotation, letting the driver load proceed
        (safe_driver_load_manager.go:57-71).  Idempotent."""
        key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        if key not in (node.get("metadata", {}).get("annotations") or {}):
            return
        self._provider.change_node_upgrade_annotation(node, key, consts.NULL_STRING)
        logger.info("unblocked safe driver load on node %s", meta.name(node))
