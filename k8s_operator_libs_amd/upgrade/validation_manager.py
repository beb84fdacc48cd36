"""Post-upgrade driver validation manager.

Capability parity with the reference's ``pkg/upgrade/validation_manager.go``:
after the new driver pod restarts, validation pods (selected by
``pod_selector``) must become Ready on the node before it is uncordoned.  Any
pod not Ready starts a 600 s timeout clock (validation_manager.go:31-33)
tracked via a node annotation; on expiry the node moves to upgrade-failed.

On AMD clusters the validation pods typically run amd-smi / rocm-smi health
checks or this library's native gfx950 GPU health check
(:mod:`k8s_operator_libs_amd.validation`) instead of NVML-based validators.
"""

from __future__ import annotations

import logging
import time
from typing import Optional

from ..core import meta
from ..core.client import Client
from ..core.events import EVENT_TYPE_WARNING, log_eventf
from ..core.meta import K8sObject
from . import consts, util
from .node_state_provider import NodeUpgradeStateProvider

logger = logging.getLogger(__name__)

# Hard-coded validation timeout (validation_manager.go:31-33).
VALIDATION_TIMEOUT_SECONDS = 600


class ValidationManager:
    def __init__(
        self,
        client: Client,
        node_state_provider: NodeUpgradeStateProvider,
        pod_selector: str = "",
        event_recorder: Optional[object] = None,
    ) -> None:
        self._client = client
        self._provider = node_state_provider
        self._pod_selector = pod_selector
        self._recorder = event_recorder

    @property
    def pod_selector(self) -> str:
        return self._pod_selector

    def validate(self, node: K8sObject) -> bool:
        """True when all validation pods on the node are Ready
        (validation_manager.go:71-116).  An empty selector short-circuits to
        success (validation disabled)."""
        if not self._pod_selector:
            return True
        node_name = meta.name(node)
        pods = self._client.list_pods(
            label_selector=self._pod_selector,
            field_selector=consts.NODE_NAME_FIELD_SELECTOR_FMT.format(node_name),
        )
        if not pods:
            logger.warning("no validation pods found on node %s (selector %r)",
                           node_name, self._pod_selector)
            return False
        for pod in pods:
            if not self._is_pod_ready(pod):
                try:
                    self._handle_timeout(node, VALIDATION_TIMEOUT_SECONDS)
                except Exception as exc:
                    log_eventf(self._recorder, node, EVENT_TYPE_WARNING,
                               util.get_event_reason(),
                               "Failed to handle timeout for validation state: {}", exc)
                    raise
                return False
        # all Ready: clear the timeout-tracking annotation
        key = util.get_validation_start_time_annotation_key()
        if key in (node.get("metadata", {}).get("annotations") or {}):
            self._provider.change_node_upgrade_annotation(node, key, consts.NULL_STRING)
        return True

    @staticmethod
    def _is_pod_ready(pod: K8sObject) -> bool:
        """(validation_manager.go:119-137): Running phase with every container
        status Ready; a pod with no container statuses is not ready."""
        if pod.get("status", {}).get("phase") != "Running":
            return False
        statuses = pod.get("status", {}).get("containerStatuses") or []
        if not statuses:
            return False
        return all(s.get("ready") for s in statuses)

    def _handle_timeout(self, node: K8sObject, timeout_seconds: int) -> None:
        """(validation_manager.go:139-175)"""
        key = util.get_validation_start_time_annotation_key()
        now = int(time.time())
        annotations = node.get("metadata", {}).get("annotations") or {}
        if key not in annotations:
            self._provider.change_node_upgrade_annotation(node, key, str(now))
            return
        try:
            start_time = int(annotations[key])
        except ValueError:
            logger.warning("node %s: corrupt validation start-time %r; re-stamping",
                           meta.name(node), annotations[key])
            self._provider.change_node_upgrade_annotation(node, key, str(now))
            return
        if now > start_time + timeout_seconds:
            self._provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_FAILED)
            logger.info("validation timeout exceeded on node %s -> upgrade-failed",
                        meta.name(node))
            self._provider.change_node_upgrade_annotation(node, key, consts.NULL_STRING)
