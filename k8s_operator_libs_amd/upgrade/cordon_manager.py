"""Cordon / uncordon manager.

Capability parity with the reference's ``pkg/upgrade/cordon_manager.go:33-48``
(a thin wrapper over kubectl's RunCordonOrUncordon): sets or clears
``spec.unschedulable`` via a merge patch and records an Event.  Patching only
the one field (instead of kubectl's full-object update) is conflict-free
against concurrent label changes by the state provider.
"""

from __future__ import annotations

import logging
from typing import Optional

from ..core import meta
from ..core.client import Client
from ..core.events import EVENT_TYPE_NORMAL, log_event
from ..core.meta import K8sObject
from . import util

logger = logging.getLogger(__name__)


class CordonManager:
    def __init__(self, client: Client, event_recorder: Optional[object] = None) -> None:
        self._client = client
        self._recorder = event_recorder

    def cordon(self, node: K8sObject) -> None:
        self._set_unschedulable(node, True)

    def uncordon(self, node: K8sObject) -> None:
        self._set_unschedulable(node, False)

    def _set_unschedulable(self, node: K8sObject, desired: bool) -> None:
        name = meta.name(node)
        current = bool(node.get("spec", {}).get("unschedulable", False))
        if current == desired:
            return
        # "null" clears the field entirely rather than storing false.
        self._client.patch(
            "v1", "Node", name, {"spec": {"unschedulable": True if desired else None}}
        )
        node.setdefault("spec", {})["unschedulable"] = desired
        if not desired:
            node["spec"].pop("unschedulable", None)
        verb = "cordoned" if desired else "uncordoned"
        logger.info("node %s %s", name, verb)
        log_event(self._recorder, node, EVENT_TYPE_NORMAL, util.get_event_reason(),
                  f"Node {verb} for driver upgrade")
