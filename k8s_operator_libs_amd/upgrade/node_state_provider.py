"""Synchronized node state accessor.

Capability parity with the reference's
``pkg/upgrade/node_upgrade_state_provider.go``: a per-node
:class:`~k8s_operator_libs_amd.upgrade.util.KeyedMutex` serializes all state
label / annotation mutations, each mutation is applied as a JSON merge patch,
and after patching the provider **waits until a fresh read reflects the
change** before returning.  That read-back barrier is what makes the state
machine safe to run over a cached (informer-backed) client: without it a
subsequent reconcile could read a stale state label and double-fire a
transition (rationale: node_upgrade_state_provider.go:92-99).

Performance redesign vs the reference: the reference polls at a fixed 1 s
interval (10 s timeout) — the dominant reconcile-latency term (SURVEY.md
§3.2).  Here the barrier starts at 0.5 ms and backs off exponentially to
100 ms under the same 10 s deadline, so an up-to-date (or uncached) client
confirms in microseconds while a slow cache still converges.
"""

from __future__ import annotations

import logging
import time
from typing import Optional

from ..core import meta
from ..core.client import Client
from ..core.events import EVENT_TYPE_NORMAL, EVENT_TYPE_WARNING, log_event, log_eventf
from ..core.meta import K8sObject
from . import consts, util

logger = logging.getLogger(__name__)

_BARRIER_TIMEOUT_S = 10.0
_BARRIER_INITIAL_S = 0.0005
_BARRIER_MAX_INTERVAL_S = 0.1


class StateChangeTimeoutError(Exception):
    pass


class NodeUpgradeStateProvider:
    """Thread-safe node get / state-label change / annotation change
    (interface parity: node_upgrade_state_provider.go:33-37)."""

    def __init__(self, client: Client, event_recorder: Optional[object] = None) -> None:
        self._client = client
        self._recorder = event_recorder
        self._mutex = util.KeyedMutex()
        # set by the state manager; transition counts land here
        self.metrics = None

    def get_node(self, name: str) -> K8sObject:
        return self._client.get_node(name)

    # -- state label ---------------------------------------------------------

    def change_node_upgrade_state(self, node: K8sObject, new_state: str) -> None:
        """Patch the upgrade-state label and wait for read-back coherency.

        Mutates ``node`` in place on success so the caller's snapshot stays
        current within the same reconcile tick.
        """
        node_name = meta.name(node)
        key = util.get_upgrade_state_label_key()
        with self._mutex.lock(node_name):
            old_state = meta.get_label(node, key)
            if old_state == new_state:
                return
            try:
                self._patch_and_confirm(
                    node_name, {"metadata": {"labels": {key: new_state}}},
                    lambda n: meta.get_label(n, key) == new_state,
                )
            except Exception as exc:
                log_eventf(
                    self._recorder, node, EVENT_TYPE_WARNING, util.get_event_reason(),
                    "Failed to update node state label to '{}': {}", new_state, exc,
                )
                raise
            meta.labels(node)[key] = new_state
            if self.metrics is not None:
                self.metrics.state_transitions.inc(old_state, new_state)
                if new_state == consts.UPGRADE_STATE_FAILED:
                    self.metrics.upgrade_failures.inc()
            logger.info("node %s upgrade state: %r -> %r", node_name, old_state, new_state)
            log_eventf(
                self._recorder, node, EVENT_TYPE_NORMAL, util.get_event_reason(),
                "Successfully updated node state label to '{}'", new_state,
            )

    # -- annotations ---------------------------------------------------------

    def change_node_upgrade_annotation(self, node: K8sObject, key: str, value: str) -> None:
        """Set (or with value ``"null"`` delete) a node annotation, with the
        same read-back barrier (node_upgrade_state_provider.go:138-216)."""
        node_name = meta.name(node)
        delete = value == consts.NULL_STRING
        with self._mutex.lock(node_name):
            patch_value = None if delete else value
            if delete:
                def confirmed(n: K8sObject) -> bool:
                    return key not in (n.get("metadata", {}).get("annotations") or {})
            else:
                def confirmed(n: K8sObject) -> bool:
                    return meta.get_annotation(n, key) == value
            try:
                self._patch_and_confirm(
                    node_name,
                    {"metadata": {"annotations": {key: patch_value}}},
                    confirmed,
                )
            except Exception as exc:
                log_eventf(
                    self._recorder, node, EVENT_TYPE_WARNING, util.get_event_reason(),
                    "Failed to update node annotation '{}': {}", key, exc,
                )
                raise
            if delete:
                meta.annotations(node).pop(key, None)
            else:
                meta.annotations(node)[key] = value
            log_event(
                self._recorder, node, EVENT_TYPE_NORMAL, util.get_event_reason(),
                f"Successfully {'deleted' if delete else 'updated'} node annotation '{key}'",
            )

    # -- internals -----------------------------------------------------------

    def _patch_and_confirm(self, node_name: str, patch: K8sObject, confirmed) -> None:
        resp = self._client.patch("v1", "Node", node_name, patch)
        deadline = time.monotonic() + _BARRIER_TIMEOUT_S

        # Fast path: an informer-backed client exposes an event-driven
        # RV barrier — block until the cache has seen exactly the
        # resourceVersion our patch produced, no polling.
        waiter = getattr(self._client, "wait_for_resource_version", None)
        rv = meta.resource_version(resp) if isinstance(resp, dict) else ""
        if waiter is not None and rv:
            if waiter("v1", "Node", node_name, "", rv, _BARRIER_TIMEOUT_S):
                live = self._client.get_node(node_name)
                if confirmed(live):
                    return
            # cache caught up but the condition doesn't hold (a competing
            # writer overwrote us) or the wait timed out: fall through to
            # the poll loop, which decides between converged and timeout

        interval = _BARRIER_INITIAL_S
        while True:
            live = self._client.get_node(node_name)
            if confirmed(live):
                return
            if time.monotonic() >= deadline:
                raise StateChangeTimeoutError(
                    f"node {node_name}: patched state not visible after "
                    f"{_BARRIER_TIMEOUT_S}s (stale cache?)"
                )
            time.sleep(interval)
            interval = min(interval * 2, _BARRIER_MAX_INTERVAL_S)
