"""Shared state-machine logic: phase processors and scheduling arithmetic.

Capability parity with the reference's ``pkg/upgrade/common_manager.go``.
Holds the client, event recorder and all L3 managers; implements every phase
processor shared between the in-place and requestor modes plus the rolling
window arithmetic (maxParallelUpgrades × maxUnavailable clamping).

State is a node label, so every processor is stateless and idempotent per
reconcile: if a tick dies halfway, the next tick resumes from the labels.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..api.upgrade.v1alpha1 import DrainSpec, PodDeletionSpec, WaitForCompletionSpec
from ..core import meta
from ..core.client import Client
from ..core.meta import K8sObject
from . import consts, util
from .cordon_manager import CordonManager
from .drain_manager import DrainConfiguration, DrainManager
from .node_state_provider import NodeUpgradeStateProvider
from .pod_manager import PodDeletionFilter, PodManager, PodManagerConfig
from .safe_driver_load_manager import SafeDriverLoadManager
from .validation_manager import ValidationManager

logger = logging.getLogger(__name__)

# A driver container restarting more than this many times fails the upgrade
# (common_manager.go:636-648).
POD_RESTART_FAILURE_THRESHOLD = 10


@dataclass
class NodeUpgradeState:
    """Join of node + driver pod + owning DaemonSet (+ NodeMaintenance in
    requestor mode) — common_manager.go:58-63."""

    node: K8sObject
    driver_pod: Optional[K8sObject] = None
    driver_daemonset: Optional[K8sObject] = None
    node_maintenance: Optional[K8sObject] = None

    def is_orphaned_pod(self) -> bool:
        """A driver pod with no owner references (common_manager.go:65-68)."""
        return self.driver_pod is not None and not meta.owner_references(self.driver_pod)


@dataclass
class ClusterUpgradeState:
    """Nodes grouped by their current state label (common_manager.go:73-75)."""

    node_states: Dict[str, List[NodeUpgradeState]] = field(default_factory=dict)

    def nodes_in(self, state: str) -> List[NodeUpgradeState]:
        return self.node_states.get(state, [])

    def add(self, state: str, node_state: NodeUpgradeState) -> None:
        self.node_states.setdefault(state, []).append(node_state)

    def regroup(self, state_label_key: str) -> None:
        """Re-bucket every node by its CURRENT state label.

        Node objects are mutated in place by the state provider, so after a
        phase runs, regrouping makes nodes visible to later phases of the
        same pass.  Used by the live (pipelined) apply mode — the reference
        always processes the fixed snapshot (one transition per tick)."""
        from ..core import meta as _meta

        all_states = [ns for lst in self.node_states.values() for ns in lst]
        self.node_states = {}
        for node_state in all_states:
            self.add(_meta.get_label(node_state.node, state_label_key), node_state)


def is_orphaned_pod(pod: K8sObject) -> bool:
    return not meta.owner_references(pod)


def is_node_unschedulable(node: K8sObject) -> bool:
    return bool(node.get("spec", {}).get("unschedulable", False))


def is_node_in_requestor_mode(node: K8sObject) -> bool:
    """(util.go:134-138)"""
    key = util.get_upgrade_requestor_mode_annotation_key()
    return meta.get_annotation(node, key) == consts.TRUE_STRING


class CommonUpgradeManager:
    """(CommonUpgradeManagerImpl, common_manager.go:84-133)"""

    def __init__(
        self,
        client: Client,
        event_recorder: Optional[object] = None,
        node_state_provider: Optional[NodeUpgradeStateProvider] = None,
        pod_deletion_filter: Optional[PodDeletionFilter] = None,
        validation_pod_selector: str = "",
    ) -> None:
        self.client = client
        self.event_recorder = event_recorder
        self.node_state_provider = node_state_provider or NodeUpgradeStateProvider(
            client, event_recorder
        )
        self.cordon_manager = CordonManager(client, event_recorder)
        self.drain_manager = DrainManager(client, self.node_state_provider, event_recorder)
        self.pod_manager = PodManager(
            client, self.node_state_provider,
            pod_deletion_filter=pod_deletion_filter, event_recorder=event_recorder,
        )
        self.validation_manager = ValidationManager(
            client, self.node_state_provider,
            pod_selector=validation_pod_selector, event_recorder=event_recorder,
        )
        self.safe_driver_load_manager = SafeDriverLoadManager(self.node_state_provider)
        self.pod_deletion_state_enabled = pod_deletion_filter is not None
        self.validation_state_enabled = bool(validation_pod_selector)

    #: per-phase node fan-out width (0/1 = sequential, the default).
    #: Per-node operations in a phase are independent (each provider
    #: mutation takes the per-node KeyedMutex), so against a REMOTE
    #: apiserver — where each transition pays a genuine network round
    #: trip — raising this takes a phase from O(nodes) round trips to
    #: O(1) wall-clock.  It is OFF by default because it measured 2-3x
    #: SLOWER on the benchmark topology (apiserver in the same process:
    #: the "concurrent" requests contend for the GIL and the server's
    #: event loop instead of overlapping wire latency).  Raise it only
    #: for real out-of-process clusters, verified with your own numbers.
    MAX_PARALLEL_NODE_OPS = 1

    def for_each_node(self, node_states, fn) -> None:
        """Run ``fn(node_state)`` for every node in the phase, fanned out
        over a bounded thread pool (sequential for 0/1 nodes).  All nodes
        are processed even if some fail; the first exception is re-raised
        afterwards, matching the idempotent requeue-on-error contract
        (upgrade_state.go:171-281 processes phases best-effort and relies
        on the next reconcile)."""
        node_states = list(node_states)
        width = min(len(node_states), self.MAX_PARALLEL_NODE_OPS)
        if width <= 1:
            for ns in node_states:
                fn(ns)
            return
        pool = getattr(self, "_node_op_pool", None)
        if pool is None:
            # persistent pool: per-call executor creation/teardown costs
            # more than the fan-out saves (threads join on every phase)
            from concurrent.futures import ThreadPoolExecutor

            pool = ThreadPoolExecutor(
                max_workers=self.MAX_PARALLEL_NODE_OPS,
                thread_name_prefix="node-op",
            )
            self._node_op_pool = pool
        futures = [pool.submit(fn, ns) for ns in node_states]
        first_exc = None
        for fut in futures:
            exc = fut.exception()  # blocks until done
            if exc is not None and first_exc is None:
                first_exc = exc
            elif exc is not None:
                logger.error("per-node phase op failed: %s", exc)
        if first_exc is not None:
            raise first_exc

    # -- feature flags (common_manager.go:136-144) ---------------------------

    def is_pod_deletion_enabled(self) -> bool:
        return self.pod_deletion_state_enabled

    def is_validation_enabled(self) -> bool:
        return self.validation_state_enabled

    # -- snapshot helpers ----------------------------------------------------

    def get_driver_daemonsets(self, namespace: str, labels: Dict[str, str]) -> Dict[str, K8sObject]:
        """UID -> DaemonSet map (common_manager.go:168-188)."""
        selector = ",".join(f"{k}={v}" for k, v in sorted(labels.items()))
        daemonsets = self.client.list_daemonsets(namespace=namespace, label_selector=selector)
        return {meta.uid(ds): ds for ds in daemonsets}

    def get_pods_owned_by_ds(self, ds: K8sObject, pods: List[K8sObject]) -> List[K8sObject]:
        """(common_manager.go:190-209)"""
        out = []
        for pod in pods:
            refs = meta.owner_references(pod)
            if not refs:
                continue
            if refs[0].get("uid") == meta.uid(ds):
                out.append(pod)
        return out

    def get_orphaned_pods(self, pods: List[K8sObject]) -> List[K8sObject]:
        """(common_manager.go:211-221)"""
        return [p for p in pods if is_orphaned_pod(p)]

    # -- node predicates -----------------------------------------------------

    def is_upgrade_requested(self, node: K8sObject) -> bool:
        """(common_manager.go:322-325)"""
        key = util.get_upgrade_requested_annotation_key()
        return meta.get_annotation(node, key) == consts.TRUE_STRING

    def skip_node_upgrade(self, node: K8sObject) -> bool:
        """(common_manager.go:666-668)"""
        return meta.get_label(node, util.get_upgrade_skip_node_label_key()) == consts.TRUE_STRING

    @staticmethod
    def is_node_unschedulable(node: K8sObject) -> bool:
        return is_node_unschedulable(node)

    @staticmethod
    def _is_node_condition_ready(node: K8sObject) -> bool:
        """(common_manager.go:656-663)"""
        for cond in node.get("status", {}).get("conditions", []) or []:
            if cond.get("type") == "Ready" and cond.get("status") != "True":
                return False
        return True

    # -- driver-pod sync checks ----------------------------------------------

    def pod_in_sync_with_ds(self, node_state: NodeUpgradeState) -> tuple:
        """Returns (is_pod_synced, is_orphaned) — common_manager.go:299-320.
        Orphaned pods are never "synced" (there is no DS to compare)."""
        if node_state.is_orphaned_pod():
            return False, True
        pod_hash = self.pod_manager.get_pod_controller_revision_hash(node_state.driver_pod)
        ds_hash = self.pod_manager.get_daemonset_controller_revision_hash(
            node_state.driver_daemonset
        )
        return pod_hash == ds_hash, False

    def is_driver_pod_in_sync(self, node_state: NodeUpgradeState) -> bool:
        """Synced AND Running AND all containers Ready
        (common_manager.go:606-634)."""
        synced, orphaned = self.pod_in_sync_with_ds(node_state)
        if orphaned or not synced:
            return False
        pod = node_state.driver_pod
        if pod.get("status", {}).get("phase") != "Running":
            return False
        statuses = pod.get("status", {}).get("containerStatuses") or []
        if not statuses:
            return False
        return all(s.get("ready") for s in statuses)

    @staticmethod
    def is_driver_pod_failing(pod: K8sObject) -> bool:
        """Any not-ready container with >10 restarts
        (common_manager.go:636-648)."""
        status = pod.get("status", {})
        for key in ("initContainerStatuses", "containerStatuses"):
            for s in status.get(key) or []:
                if not s.get("ready") and s.get("restartCount", 0) > POD_RESTART_FAILURE_THRESHOLD:
                    return True
        return False

    # -- phase processors ----------------------------------------------------

    def process_done_or_unknown_nodes(
        self, state: ClusterUpgradeState, state_name: str
    ) -> None:
        """(common_manager.go:229-291)"""

        def one(node_state):
            synced, orphaned = self.pod_in_sync_with_ds(node_state)
            waiting_safe_load = self.safe_driver_load_manager.is_waiting_for_safe_driver_load(
                node_state.node
            )
            upgrade_requested = self.is_upgrade_requested(node_state.node)
            if (not synced and not orphaned) or waiting_safe_load or upgrade_requested:
                if is_node_unschedulable(node_state.node):
                    # Remember the node started cordoned so it is never
                    # uncordoned by us (common_manager.go:250-264).
                    self.node_state_provider.change_node_upgrade_annotation(
                        node_state.node,
                        util.get_upgrade_initial_state_annotation_key(),
                        consts.TRUE_STRING,
                    )
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_UPGRADE_REQUIRED
                )
                return
            if state_name == consts.UPGRADE_STATE_UNKNOWN:
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_DONE
                )

        self.for_each_node(state.nodes_in(state_name), one)

    def process_cordon_required_nodes(self, state: ClusterUpgradeState) -> None:
        """(common_manager.go:361-380)"""

        def one(node_state):
            self.cordon_manager.cordon(node_state.node)
            self.node_state_provider.change_node_upgrade_state(
                node_state.node, consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED
            )

        self.for_each_node(
            state.nodes_in(consts.UPGRADE_STATE_CORDON_REQUIRED), one
        )

    def process_wait_for_jobs_required_nodes(
        self,
        state: ClusterUpgradeState,
        wait_for_completion_spec: Optional[WaitForCompletionSpec],
    ) -> None:
        """(common_manager.go:384-419)"""
        node_states = state.nodes_in(consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED)
        no_selector = (
            wait_for_completion_spec is None or not wait_for_completion_spec.pod_selector
        )
        if no_selector:
            # nothing to wait for: fall through (to pod-deletion if enabled,
            # else straight to drain)
            next_state = (
                consts.UPGRADE_STATE_POD_DELETION_REQUIRED
                if self.is_pod_deletion_enabled()
                else consts.UPGRADE_STATE_DRAIN_REQUIRED
            )
            for node_state in node_states:
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, next_state
                )
            return
        if not node_states:
            return
        self.pod_manager.schedule_check_on_pod_completion(
            PodManagerConfig(
                nodes=[ns.node for ns in node_states],
                wait_for_completion_spec=wait_for_completion_spec,
            )
        )

    def process_pod_deletion_required_nodes(
        self,
        state: ClusterUpgradeState,
        pod_deletion_spec: Optional[PodDeletionSpec],
        drain_enabled: bool,
    ) -> None:
        """(common_manager.go:424-453)"""
        node_states = state.nodes_in(consts.UPGRADE_STATE_POD_DELETION_REQUIRED)
        if not self.is_pod_deletion_enabled():
            for node_state in node_states:
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_DRAIN_REQUIRED
                )
            return
        if not node_states:
            return
        self.pod_manager.schedule_pod_eviction(
            PodManagerConfig(
                nodes=[ns.node for ns in node_states],
                deletion_spec=pod_deletion_spec or PodDeletionSpec(),
                drain_enabled=drain_enabled,
            )
        )

    def process_drain_nodes(
        self, state: ClusterUpgradeState, drain_spec: Optional[DrainSpec]
    ) -> None:
        """(common_manager.go:329-357)"""
        node_states = state.nodes_in(consts.UPGRADE_STATE_DRAIN_REQUIRED)
        if drain_spec is None or not drain_spec.enable:
            # drain disabled: straight to pod-restart
            for node_state in node_states:
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_POD_RESTART_REQUIRED
                )
            return
        self.drain_manager.schedule_nodes_drain(
            DrainConfiguration(spec=drain_spec, nodes=[ns.node for ns in node_states])
        )

    def process_pod_restart_nodes(self, state: ClusterUpgradeState) -> None:
        """(common_manager.go:457-524)"""
        pods_to_restart = []  # .append is atomic under the GIL

        def one(node_state):
            synced, orphaned = self.pod_in_sync_with_ds(node_state)
            if not synced or orphaned:
                # restart unless already terminating
                if "deletionTimestamp" not in node_state.driver_pod.get("metadata", {}):
                    pods_to_restart.append(node_state.driver_pod)
                return
            # template in sync: unblock safe driver load, then wait for Ready
            self.safe_driver_load_manager.unblock_loading(node_state.node)
            if self.is_driver_pod_in_sync(node_state):
                if not self.is_validation_enabled():
                    self.update_node_to_uncordon_or_done_state(node_state)
                    return
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_VALIDATION_REQUIRED
                )
            elif self.is_driver_pod_failing(node_state.driver_pod):
                self.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_FAILED
                )

        self.for_each_node(
            state.nodes_in(consts.UPGRADE_STATE_POD_RESTART_REQUIRED), one
        )
        self.pod_manager.schedule_pods_restart(pods_to_restart)

    def process_upgrade_failed_nodes(self, state: ClusterUpgradeState) -> None:
        """Auto-recovery once the driver pod is back in sync
        (common_manager.go:528-570)."""
        for node_state in state.nodes_in(consts.UPGRADE_STATE_FAILED):
            if not self.is_driver_pod_in_sync(node_state):
                continue
            new_state = consts.UPGRADE_STATE_UNCORDON_REQUIRED
            key = util.get_upgrade_initial_state_annotation_key()
            if key in (node_state.node.get("metadata", {}).get("annotations") or {}):
                new_state = consts.UPGRADE_STATE_DONE
            self.node_state_provider.change_node_upgrade_state(node_state.node, new_state)
            if new_state == consts.UPGRADE_STATE_DONE:
                self.node_state_provider.change_node_upgrade_annotation(
                    node_state.node, key, consts.NULL_STRING
                )

    def process_validation_required_nodes(self, state: ClusterUpgradeState) -> None:
        """(common_manager.go:573-604)"""

        def one(node_state):
            # The driver may have restarted after reaching this state and be
            # blocked on safe load again — always unblock here
            # (common_manager.go:581-586).
            self.safe_driver_load_manager.unblock_loading(node_state.node)
            if not self.validation_manager.validate(node_state.node):
                return
            self.update_node_to_uncordon_or_done_state(node_state)

        self.for_each_node(
            state.nodes_in(consts.UPGRADE_STATE_VALIDATION_REQUIRED), one
        )

    def update_node_to_uncordon_or_done_state(self, node_state: NodeUpgradeState) -> None:
        """(common_manager.go:673-708): initially-unschedulable nodes skip
        uncordon and go straight to done (unless handled by requestor mode,
        whose uncordon processor owns the transition)."""
        node = node_state.node
        new_state = consts.UPGRADE_STATE_UNCORDON_REQUIRED
        key = util.get_upgrade_initial_state_annotation_key()
        requestor_mode = is_node_in_requestor_mode(node)
        has_initial = key in (node.get("metadata", {}).get("annotations") or {})
        if has_initial and not requestor_mode:
            new_state = consts.UPGRADE_STATE_DONE
        self.node_state_provider.change_node_upgrade_state(node, new_state)
        if has_initial and (new_state == consts.UPGRADE_STATE_DONE or requestor_mode):
            self.node_state_provider.change_node_upgrade_annotation(
                node, key, consts.NULL_STRING
            )

    # -- scheduling arithmetic (common_manager.go:146-165, 715-788) ----------

    _MANAGED_STATES = (
        consts.UPGRADE_STATE_UNKNOWN,
        consts.UPGRADE_STATE_DONE,
        consts.UPGRADE_STATE_UPGRADE_REQUIRED,
        consts.UPGRADE_STATE_CORDON_REQUIRED,
        consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED,
        consts.UPGRADE_STATE_POD_DELETION_REQUIRED,
        consts.UPGRADE_STATE_FAILED,
        consts.UPGRADE_STATE_DRAIN_REQUIRED,
        consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
        consts.UPGRADE_STATE_UNCORDON_REQUIRED,
        consts.UPGRADE_STATE_VALIDATION_REQUIRED,
    )

    def get_total_managed_nodes(self, state: ClusterUpgradeState) -> int:
        """(common_manager.go:715-730) — note: node-maintenance-required and
        post-maintenance-required are requestor-mode states counted separately,
        matching the reference."""
        return sum(len(state.nodes_in(s)) for s in self._MANAGED_STATES)

    def get_upgrades_in_progress(self, state: ClusterUpgradeState) -> int:
        """(common_manager.go:733-739)"""
        return self.get_total_managed_nodes(state) - (
            len(state.nodes_in(consts.UPGRADE_STATE_UNKNOWN))
            + len(state.nodes_in(consts.UPGRADE_STATE_DONE))
            + len(state.nodes_in(consts.UPGRADE_STATE_UPGRADE_REQUIRED))
        )

    def get_upgrades_done(self, state: ClusterUpgradeState) -> int:
        return len(state.nodes_in(consts.UPGRADE_STATE_DONE))

    def get_upgrades_failed(self, state: ClusterUpgradeState) -> int:
        return len(state.nodes_in(consts.UPGRADE_STATE_FAILED))

    def get_upgrades_pending(self, state: ClusterUpgradeState) -> int:
        return len(state.nodes_in(consts.UPGRADE_STATE_UPGRADE_REQUIRED))

    def get_current_unavailable_nodes(self, state: ClusterUpgradeState) -> int:
        """Cordoned or NotReady nodes (common_manager.go:146-165)."""
        unavailable = 0
        for node_states in state.node_states.values():
            for node_state in node_states:
                if is_node_unschedulable(node_state.node):
                    unavailable += 1
                elif not self._is_node_condition_ready(node_state.node):
                    unavailable += 1
        return unavailable

    def get_upgrades_available(
        self, state: ClusterUpgradeState, max_parallel_upgrades: int, max_unavailable: int
    ) -> int:
        """The rolling-window slot computation (common_manager.go:748-776):
        maxParallel bounds concurrent upgrades (0 = unlimited), then the slots
        are clamped so cordoned/NotReady/about-to-cordon nodes never exceed
        maxUnavailable."""
        in_progress = self.get_upgrades_in_progress(state)
        total = self.get_total_managed_nodes(state)
        if max_parallel_upgrades == 0:
            available = len(state.nodes_in(consts.UPGRADE_STATE_UPGRADE_REQUIRED))
        else:
            available = max_parallel_upgrades - in_progress
        current_unavailable = self.get_current_unavailable_nodes(state) + len(
            state.nodes_in(consts.UPGRADE_STATE_CORDON_REQUIRED)
        )
        if available > max_unavailable:
            available = max_unavailable
        if current_unavailable >= max_unavailable:
            available = 0
        elif max_unavailable < total and current_unavailable + available > max_unavailable:
            available = max_unavailable - current_unavailable
        return available

    # -- worker draining (test/bench determinism) -----------------------------

    def wait_idle(self, timeout: float = 60.0) -> None:
        """Join all async drain/eviction workers spawned by the last tick."""
        self.pod_manager.wait_idle(timeout)
        self.drain_manager.wait_idle(timeout)
