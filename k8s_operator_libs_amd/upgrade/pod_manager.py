"""Pod manager: eviction, driver-pod restart, job-completion wait.

Capability parity with the reference's ``pkg/upgrade/pod_manager.go``
(interface at pod_manager.go:53-60).  Three responsibilities:

1. **Eviction** (:meth:`PodManager.schedule_pod_eviction`,
   pod_manager.go:122-232): per node, a worker thread lists all pods on the
   node, counts those matching the operator-supplied ``pod_deletion_filter``
   (for AMD operators typically pods consuming ``amd.com/gpu`` resources),
   runs them through the drain filter chain, and deletes them.  If some
   matching pods cannot be deleted, the node moves to drain-required (when
   drain is enabled) or upgrade-failed (pod_manager.go:393-403).
2. **Driver pod restart** (:meth:`schedule_pods_restart`,
   pod_manager.go:233-251): plain delete of the out-of-date driver DaemonSet
   pods; the DaemonSet controller recreates them with the new template.
3. **Job-completion wait** (:meth:`schedule_check_on_pod_completion`,
   pod_manager.go:256-368): per node, checks pods matching the
   WaitForCompletionSpec selector; when none are running/pending the node
   moves to pod-deletion-required; otherwise a start-time annotation is
   stamped and on expiry the node is forced onwards.

Worker threads are the Python analogue of the reference's goroutines; the
:class:`~k8s_operator_libs_amd.upgrade.util.StringSet` in-progress guard and
the provider's per-node locks make them safe.  ``wait_idle()`` joins all
outstanding workers (tests and the benchmark use it for determinism).
"""

from __future__ import annotations

import logging
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, List, Optional

from ..api.upgrade.v1alpha1 import PodDeletionSpec, WaitForCompletionSpec
from ..core import meta
from ..core.client import Client
from ..core.events import EVENT_TYPE_NORMAL, EVENT_TYPE_WARNING, log_event, log_eventf
from ..core.meta import K8sObject
from . import consts, util
from .drain import delete_or_evict_pods, get_pods_for_deletion
from .node_state_provider import NodeUpgradeStateProvider

logger = logging.getLogger(__name__)

# Pod label set by the DaemonSet controller with the ControllerRevision hash.
class StaleClusterViewError(RuntimeError):
    """The client's view is missing objects that must exist (e.g. a managed
    DaemonSet with no ControllerRevisions): with an informer-backed client
    this is a transient staleness window, not a cluster defect.  Callers
    treat it like any reconcile error — requeue (upgrade_state.go:128-131
    pattern)."""


POD_CONTROLLER_REVISION_HASH_LABEL = "controller-revision-hash"

PodDeletionFilter = Callable[[K8sObject], bool]


@dataclass
class PodManagerConfig:
    """(pod_manager.go:63-68)"""

    nodes: List[K8sObject] = field(default_factory=list)
    deletion_spec: Optional[PodDeletionSpec] = None
    wait_for_completion_spec: Optional[WaitForCompletionSpec] = None
    drain_enabled: bool = False


class PodManager:
    def __init__(
        self,
        client: Client,
        node_state_provider: NodeUpgradeStateProvider,
        pod_deletion_filter: Optional[PodDeletionFilter] = None,
        event_recorder: Optional[object] = None,
    ) -> None:
        self._client = client
        self._provider = node_state_provider
        self._filter = pod_deletion_filter
        self._recorder = event_recorder
        self._nodes_in_progress = util.StringSet()
        self._workers: List[threading.Thread] = []
        self._workers_lock = threading.Lock()
        # bounded eviction-worker concurrency (see DrainManager rationale)
        self._worker_slots = threading.Semaphore(self.MAX_CONCURRENT_NODE_WORKERS)
        # (namespace, name, resourceVersion) -> latest revision hash; a DS's
        # hash can only change when the DS object itself changes, so this
        # collapses the per-node ControllerRevision LISTs of a reconcile pass
        # into one
        self._ds_hash_cache: dict = {}

    # -- revision-hash helpers (pod_manager.go:84-118) -----------------------

    def get_pod_controller_revision_hash(self, pod: K8sObject) -> str:
        """Hash label stamped on DaemonSet pods (pod_manager.go:84-89)."""
        hash_ = meta.get_label(pod, POD_CONTROLLER_REVISION_HASH_LABEL)
        if not hash_:
            raise ValueError(
                f"pod {meta.name(pod)} has no {POD_CONTROLLER_REVISION_HASH_LABEL} label"
            )
        return hash_

    def get_daemonset_controller_revision_hash(self, daemonset: K8sObject) -> str:
        """Latest ControllerRevision hash for the DaemonSet
        (pod_manager.go:92-118): list revisions by the DS's selector labels,
        take the highest ``revision``, strip the ``<dsname>-`` name prefix.
        Cached per DS resourceVersion."""
        cache_key = (meta.namespace(daemonset), meta.name(daemonset),
                     meta.resource_version(daemonset))
        cached = self._ds_hash_cache.get(cache_key)
        if cached is not None:
            return cached
        selector = daemonset.get("spec", {}).get("selector", {}).get("matchLabels", {})
        label_selector = ",".join(f"{k}={v}" for k, v in sorted(selector.items()))
        revisions = self._client.list_controller_revisions(
            namespace=meta.namespace(daemonset), label_selector=label_selector
        )
        if not revisions:
            raise StaleClusterViewError(
                f"no ControllerRevisions found for DaemonSet {meta.name(daemonset)}"
            )
        latest = max(revisions, key=lambda r: r.get("revision", 0))
        prefix = f"{meta.name(daemonset)}-"
        name = meta.name(latest)
        hash_ = name[len(prefix):] if name.startswith(prefix) else name
        if len(self._ds_hash_cache) > 256:
            self._ds_hash_cache.clear()
        self._ds_hash_cache[cache_key] = hash_
        return hash_

    # -- eviction (pod_manager.go:122-232) -----------------------------------

    def schedule_pod_eviction(self, config: PodManagerConfig) -> None:
        if not config.nodes:
            logger.info("no nodes scheduled for pod deletion")
            return
        if config.deletion_spec is None:
            raise ValueError("pod deletion spec should not be empty")
        for node in config.nodes:
            node_name = meta.name(node)
            if not self._nodes_in_progress.add_if_absent(node_name):
                logger.info("node %s already getting pods deleted, skipping", node_name)
                continue
            self._spawn(self._evict_node_worker, node, config)

    def _evict_node_worker(self, node: K8sObject, config: PodManagerConfig) -> None:
        node_name = meta.name(node)
        spec = config.deletion_spec
        try:
            pods = self._client.list_pods(
                field_selector=consts.NODE_NAME_FIELD_SELECTOR_FMT.format(node_name)
            )
            to_delete = [p for p in pods if self._filter and self._filter(p)]
            if not to_delete:
                logger.info("no pods require deletion on node %s", node_name)
                self._provider.change_node_upgrade_state(
                    node, consts.UPGRADE_STATE_POD_RESTART_REQUIRED
                )
                return
            plist = get_pods_for_deletion(
                self._client,
                node_name,
                force=spec.force,
                delete_emptydir_data=spec.delete_emptydir_data,
                ignore_daemonsets=True,
                additional_filter=self._filter,
            )
            if len(plist.pods) != len(to_delete):
                logger.error(
                    "cannot delete all required pods on node %s: %s",
                    node_name, plist.errors,
                )
                self._update_node_to_drain_or_failed(node, config.drain_enabled)
                return
            try:
                delete_or_evict_pods(
                    self._client, plist.pods,
                    use_eviction=True, timeout_seconds=spec.timeout_seconds,
                )
            except Exception as exc:
                log_eventf(
                    self._recorder, node, EVENT_TYPE_WARNING, util.get_event_reason(),
                    "Failed to delete workload pods on the node for the driver upgrade, {}",
                    exc,
                )
                self._update_node_to_drain_or_failed(node, config.drain_enabled)
                return
            self._provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_POD_RESTART_REQUIRED
            )
            log_event(self._recorder, node, EVENT_TYPE_NORMAL, util.get_event_reason(),
                      "Deleted workload pods on the node for the driver upgrade")
        except Exception:
            logger.exception("pod eviction worker failed for node %s", node_name)
        finally:
            self._nodes_in_progress.remove(node_name)

    def _update_node_to_drain_or_failed(self, node: K8sObject, drain_enabled: bool) -> None:
        """(pod_manager.go:393-403)"""
        next_state = consts.UPGRADE_STATE_FAILED
        if drain_enabled:
            log_event(self._recorder, node, EVENT_TYPE_WARNING, util.get_event_reason(),
                      "Pod deletion failed but drain is enabled in spec. Will attempt a node drain")
            next_state = consts.UPGRADE_STATE_DRAIN_REQUIRED
        self._provider.change_node_upgrade_state(node, next_state)

    # -- driver pod restart (pod_manager.go:233-251) --------------------------

    def schedule_pods_restart(self, pods: List[K8sObject]) -> None:
        if not pods:
            logger.info("no pods scheduled to restart")
            return
        for pod in pods:
            logger.info("deleting driver pod %s for restart", meta.name(pod))
            try:
                self._client.delete_pod(meta.name(pod), meta.namespace(pod))
            except Exception as exc:
                log_eventf(self._recorder, pod, EVENT_TYPE_WARNING, util.get_event_reason(),
                           "Failed to restart driver pod {}", exc)
                raise

    # -- job-completion wait (pod_manager.go:256-368) --------------------------

    def schedule_check_on_pod_completion(self, config: PodManagerConfig) -> None:
        """Synchronous across nodes (the reference wg.Wait()s its goroutines)."""
        spec = config.wait_for_completion_spec or WaitForCompletionSpec()
        threads = []
        for node in config.nodes:
            t = threading.Thread(
                target=self._completion_check_worker, args=(node, spec), daemon=True
            )
            t.start()
            threads.append(t)
        for t in threads:
            t.join()

    def _completion_check_worker(self, node: K8sObject, spec: WaitForCompletionSpec) -> None:
        node_name = meta.name(node)
        try:
            pods = self._client.list_pods(
                label_selector=spec.pod_selector,
                field_selector=consts.NODE_NAME_FIELD_SELECTOR_FMT.format(node_name),
            )
            running = any(self.is_pod_running_or_pending(p) for p in pods)
            if running:
                logger.info("workload pods still running on node %s", node_name)
                if spec.timeout_seconds != 0:
                    self.handle_timeout_on_pod_completions(node, spec.timeout_seconds)
                return
            # all matching pods finished (or none exist): clear the
            # start-time annotation and move on
            key = util.get_wait_for_pod_completion_start_time_annotation_key()
            if key in (node.get("metadata", {}).get("annotations") or {}):
                self._provider.change_node_upgrade_annotation(node, key, consts.NULL_STRING)
            self._provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_POD_DELETION_REQUIRED
            )
        except Exception:
            logger.exception("pod completion check failed for node %s", node_name)

    def handle_timeout_on_pod_completions(self, node: K8sObject, timeout_seconds: int) -> None:
        """(pod_manager.go:330-368)"""
        key = util.get_wait_for_pod_completion_start_time_annotation_key()
        now = int(time.time())
        annotations = node.get("metadata", {}).get("annotations") or {}
        if key not in annotations:
            self._provider.change_node_upgrade_annotation(node, key, str(now))
            return
        try:
            start_time = int(annotations[key])
        except ValueError:
            # corrupt stamp (manual edit?): re-stamp rather than wedging the
            # phase forever (improves on the reference, which errors out)
            logger.warning("node %s: corrupt completion start-time %r; re-stamping",
                           meta.name(node), annotations[key])
            self._provider.change_node_upgrade_annotation(node, key, str(now))
            return
        if now > start_time + timeout_seconds:
            self._provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_POD_DELETION_REQUIRED
            )
            self._provider.change_node_upgrade_annotation(node, key, consts.NULL_STRING)

    # -- helpers --------------------------------------------------------------

    @staticmethod
    def is_pod_running_or_pending(pod: K8sObject) -> bool:
        """(pod_manager.go:371-391)"""
        return pod.get("status", {}).get("phase") in ("Running", "Pending")

    MAX_CONCURRENT_NODE_WORKERS = 8

    def _spawn(self, target, *args) -> None:
        def bounded():
            with self._worker_slots:
                target(*args)

        t = threading.Thread(target=bounded, daemon=True)
        with self._workers_lock:
            self._workers = [w for w in self._workers if w.is_alive()]
            self._workers.append(t)
        t.start()

    def wait_idle(self, timeout: float = 30.0) -> None:
        """Join all outstanding eviction workers (test/bench determinism)."""
        deadline = time.monotonic() + timeout
        with self._workers_lock:
            workers = list(self._workers)
        for w in workers:
            w.join(max(0.0, deadline - time.monotonic()))
