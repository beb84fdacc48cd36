"""Event-driven upgrade controller: the controller-runtime manager analogue.

The reference is a library whose consumers wire it into controller-runtime:
watches on Nodes/NodeMaintenance (filtered by the predicates in
:mod:`k8s_operator_libs_amd.upgrade.requestor`) trigger Reconcile, which
calls BuildState + ApplyState.  This module provides that wiring natively so
an operator built on this library is complete without any external
framework:

- watches Nodes, driver Pods, and (in requestor mode) NodeMaintenance
  objects through any :class:`~k8s_operator_libs_amd.core.client.Client`
  with watch support;
- coalesces events (a reconcile is already pending -> new events fold into
  it) and applies the requestor-ID / condition-changed predicates so
  irrelevant NodeMaintenance churn does not wake the loop;
- falls back to a resync interval (like an informer's periodic resync) so
  missed events can never wedge the machine — the state machine is
  idempotent, so spurious wakeups are merely cheap.
"""

from __future__ import annotations

import logging
import threading
from typing import Dict

from ..api.upgrade.v1alpha1 import DriverUpgradePolicySpec
from ..core import meta
from ..core.client import Client
from .requestor import (
    NODE_MAINTENANCE_API_VERSION,
    NODE_MAINTENANCE_KIND,
    condition_changed_predicate,
    requestor_id_predicate,
)
from .state_manager import ClusterUpgradeStateManager

logger = logging.getLogger(__name__)


class UpgradeController:
    def __init__(
        self,
        manager: ClusterUpgradeStateManager,
        namespace: str,
        driver_labels: Dict[str, str],
        policy: DriverUpgradePolicySpec,
        resync_seconds: float = 30.0,
        converge: bool = True,
    ) -> None:
        self.manager = manager
        self.namespace = namespace
        self.driver_labels = driver_labels
        self.policy = policy
        self.resync_seconds = resync_seconds
        self.converge = converge
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._watch_threads = []
        self._watches = []
        self.reconcile_count = 0

    # -- watch wiring --------------------------------------------------------

    def _start_watch(self, api_version: str, kind: str, predicate=None) -> None:
        client: Client = self.manager.common.client
        watch = client.watch(api_version, kind)
        self._watches.append(watch)

        last_seen: Dict[str, dict] = {}

        def pump():
            while not self._stop.is_set():
                item = watch.next(timeout=0.5)
                if item is None:
                    continue
                event_type, obj = item
                if predicate is not None and obj is not None:
                    key = f"{meta.namespace(obj)}/{meta.name(obj)}"
                    old = last_seen.get(key)
                    last_seen[key] = obj
                    if event_type == "MODIFIED" and not predicate(old, obj):
                        continue
                self._wake.set()

        t = threading.Thread(target=pump, daemon=True)
        t.start()
        self._watch_threads.append(t)

    def start_watches(self) -> None:
        self._start_watch("v1", "Node")
        self._start_watch("v1", "Pod")
        if self.manager.requestor is not None:
            requestor_id = self.manager.opts.requestor.requestor_id
            id_pred = requestor_id_predicate(requestor_id)

            def nm_predicate(old, new):
                if new is not None and not id_pred(new):
                    return False
                return condition_changed_predicate(old, new)

            self._start_watch(
                NODE_MAINTENANCE_API_VERSION, NODE_MAINTENANCE_KIND, nm_predicate
            )

    # -- loop ----------------------------------------------------------------

    def reconcile_once(self) -> dict:
        state = self.manager.reconcile(
            self.namespace, self.driver_labels, self.policy, converge=self.converge
        )
        self.reconcile_count += 1
        return self.manager.counts(state)

    def run(self, until_all_done: bool = False, max_reconciles: int = 0) -> bool:
        """Run until stopped.  With ``until_all_done`` returns True once every
        managed node reports upgrade-done (used by tests/demos)."""
        self.start_watches()
        try:
            while not self._stop.is_set():
                try:
                    counts = self.reconcile_once()
                except Exception:
                    logger.exception("reconcile failed; will retry on next event")
                    counts = None
                if (
                    until_all_done
                    and counts
                    and counts["total"] > 0
                    and counts["done"] == counts["total"]
                ):
                    return True
                if max_reconciles and self.reconcile_count >= max_reconciles:
                    return False
                self._wake.wait(self.resync_seconds)
                self._wake.clear()
            return False
        finally:
            self.stop_watches()

    def run_with_leader_election(
        self,
        lease_name: str = "amd-gpu-operator-upgrade",
        lease_namespace: str = "default",
        identity: str = "",
        **elector_kwargs,
    ) -> None:
        """Run the reconcile loop under Lease-based leader election (the
        controller-runtime manager's election): only the
        lease holder reconciles; on lost leadership the loop stops so a
        demoted replica can never split-brain the state machine.  Blocks
        until :meth:`stop` or leadership is lost."""
        from ..core.leaderelection import LeaderElector

        elector = LeaderElector(
            self.manager.common.client, lease_name,
            namespace=lease_namespace,
            **({"identity": identity} if identity else {}),
            **elector_kwargs,
        )
        self._elector = elector

        def stop_all():
            self.stop()

        try:
            elector.run(on_started_leading=self.run,
                        on_stopped_leading=stop_all)
        finally:
            elector.stop()

    def wake(self) -> None:
        self._wake.set()

    def stop(self) -> None:
        self._stop.set()
        self._wake.set()
        elector = getattr(self, "_elector", None)
        if elector is not None:
            elector.stop()

    def stop_watches(self) -> None:
        self._stop.set()
        for w in self._watches:
            w.stop()
        for t in self._watch_threads:
            t.join(timeout=2)
        self._watches.clear()
        self._watch_threads.clear()
