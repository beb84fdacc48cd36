"""Simulated cluster actors for end-to-end state machine tests and benchmarks.

envtest (and our FakeCluster) has no kubelet or controllers; these helpers
play the missing roles deterministically:

- :class:`SimDaemonSetController` recreates deleted driver pods with the
  DaemonSet's current revision hash (Running+Ready), like the real DS
  controller + kubelet would after a driver-pod restart.
- :class:`SimMaintenanceOperator` reconciles NodeMaintenance objects:
  cordons the node, evicts matching pods, marks the Ready condition, and on
  deletion uncordons and removes its finalizer.
"""

from __future__ import annotations

import threading

from k8s_operator_libs_amd.core import meta


class SimDaemonSetController:
    """Recreate driver pods deleted during pod-restart with the current
    DaemonSet revision hash."""

    def __init__(self, cluster, ds, current_hash, ready=True):
        self.cluster = cluster
        self.ds = ds
        self.current_hash = current_hash
        self.ready = ready
        self._lock = threading.RLock()
        cluster.add_change_hook(self._on_change)

    def _on_change(self, event_type, obj):
        if event_type != "DELETED" or meta.kind(obj) != "Pod":
            return
        refs = meta.owner_references(obj)
        if not refs or refs[0].get("uid") != meta.uid(self.ds):
            return
        with self._lock:
            node = obj["spec"].get("nodeName", "")
            labels = dict(self.ds["spec"]["selector"]["matchLabels"])
            labels["controller-revision-hash"] = self.current_hash
            new_pod = {
                "apiVersion": "v1",
                "kind": "Pod",
                "metadata": {
                    "name": obj["metadata"]["name"],
                    "namespace": obj["metadata"]["namespace"],
                    "labels": labels,
                    "ownerReferences": [
                        {"apiVersion": "apps/v1", "kind": "DaemonSet",
                         "name": meta.name(self.ds), "uid": meta.uid(self.ds),
                         "controller": True}
                    ],
                },
                "spec": {"nodeName": node,
                         "containers": [{"name": "driver", "image": "amdgpu-dkms:new"}]},
                "status": {"phase": "Running",
                           "containerStatuses": [
                               {"name": "driver", "ready": self.ready, "restartCount": 0}
                           ]},
            }
            self.cluster.create(new_pod)


class SimMaintenanceOperator:
    """Minimal maintenance-operator: drive NodeMaintenance objects to Ready
    and honour deletion with a finalizer (uncordon on release)."""

    FINALIZER = "maintenance.amd.com/guard"

    def __init__(self, cluster, evict=True):
        self.cluster = cluster
        self.evict = evict
        self._lock = threading.RLock()
        cluster.add_change_hook(self._on_change)

    def _on_change(self, event_type, obj):
        if meta.kind(obj) != "NodeMaintenance":
            return
        with self._lock:
            if event_type in ("ADDED", "MODIFIED"):
                try:
                    live = self.cluster.get(
                        "maintenance.amd.com/v1alpha1", "NodeMaintenance",
                        meta.name(obj), meta.namespace(obj),
                    )
                except Exception:
                    return
                node_name = live.get("spec", {}).get("nodeName", "")
                if "deletionTimestamp" in live["metadata"]:
                    # release: uncordon node, drop finalizer
                    if node_name:
                        try:
                            self.cluster.patch("v1", "Node", node_name,
                                               {"spec": {"unschedulable": None}})
                        except Exception:
                            pass
                    self.cluster.patch(
                        "maintenance.amd.com/v1alpha1", "NodeMaintenance",
                        meta.name(live), {"metadata": {"finalizers": []}},
                        meta.namespace(live),
                    )
                    return
                fins = live["metadata"].get("finalizers") or []
                conds = live.get("status", {}).get("conditions") or []
                if self.FINALIZER not in fins:
                    self.cluster.patch(
                        "maintenance.amd.com/v1alpha1", "NodeMaintenance",
                        meta.name(live),
                        {"metadata": {"finalizers": fins + [self.FINALIZER]}},
                        meta.namespace(live),
                    )
                if not any(c.get("type") == "Ready" for c in conds):
                    # perform maintenance: cordon + evict workload pods
                    if node_name:
                        self.cluster.patch("v1", "Node", node_name,
                                           {"spec": {"unschedulable": True}})
                        if self.evict:
                            for pod in self.cluster.list(
                                "v1", "Pod",
                                field_selector=f"spec.nodeName={node_name}",
                            ):
                                refs = meta.owner_references(pod)
                                if refs and refs[0].get("kind") == "DaemonSet":
                                    continue
                                try:
                                    self.cluster.delete("v1", "Pod", meta.name(pod),
                                                        meta.namespace(pod))
                                except Exception:
                                    pass
                    self.cluster.patch(
                        "maintenance.amd.com/v1alpha1", "NodeMaintenance",
                        meta.name(live),
                        {"status": {"conditions": [
                            {"type": "Ready", "status": "True", "reason": "Ready"}
                        ]}},
                        meta.namespace(live),
                    )
