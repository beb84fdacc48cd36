"""Public test fixtures for operators built on this library.

Builder fixtures (the analogue of the reference's suite builders,
upgrade_suit_test.go:216-428) and simulated cluster actors (DaemonSet
controller + kubelet role, maintenance operator) usable against FakeCluster
or the HTTP mini-apiserver.  Consumer test suites import from here:

    from k8s_operator_libs_amd.testing import (
        NodeBuilder, PodBuilder, DaemonSetBuilder, NodeMaintenanceBuilder,
        SimDaemonSetController, SimMaintenanceOperator,
    )
"""

from __future__ import annotations

import threading
import uuid

from ..core import meta
from ..upgrade import util  # noqa: F401  (driver-name-sensitive helpers)

DRIVER_NS = "amd-gpu-operator"
DRIVER_LABELS = {"app": "amdgpu-driver-daemonset"}


class NodeBuilder:
    def __init__(self, name):
        self.obj = {
            "apiVersion": "v1",
            "kind": "Node",
            "metadata": {"name": name, "labels": {}, "annotations": {}},
            "spec": {},
            "status": {"conditions": [{"type": "Ready", "status": "True"}]},
        }

    def with_upgrade_state(self, state):
        self.obj["metadata"]["labels"][util.get_upgrade_state_label_key()] = state
        return self

    def with_label(self, key, value):
        self.obj["metadata"]["labels"][key] = value
        return self

    def with_annotation(self, key, value):
        self.obj["metadata"]["annotations"][key] = value
        return self

    def unschedulable(self, value=True):
        if value:
            self.obj["spec"]["unschedulable"] = True
        return self

    def not_ready(self):
        self.obj["status"]["conditions"] = [{"type": "Ready", "status": "False"}]
        return self

    def build(self, cluster=None):
        if cluster is not None:
            return cluster.create(self.obj)
        return self.obj


class PodBuilder:
    def __init__(self, name, node="", namespace="default"):
        self.obj = {
            "apiVersion": "v1",
            "kind": "Pod",
            "metadata": {
                "name": name, "namespace": namespace,
                "labels": {}, "annotations": {},
            },
            "spec": {"nodeName": node, "containers": [{"name": "main", "image": "x"}]},
            "status": {"phase": "Running",
                       "containerStatuses": [{"name": "main", "ready": True,
                                              "restartCount": 0}]},
        }

    def with_labels(self, labels):
        self.obj["metadata"]["labels"].update(labels)
        return self

    def with_phase(self, phase):
        self.obj["status"]["phase"] = phase
        if phase in ("Succeeded", "Failed"):
            self.obj["status"]["containerStatuses"][0]["ready"] = False
        return self

    def not_ready(self):
        self.obj["status"]["containerStatuses"][0]["ready"] = False
        return self

    def with_restart_count(self, n):
        self.obj["status"]["containerStatuses"][0]["restartCount"] = n
        return self

    def with_resource(self, name, qty="1"):
        self.obj["spec"]["containers"][0].setdefault("resources", {}).setdefault(
            "limits", {}
        )[name] = qty
        return self

    def with_emptydir(self):
        self.obj["spec"].setdefault("volumes", []).append(
            {"name": "scratch", "emptyDir": {}}
        )
        return self

    def with_owner_reference(self, kind, name, controller=True, uid=None):
        self.obj["metadata"].setdefault("ownerReferences", []).append(
            {"apiVersion": "apps/v1", "kind": kind, "name": name,
             "uid": uid or str(uuid.uuid4()), "controller": controller}
        )
        return self

    def with_revision_hash(self, hash_):
        self.obj["metadata"]["labels"]["controller-revision-hash"] = hash_
        return self

    def build(self, cluster=None):
        if cluster is not None:
            return cluster.create(self.obj)
        return self.obj


class DaemonSetBuilder:
    def __init__(self, name, namespace=DRIVER_NS, labels=None):
        labels = dict(labels or DRIVER_LABELS)
        self.obj = {
            "apiVersion": "apps/v1",
            "kind": "DaemonSet",
            "metadata": {"name": name, "namespace": namespace, "labels": labels},
            "spec": {"selector": {"matchLabels": labels},
                     "template": {"metadata": {"labels": labels},
                                  "spec": {"containers": [
                                      {"name": "driver",
                                       "image": "amdgpu-dkms:latest"}]}}},
            "status": {"desiredNumberScheduled": 0, "numberMisscheduled": 0},
        }

    def with_desired_number_scheduled(self, n):
        self.obj["status"]["desiredNumberScheduled"] = n
        return self

    def build(self, cluster=None):
        if cluster is not None:
            return cluster.create(self.obj)
        return self.obj


def make_controller_revision(ds, hash_, revision=1, cluster=None):
    obj = {
        "apiVersion": "apps/v1",
        "kind": "ControllerRevision",
        "metadata": {
            "name": f"{ds['metadata']['name']}-{hash_}",
            "namespace": ds["metadata"]["namespace"],
            "labels": dict(ds["spec"]["selector"]["matchLabels"]),
        },
        "revision": revision,
    }
    if cluster is not None:
        return cluster.create(obj)
    return obj


def driver_pod_for(ds, node, hash_="rev1", ready=True, namespace=None):
    """A driver DaemonSet pod on the given node with a revision hash."""
    b = (
        PodBuilder(f"{ds['metadata']['name']}-{node}", node=node,
                   namespace=namespace or ds["metadata"]["namespace"])
        .with_labels(dict(ds["spec"]["selector"]["matchLabels"]))
        .with_owner_reference("DaemonSet", ds["metadata"]["name"],
                              uid=ds["metadata"].get("uid"))
        .with_revision_hash(hash_)
    )
    if not ready:
        b.not_ready()
    return b


class NodeMaintenanceBuilder:
    def __init__(self, name, namespace="default"):
        self.obj = {
            "apiVersion": "maintenance.amd.com/v1alpha1",
            "kind": "NodeMaintenance",
            "metadata": {"name": name, "namespace": namespace},
            "spec": {},
            "status": {},
        }

    def with_node(self, node_name):
        self.obj["spec"]["nodeName"] = node_name
        return self

    def with_requestor(self, requestor_id):
        self.obj["spec"]["requestorID"] = requestor_id
        return self

    def with_conditions(self, cond_type, status="True", reason=""):
        self.obj.setdefault("status", {}).setdefault("conditions", []).append(
            {"type": cond_type, "status": status, "reason": reason}
        )
        return self

    def build(self, cluster=None):
        if cluster is not None:
            return cluster.create(self.obj)
        return self.obj

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

    def __init__(self, cluster, evict=True, evict_filter=None):
        """``evict_filter(pod) -> bool``: which non-DaemonSet pods to evict
        during maintenance (the real operator applies the NodeMaintenance
        spec's podEvictionFilters); default evicts all of them."""
        self.cluster = cluster
        self.evict = evict
        self.evict_filter = evict_filter
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
                                if self.evict_filter is not None and                                         not self.evict_filter(pod):
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

class ClientHookAdapter:
    """Give any :class:`~k8s_operator_libs_amd.core.client.Client` the
    FakeCluster surface the simulated actors use (CRUD passthrough plus
    ``add_change_hook``), feeding hooks from watch streams instead of
    in-process mutation hooks.

    This is what lets the same :class:`SimDaemonSetController` /
    :class:`SimMaintenanceOperator` / :class:`SimKubelet` drive a REAL
    kube-apiserver (tests/test_conformance.py) where no change-hook exists.
    Two real-apiserver differences are papered over:

    - ``create`` with an inline ``status`` (the builder convention, matching
      envtest's forced Status().Update pattern, reference
      upgrade_suit_test.go:344-355) becomes create + status patch, because a
      real apiserver drops status on create;
    - events arrive asynchronously, so hooks must be level-triggered (all of
      the in-repo sims are).
    """

    def __init__(self, client, kinds=(("v1", "Pod"),)):
        self._client = client
        self._hooks = []
        self._lock = threading.RLock()
        self._stopped = threading.Event()
        self._watch_threads = []
        self._watches = []
        for api_version, kind in kinds:
            w = client.watch(api_version, kind)
            self._watches.append(w)
            t = threading.Thread(
                target=self._pump, args=(w, api_version, kind), daemon=True
            )
            t.start()
            self._watch_threads.append(t)

    def _pump(self, watch, api_version, kind):
        while not self._stopped.is_set():
            item = watch.next(timeout=0.2)
            if item is None:
                alive = getattr(watch, "alive", None)
                if alive is not None and not alive():
                    if self._stopped.is_set():
                        return
                    try:
                        watch = self._client.watch(api_version, kind)
                    except Exception:
                        self._stopped.wait(0.5)
                continue
            event_type, obj = item
            if event_type in ("BOOKMARK", "ERROR") or obj is None:
                continue
            with self._lock:
                hooks = list(self._hooks)
            for hook in hooks:
                try:
                    hook(event_type, obj)
                except Exception:
                    import logging

                    logging.getLogger(__name__).exception("sim hook failed")

    def add_change_hook(self, hook):
        with self._lock:
            self._hooks.append(hook)

    def stop(self):
        self._stopped.set()
        for w in self._watches:
            w.stop()

    # -- FakeCluster-flavoured CRUD over the client ---------------------------

    def create(self, obj):
        status = obj.pop("status", None)
        created = self._client.create(obj)
        if status:
            created = self._client.patch_status(
                meta.api_version(created), meta.kind(created),
                meta.name(created), status, meta.namespace(created),
            )
        return created

    def get(self, api_version, kind, name, namespace=""):
        return self._client.get(api_version, kind, name, namespace)

    def list(self, api_version, kind, namespace=None, label_selector="",
             field_selector=""):
        return self._client.list(api_version, kind, namespace=namespace,
                                 label_selector=label_selector,
                                 field_selector=field_selector)

    def patch(self, api_version, kind, name, patch, namespace=""):
        if set(patch) == {"status"}:
            # sims patch status through the subresource on real servers
            return self._client.patch_status(api_version, kind, name,
                                             patch["status"], namespace)
        return self._client.patch(api_version, kind, name, patch, namespace)

    def delete(self, api_version, kind, name, namespace=""):
        self._client.delete(api_version, kind, name, namespace)

    def evict_pod(self, name, namespace):
        self._client.evict_pod(name, namespace)


class SimKubelet:
    """The kubelet role a control-plane-only cluster is missing.

    envtest boots no kubelet, so (a) a deleted pod lingers Terminating
    forever (graceful deletion waits for kubelet confirmation) and (b) fresh
    pods never go Running/Ready.  This sim completes both halves: it
    force-deletes pods carrying a deletionTimestamp (the kubelet's grace-0
    confirmation) and patches phase=Running + ready containerStatuses onto
    new pods.  On FakeCluster it is unnecessary (deletes are immediate,
    builders set status) but harmless."""

    def __init__(self, cluster, ready=True):
        self.cluster = cluster
        self.ready = ready
        self._lock = threading.RLock()
        cluster.add_change_hook(self._on_change)

    def _on_change(self, event_type, obj):
        if meta.kind(obj) != "Pod" or event_type == "DELETED":
            return
        with self._lock:
            md = obj.get("metadata", {})
            name, ns = md.get("name", ""), md.get("namespace", "default")
            if "deletionTimestamp" in md:
                client = getattr(self.cluster, "_client", None)
                try:
                    if client is not None:
                        client.delete("v1", "Pod", name, ns,
                                      grace_period_seconds=0)
                    else:
                        self.cluster.delete("v1", "Pod", name, ns)
                except Exception:
                    pass
                return
            status = obj.get("status", {})
            if status.get("phase") != "Running" or not status.get("containerStatuses"):
                containers = obj.get("spec", {}).get("containers", [])
                try:
                    self.cluster.patch("v1", "Pod", name, {"status": {
                        "phase": "Running",
                        "containerStatuses": [
                            {"name": c.get("name", "c"), "ready": self.ready,
                             "restartCount": 0,
                             "state": {"running": {}}}
                            for c in containers
                        ],
                    }}, ns)
                except Exception:
                    pass
