"""Test fixture builders, the analogue of the reference's
``upgrade_suit_test.go:216-428`` Node/Pod/DaemonSet/NodeMaintenance builders.
Like envtest, nothing runs pods: statuses are set directly on the objects."""

from __future__ import annotations

import uuid

from k8s_operator_libs_amd.upgrade import consts, util

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
                     "template": {"metadata": {"labels": labels}}},
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
