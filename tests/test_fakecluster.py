"""Tests for the in-memory apiserver substrate (core.fakecluster) and the
selector/patch helpers (core.meta)."""

import pytest

from k8s_operator_libs_amd.core import (
    AlreadyExistsError,
    ConflictError,
    FakeClient,
    FakeCluster,
    NotFoundError,
)
from k8s_operator_libs_amd.core.meta import (
    FieldSelector,
    LabelSelector,
    json_merge_patch,
)


def mk_node(name, labels=None, unschedulable=False):
    n = {
        "apiVersion": "v1",
        "kind": "Node",
        "metadata": {"name": name, "labels": labels or {}},
        "spec": {},
        "status": {"conditions": [{"type": "Ready", "status": "True"}]},
    }
    if unschedulable:
        n["spec"]["unschedulable"] = True
    return n


def mk_pod(name, node="", namespace="default", labels=None, phase="Running"):
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {"name": name, "namespace": namespace, "labels": labels or {}},
        "spec": {"nodeName": node},
        "status": {"phase": phase},
    }


class TestCrud:
    def test_create_get_roundtrip(self, cluster):
        created = cluster.create(mk_node("n1", {"gpu": "mi355x"}))
        assert created["metadata"]["uid"]
        assert created["metadata"]["resourceVersion"]
        got = cluster.get("v1", "Node", "n1")
        assert got["metadata"]["labels"]["gpu"] == "mi355x"

    def test_create_duplicate_fails(self, cluster):
        cluster.create(mk_node("n1"))
        with pytest.raises(AlreadyExistsError):
            cluster.create(mk_node("n1"))

    def test_get_missing_raises(self, cluster):
        with pytest.raises(NotFoundError):
            cluster.get("v1", "Node", "nope")

    def test_generate_name(self, cluster):
        p = cluster.create(
            {"apiVersion": "v1", "kind": "Pod",
             "metadata": {"generateName": "driver-", "namespace": "ops"},
             "spec": {}}
        )
        assert p["metadata"]["name"].startswith("driver-")

    def test_update_bumps_rv_and_conflicts_on_stale(self, cluster):
        cluster.create(mk_node("n1"))
        a = cluster.get("v1", "Node", "n1")
        b = cluster.get("v1", "Node", "n1")
        a["metadata"]["labels"]["x"] = "1"
        cluster.update(a)
        b["metadata"]["labels"]["x"] = "2"
        with pytest.raises(ConflictError):
            cluster.update(b)

    def test_delete(self, cluster):
        cluster.create(mk_node("n1"))
        cluster.delete("v1", "Node", "n1")
        with pytest.raises(NotFoundError):
            cluster.get("v1", "Node", "n1")
        with pytest.raises(NotFoundError):
            cluster.delete("v1", "Node", "n1")

    def test_namespacing(self, cluster):
        cluster.create(mk_pod("p", namespace="a"))
        cluster.create(mk_pod("p", namespace="b"))
        assert len(cluster.list("v1", "Pod")) == 2
        assert len(cluster.list("v1", "Pod", namespace="a")) == 1


class TestListSelectors:
    def test_label_selector(self, cluster):
        cluster.create(mk_pod("p1", labels={"app": "train", "tier": "gpu"}))
        cluster.create(mk_pod("p2", labels={"app": "serve"}))
        out = cluster.list("v1", "Pod", label_selector="app=train")
        assert [p["metadata"]["name"] for p in out] == ["p1"]
        out = cluster.list("v1", "Pod", label_selector="app in (train,serve)")
        assert len(out) == 2
        out = cluster.list("v1", "Pod", label_selector="tier")
        assert [p["metadata"]["name"] for p in out] == ["p1"]

    def test_field_selector_node_name(self, cluster):
        cluster.create(mk_pod("p1", node="n1"))
        cluster.create(mk_pod("p2", node="n2"))
        out = cluster.list("v1", "Pod", field_selector="spec.nodeName=n1")
        assert [p["metadata"]["name"] for p in out] == ["p1"]


class TestPatch:
    def test_merge_patch_labels(self, cluster):
        cluster.create(mk_node("n1", {"a": "1", "b": "2"}))
        cluster.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": "9", "c": "3"}}})
        got = cluster.get("v1", "Node", "n1")
        assert got["metadata"]["labels"] == {"a": "9", "b": "2", "c": "3"}

    def test_merge_patch_null_deletes(self, cluster):
        cluster.create(mk_node("n1", {"a": "1"}))
        cluster.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": None}}})
        assert "a" not in cluster.get("v1", "Node", "n1")["metadata"]["labels"]

    def test_optimistic_lock_patch(self, cluster):
        cluster.create(mk_node("n1"))
        rv = cluster.get("v1", "Node", "n1")["metadata"]["resourceVersion"]
        cluster.patch("v1", "Node", "n1", {"metadata": {"resourceVersion": rv, "labels": {"x": "1"}}})
        with pytest.raises(ConflictError):
            cluster.patch(
                "v1", "Node", "n1",
                {"metadata": {"resourceVersion": rv, "labels": {"x": "2"}}},
            )


class TestFinalizers:
    def test_delete_with_finalizer_defers(self, cluster):
        nm = {
            "apiVersion": "maintenance.amd.com/v1alpha1",
            "kind": "NodeMaintenance",
            "metadata": {"name": "m1", "namespace": "ops", "finalizers": ["ext/guard"]},
            "spec": {"nodeName": "n1"},
        }
        cluster.create(nm)
        cluster.delete("maintenance.amd.com/v1alpha1", "NodeMaintenance", "m1", "ops")
        got = cluster.get("maintenance.amd.com/v1alpha1", "NodeMaintenance", "m1", "ops")
        assert "deletionTimestamp" in got["metadata"]
        # external operator removes the finalizer -> object goes away
        cluster.patch(
            "maintenance.amd.com/v1alpha1", "NodeMaintenance", "m1",
            {"metadata": {"finalizers": []}}, "ops",
        )
        with pytest.raises(NotFoundError):
            cluster.get("maintenance.amd.com/v1alpha1", "NodeMaintenance", "m1", "ops")


class TestWatchAndEviction:
    def test_watch_sees_lifecycle(self, cluster):
        w = cluster.watch("v1", "Node")
        cluster.create(mk_node("n1"))
        cluster.patch("v1", "Node", "n1", {"metadata": {"labels": {"s": "1"}}})
        cluster.delete("v1", "Node", "n1")
        events = [w.next(timeout=1)[0] for _ in range(3)]
        assert events == ["ADDED", "MODIFIED", "DELETED"]
        w.stop()

    def test_evict_pod(self, cluster):
        cluster.create(mk_pod("p1"))
        cluster.evict_pod("p1", "default")
        with pytest.raises(NotFoundError):
            cluster.get("v1", "Pod", "p1", "default")


class TestCrdRegistration:
    def test_created_crd_becomes_servable(self, cluster):
        crd = {
            "apiVersion": "apiextensions.k8s.io/v1",
            "kind": "CustomResourceDefinition",
            "metadata": {"name": "widgets.amd.com"},
            "spec": {
                "group": "amd.com",
                "scope": "Namespaced",
                "names": {"kind": "Widget", "plural": "widgets"},
                "versions": [{"name": "v1", "served": True}],
            },
        }
        cluster.create(crd)
        assert cluster.lookup_by_plural("amd.com/v1", "widgets") == "Widget"
        cluster.create(
            {"apiVersion": "amd.com/v1", "kind": "Widget",
             "metadata": {"name": "w1", "namespace": "default"}}
        )
        assert cluster.get("amd.com/v1", "Widget", "w1", "default")


class TestClientFacade:
    def test_typed_helpers(self, client):
        client.create(mk_node("n1"))
        client.create(mk_pod("p1", node="n1"))
        assert client.get_node("n1")["metadata"]["name"] == "n1"
        assert len(client.list_pods(field_selector="spec.nodeName=n1")) == 1


class TestMetaHelpers:
    def test_label_selector_negations(self):
        sel = LabelSelector("a!=x,!b,c")
        assert sel.matches({"c": "1"})
        assert sel.matches({"a": "y", "c": "1"})
        assert not sel.matches({"a": "x", "c": "1"})
        assert not sel.matches({"b": "1", "c": "1"})
        assert not sel.matches({})

    def test_field_selector(self):
        fs = FieldSelector("spec.nodeName=n1,status.phase!=Failed")
        assert fs.matches_object({"spec": {"nodeName": "n1"}, "status": {"phase": "Running"}})
        assert not fs.matches_object({"spec": {"nodeName": "n2"}, "status": {"phase": "Running"}})
        assert not fs.matches_object({"spec": {"nodeName": "n1"}, "status": {"phase": "Failed"}})

    def test_json_merge_patch_nested(self):
        t = {"a": {"b": 1, "c": 2}, "d": 3}
        json_merge_patch(t, {"a": {"b": None, "e": 4}, "d": 5})
        assert t == {"a": {"c": 2, "e": 4}, "d": 5}


class TestRfc7386Vectors:
    """The JSON merge patch test cases from RFC 7386 Appendix A."""

    VECTORS = [
        ({"a": "b"}, {"a": "c"}, {"a": "c"}),
        ({"a": "b"}, {"b": "c"}, {"a": "b", "b": "c"}),
        ({"a": "b"}, {"a": None}, {}),
        ({"a": "b", "b": "c"}, {"a": None}, {"b": "c"}),
        ({"a": ["b"]}, {"a": "c"}, {"a": "c"}),
        ({"a": "c"}, {"a": ["b"]}, {"a": ["b"]}),
        ({"a": {"b": "c"}}, {"a": {"b": "d", "c": None}}, {"a": {"b": "d"}}),
        ({"a": [{"b": "c"}]}, {"a": [1]}, {"a": [1]}),
        (["a", "b"], ["c", "d"], ["c", "d"]),
        ({"a": "b"}, ["c"], ["c"]),
        ({"a": "foo"}, None, None),
        ({"a": "foo"}, "bar", "bar"),
        ({"e": None}, {"a": 1}, {"e": None, "a": 1}),
        ([1, 2], {"a": "b", "c": None}, {"a": "b"}),
        ({}, {"a": {"bb": {"ccc": None}}}, {"a": {"bb": {}}}),
    ]

    def test_all_vectors(self):
        for original, patch, expected in self.VECTORS:
            result = json_merge_patch(
                original if not isinstance(original, dict) else dict(original),
                patch,
            )
            assert result == expected, (original, patch, result, expected)


class TestDumpLoad:
    def test_roundtrip_preserves_objects_and_kinds(self, cluster):
        cluster.create(mk_node("n1", {"s": "done"}))
        cluster.create(mk_pod("p1", node="n1", labels={"app": "x"}))
        cluster.register_kind("x.amd.com/v1", "Thing", "things", True)
        cluster.create({"apiVersion": "x.amd.com/v1", "kind": "Thing",
                        "metadata": {"name": "t", "namespace": "default"},
                        "spec": {"v": 1}})
        snap = cluster.dump()
        clone = FakeCluster.load(snap)
        assert clone.get("v1", "Node", "n1")["metadata"]["labels"]["s"] == "done"
        assert clone.get("x.amd.com/v1", "Thing", "t", "default")["spec"]["v"] == 1
        # pod index rebuilt: per-node list works on the clone
        assert len(clone.list("v1", "Pod", field_selector="spec.nodeName=n1")) == 1

    def test_replay_continues_upgrade(self, cluster):
        """A dumped mid-upgrade cluster replays to completion — the
        checkpoint/resume story in one test."""
        from k8s_operator_libs_amd.core import FakeClient
        from k8s_operator_libs_amd.upgrade.state_manager import (
            ClusterUpgradeStateManager,
        )
        from builders import DRIVER_LABELS, DRIVER_NS
        from simenv import SimDaemonSetController
        from test_state_manager import policy, setup_cluster, state_of

        client = FakeClient(cluster)
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        manager = ClusterUpgradeStateManager(client)
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%")
        for _ in range(3):  # advance partway
            state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
            manager.apply_state(state, pol)
            manager.wait_idle()
        mid_state = state_of(client, "node-0")
        assert mid_state not in ("", "upgrade-done")
        # dump, load elsewhere, continue with a fresh manager
        clone_client = FakeClient(FakeCluster.load(cluster.dump()))
        ds_clone = clone_client.get("apps/v1", "DaemonSet", "amdgpu-driver", DRIVER_NS)
        SimDaemonSetController(clone_client.cluster, ds_clone, current_hash="new")
        clone_mgr = ClusterUpgradeStateManager(clone_client)
        for _ in range(12):
            clone_mgr.reconcile(DRIVER_NS, DRIVER_LABELS, pol, converge=True)
            if state_of(clone_client, "node-0") == "upgrade-done":
                break
        assert state_of(clone_client, "node-0") == "upgrade-done"


class TestOptimisticConcurrencyUnderContention:
    def test_rv_locked_patches_never_lose_updates(self, cluster):
        """4 writers x 50 increments through RV-locked merge patches with
        conflict retry: every increment must land (no lost updates)."""
        import threading

        cluster.create(mk_node("n1"))
        cluster.patch("v1", "Node", "n1",
                      {"metadata": {"annotations": {"counter": "0"}}})
        errors = []

        def writer():
            try:
                for _ in range(50):
                    while True:
                        live = cluster.get("v1", "Node", "n1")
                        val = int(live["metadata"]["annotations"]["counter"])
                        try:
                            cluster.patch("v1", "Node", "n1", {
                                "metadata": {
                                    "resourceVersion": live["metadata"]["resourceVersion"],
                                    "annotations": {"counter": str(val + 1)},
                                }
                            })
                            break
                        except ConflictError:
                            continue
            except Exception as exc:
                errors.append(exc)

        threads = [threading.Thread(target=writer) for _ in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
        assert not errors, errors
        final = cluster.get("v1", "Node", "n1")["metadata"]["annotations"]["counter"]
        assert final == "200"


class TestDumpLoadWithCrdSchemas:
    def test_snapshot_restores_cr_status_and_enforcement(self):
        """dump/load with a status-subresource CRD: statuses survive the
        round trip (restored via /status) and schema enforcement is
        re-armed on the loaded cluster."""
        from k8s_operator_libs_amd.core.errors import InvalidError
        from k8s_operator_libs_amd.core.fakecluster import FakeCluster

        c = FakeCluster()
        c.create({
            "apiVersion": "apiextensions.k8s.io/v1",
            "kind": "CustomResourceDefinition",
            "metadata": {"name": "things.snap.amd.com"},
            "spec": {
                "group": "snap.amd.com", "scope": "Namespaced",
                "names": {"kind": "Thing", "plural": "things",
                          "singular": "thing"},
                "versions": [{
                    "name": "v1", "served": True, "storage": True,
                    "subresources": {"status": {}},
                    "schema": {"openAPIV3Schema": {
                        "type": "object",
                        "properties": {
                            "spec": {"type": "object",
                                     "properties": {"n": {"type": "integer"}}},
                            "status": {"type": "object",
                                       "properties": {"ok": {"type": "boolean"}}},
                        },
                    }},
                }],
            },
        })
        c.create({"apiVersion": "snap.amd.com/v1", "kind": "Thing",
                  "metadata": {"name": "t1", "namespace": "default"},
                  "spec": {"n": 5}})
        c.patch_status("snap.amd.com/v1", "Thing", "t1", {"ok": True},
                       "default")

        loaded = FakeCluster.load(c.dump())
        got = loaded.get("snap.amd.com/v1", "Thing", "t1", "default")
        assert got["spec"]["n"] == 5
        assert got["status"]["ok"] is True  # status survived the round trip
        # schema enforcement re-armed
        import pytest as _pytest

        with _pytest.raises(InvalidError):
            loaded.create({"apiVersion": "snap.amd.com/v1", "kind": "Thing",
                           "metadata": {"name": "t2", "namespace": "default"},
                           "spec": {"n": "NaN"}})
