"""DrainManager + drain engine tests
(reference pkg/upgrade/drain_manager_test.go:33-160)."""

import pytest

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import DrainSpec
from k8s_operator_libs_amd.core.errors import NotFoundError
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.drain import get_pods_for_deletion
from k8s_operator_libs_amd.upgrade.drain_manager import DrainConfiguration, DrainManager
from k8s_operator_libs_amd.upgrade.node_state_provider import NodeUpgradeStateProvider

from builders import DaemonSetBuilder, NodeBuilder, PodBuilder, driver_pod_for


def state_of(client, node_name):
    return (
        client.get_node(node_name)["metadata"]["labels"]
        .get(util.get_upgrade_state_label_key(), "")
    )


@pytest.fixture
def provider(client):
    return NodeUpgradeStateProvider(client)


@pytest.fixture
def manager(client, provider):
    return DrainManager(client, provider)


def drain_node_fixture(client, name="n1"):
    return (
        NodeBuilder(name)
        .with_upgrade_state(consts.UPGRADE_STATE_DRAIN_REQUIRED)
        .build(client.cluster)
    )


class TestDrainEngine:
    def test_daemonset_pods_skipped(self, client):
        ds = DaemonSetBuilder("amdgpu-driver").build(client.cluster)
        driver_pod_for(ds, "n1").build(client.cluster)
        PodBuilder("workload", node="n1").with_owner_reference(
            "ReplicaSet", "rs"
        ).build(client.cluster)
        plist = get_pods_for_deletion(client, "n1")
        assert [p["metadata"]["name"] for p in plist.pods] == ["workload"]
        assert len(plist.skipped) == 1
        assert not plist.errors

    def test_mirror_pod_skipped(self, client):
        p = PodBuilder("static-pod", node="n1").build()
        p["metadata"]["annotations"]["kubernetes.io/config.mirror"] = "x"
        client.create(p)
        plist = get_pods_for_deletion(client, "n1")
        assert not plist.pods and not plist.errors and len(plist.skipped) == 1

    def test_finished_bare_pod_deletable_without_force(self, client):
        PodBuilder("done", node="n1").with_phase("Succeeded").build(client.cluster)
        plist = get_pods_for_deletion(client, "n1", force=False)
        assert [p["metadata"]["name"] for p in plist.pods] == ["done"]

    def test_pod_selector_narrows(self, client):
        PodBuilder("a", node="n1").with_labels({"team": "x"}).with_owner_reference(
            "ReplicaSet", "rs"
        ).build(client.cluster)
        PodBuilder("b", node="n1").with_owner_reference("ReplicaSet", "rs").build(client.cluster)
        plist = get_pods_for_deletion(client, "n1", pod_selector="team=x")
        assert [p["metadata"]["name"] for p in plist.pods] == ["a"]


class TestDrainManager:
    def test_drains_all_given_nodes(self, client, provider, manager):
        nodes = [drain_node_fixture(client, f"n{i}") for i in range(3)]
        for i in range(3):
            PodBuilder(f"w{i}", node=f"n{i}").with_owner_reference(
                "ReplicaSet", "rs"
            ).build(client.cluster)
        manager.schedule_nodes_drain(
            DrainConfiguration(spec=DrainSpec(enable=True), nodes=nodes)
        )
        manager.wait_idle()
        for i in range(3):
            assert state_of(client, f"n{i}") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED
            assert client.get_node(f"n{i}")["spec"].get("unschedulable") is True
            with pytest.raises(NotFoundError):
                client.get("v1", "Pod", f"w{i}", "default")

    def test_empty_node_list_noop(self, client, manager):
        manager.schedule_nodes_drain(
            DrainConfiguration(spec=DrainSpec(enable=True), nodes=[])
        )
        manager.wait_idle()

    def test_nil_spec_raises(self, client, manager):
        with pytest.raises(ValueError):
            manager.schedule_nodes_drain(DrainConfiguration(spec=None, nodes=[]))

    def test_disabled_spec_noop(self, client, provider, manager):
        node = drain_node_fixture(client)
        manager.schedule_nodes_drain(
            DrainConfiguration(spec=DrainSpec(enable=False), nodes=[node])
        )
        manager.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_DRAIN_REQUIRED

    def test_blocked_drain_fails_node(self, client, provider, manager):
        node = drain_node_fixture(client)
        PodBuilder("bare", node="n1").build(client.cluster)  # no controller, no force
        manager.schedule_nodes_drain(
            DrainConfiguration(spec=DrainSpec(enable=True, force=False), nodes=[node])
        )
        manager.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_FAILED

    def test_driver_daemonset_pod_survives_drain(self, client, provider, manager):
        node = drain_node_fixture(client)
        ds = DaemonSetBuilder("amdgpu-driver").build(client.cluster)
        driver = driver_pod_for(ds, "n1").build(client.cluster)
        manager.schedule_nodes_drain(
            DrainConfiguration(spec=DrainSpec(enable=True), nodes=[node])
        )
        manager.wait_idle()
        assert client.get("v1", "Pod", driver["metadata"]["name"], driver["metadata"]["namespace"])
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_dedup_guard_prevents_double_drain(self, client, provider, manager):
        node = drain_node_fixture(client)
        manager._draining_nodes.add("n1")  # simulate in-flight drain
        manager.schedule_nodes_drain(
            DrainConfiguration(spec=DrainSpec(enable=True), nodes=[node])
        )
        manager.wait_idle()
        # still in drain-required: the new request was deduplicated
        assert state_of(client, "n1") == consts.UPGRADE_STATE_DRAIN_REQUIRED
