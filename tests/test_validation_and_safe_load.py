"""ValidationManager tests (reference validation_manager_test.go:45-171) and
SafeDriverLoadManager tests (safe_driver_load_manager_test.go:44-70)."""

import time

import pytest

from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.node_state_provider import NodeUpgradeStateProvider
from k8s_operator_libs_amd.upgrade.safe_driver_load_manager import SafeDriverLoadManager
from k8s_operator_libs_amd.upgrade.validation_manager import ValidationManager

from builders import NodeBuilder, PodBuilder

SELECTOR = "app=amd-gpu-validator"


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
    return ValidationManager(client, provider, pod_selector=SELECTOR)


class TestValidation:
    def test_empty_selector_short_circuits(self, client, provider):
        mgr = ValidationManager(client, provider, pod_selector="")
        node = NodeBuilder("n1").build(client.cluster)
        assert mgr.validate(node) is True

    def test_no_validation_pods_not_done(self, client, manager):
        node = NodeBuilder("n1").build(client.cluster)
        assert manager.validate(node) is False

    def test_ready_pod_validates(self, client, manager):
        node = NodeBuilder("n1").build(client.cluster)
        PodBuilder("val", node="n1").with_labels(
            {"app": "amd-gpu-validator"}
        ).build(client.cluster)
        assert manager.validate(node) is True

    def test_not_ready_pod_stamps_timeout(self, client, manager):
        node = NodeBuilder("n1").build(client.cluster)
        PodBuilder("val", node="n1").with_labels(
            {"app": "amd-gpu-validator"}
        ).not_ready().build(client.cluster)
        assert manager.validate(node) is False
        key = util.get_validation_start_time_annotation_key()
        assert key in client.get_node("n1")["metadata"]["annotations"]

    def test_not_running_pod_not_ready(self, client, manager):
        node = NodeBuilder("n1").build(client.cluster)
        PodBuilder("val", node="n1").with_labels(
            {"app": "amd-gpu-validator"}
        ).with_phase("Pending").build(client.cluster)
        assert manager.validate(node) is False

    def test_timeout_fails_node_and_cleans_annotation(self, client, provider, manager):
        node = NodeBuilder("n1").with_upgrade_state(
            consts.UPGRADE_STATE_VALIDATION_REQUIRED
        ).build(client.cluster)
        PodBuilder("val", node="n1").with_labels(
            {"app": "amd-gpu-validator"}
        ).not_ready().build(client.cluster)
        key = util.get_validation_start_time_annotation_key()
        provider.change_node_upgrade_annotation(node, key, str(int(time.time()) - 601))
        assert manager.validate(node) is False
        assert state_of(client, "n1") == consts.UPGRADE_STATE_FAILED
        assert key not in client.get_node("n1")["metadata"]["annotations"]

    def test_success_clears_stale_annotation(self, client, provider, manager):
        node = NodeBuilder("n1").build(client.cluster)
        key = util.get_validation_start_time_annotation_key()
        provider.change_node_upgrade_annotation(node, key, str(int(time.time())))
        PodBuilder("val", node="n1").with_labels(
            {"app": "amd-gpu-validator"}
        ).build(client.cluster)
        assert manager.validate(node) is True
        assert key not in client.get_node("n1")["metadata"]["annotations"]


class TestSafeDriverLoad:
    def test_detect_and_unblock(self, client, provider):
        mgr = SafeDriverLoadManager(provider)
        key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        node = NodeBuilder("n1").with_annotation(key, "true").build(client.cluster)
        assert mgr.is_waiting_for_safe_driver_load(node) is True
        mgr.unblock_loading(node)
        assert key not in client.get_node("n1")["metadata"]["annotations"]
        assert mgr.is_waiting_for_safe_driver_load(node) is False
        mgr.unblock_loading(node)  # idempotent

    def test_not_waiting(self, client, provider):
        mgr = SafeDriverLoadManager(provider)
        node = NodeBuilder("n1").build(client.cluster)
        assert mgr.is_waiting_for_safe_driver_load(node) is False
