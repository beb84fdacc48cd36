"""Tests driving the state machine phase processors through the mock L3
managers — the reference's mockery-mock isolation pattern
(upgrade_state_test.go:54-71)."""

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import (
    DrainSpec,
        WaitForCompletionSpec,
)
from k8s_operator_libs_amd.core.client import FakeClient
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.common_manager import (
    ClusterUpgradeState,
    CommonUpgradeManager,
    NodeUpgradeState,
)
from k8s_operator_libs_amd.upgrade.mocks import (
    MockCordonManager,
    MockDrainManager,
    MockNodeUpgradeStateProvider,
    MockPodManager,
    MockSafeDriverLoadManager,
    MockValidationManager,
)

from builders import NodeBuilder


def make_common_with_mocks(**overrides):
    common = CommonUpgradeManager(FakeClient())
    provider = MockNodeUpgradeStateProvider()
    common.node_state_provider = provider
    common.cordon_manager = overrides.get("cordon", MockCordonManager())
    common.drain_manager = overrides.get("drain", MockDrainManager(provider))
    common.pod_manager = overrides.get("pod", MockPodManager(provider))
    common.validation_manager = overrides.get("validation", MockValidationManager())
    common.safe_driver_load_manager = overrides.get(
        "safe", MockSafeDriverLoadManager()
    )
    return common, provider


def node_state(name, state, provider):
    node = provider.register(NodeBuilder(name).with_upgrade_state(state).build())
    return NodeUpgradeState(node=node)


def test_cordon_phase_uses_mock_and_advances(client):
    common, provider = make_common_with_mocks()
    ns = node_state("n1", consts.UPGRADE_STATE_CORDON_REQUIRED, provider)
    state = ClusterUpgradeState()
    state.add(consts.UPGRADE_STATE_CORDON_REQUIRED, ns)
    common.process_cordon_required_nodes(state)
    assert common.cordon_manager.calls_to("cordon") == [("n1",)]
    key = util.get_upgrade_state_label_key()
    assert ns.node["metadata"]["labels"][key] == consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED


def test_drain_phase_delegates_to_mock(client):
    common, provider = make_common_with_mocks()
    ns = node_state("n1", consts.UPGRADE_STATE_DRAIN_REQUIRED, provider)
    state = ClusterUpgradeState()
    state.add(consts.UPGRADE_STATE_DRAIN_REQUIRED, ns)
    common.process_drain_nodes(state, DrainSpec(enable=True))
    assert common.drain_manager.calls_to("schedule_nodes_drain") == [(("n1",),)]
    key = util.get_upgrade_state_label_key()
    assert ns.node["metadata"]["labels"][key] == consts.UPGRADE_STATE_POD_RESTART_REQUIRED


def test_wait_for_jobs_delegates(client):
    common, provider = make_common_with_mocks()
    ns = node_state("n1", consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED, provider)
    state = ClusterUpgradeState()
    state.add(consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED, ns)
    common.process_wait_for_jobs_required_nodes(
        state, WaitForCompletionSpec(podSelector="app=job")
    )
    assert common.pod_manager.calls_to("schedule_check_on_pod_completion")


def test_provider_failure_propagates(client):
    common, provider = make_common_with_mocks()
    provider.failures["change_node_upgrade_state"] = RuntimeError("apiserver down")
    ns = node_state("n1", consts.UPGRADE_STATE_CORDON_REQUIRED, provider)
    state = ClusterUpgradeState()
    state.add(consts.UPGRADE_STATE_CORDON_REQUIRED, ns)
    try:
        common.process_cordon_required_nodes(state)
        raised = False
    except RuntimeError:
        raised = True
    assert raised


def test_safe_load_mock_tracks_unblock(client):
    safe = MockSafeDriverLoadManager({"n1"})
    node = NodeBuilder("n1").build()
    assert safe.is_waiting_for_safe_driver_load(node)
    safe.unblock_loading(node)
    assert not safe.is_waiting_for_safe_driver_load(node)
    assert safe.calls_to("unblock_loading") == [("n1",)]
