"""PodManager behavioral tests (reference pkg/upgrade/pod_manager_test.go)."""

import time

import pytest

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import (
    PodDeletionSpec,
    WaitForCompletionSpec,
)
from k8s_operator_libs_amd.core.errors import NotFoundError
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
from k8s_operator_libs_amd.upgrade.node_state_provider import NodeUpgradeStateProvider
from k8s_operator_libs_amd.upgrade.pod_manager import PodManager, PodManagerConfig

from builders import (
    DaemonSetBuilder,
    NodeBuilder,
    PodBuilder,
    driver_pod_for,
    make_controller_revision,
)


def state_of(client, node_name):
    return (
        client.get_node(node_name)["metadata"]["labels"]
        .get(util.get_upgrade_state_label_key(), "")
    )


@pytest.fixture
def provider(client):
    return NodeUpgradeStateProvider(client)


def make_manager(client, provider, filter_=gpu_pod_deletion_filter):
    return PodManager(client, provider, pod_deletion_filter=filter_)


class TestRevisionHash:
    def test_pod_hash(self, client, provider):
        ds = DaemonSetBuilder("amdgpu-driver").build(client.cluster)
        pod = driver_pod_for(ds, "n1", hash_="abc123").build(client.cluster)
        mgr = make_manager(client, provider)
        assert mgr.get_pod_controller_revision_hash(pod) == "abc123"

    def test_pod_without_hash_raises(self, client, provider):
        pod = PodBuilder("p", node="n1").build(client.cluster)
        with pytest.raises(ValueError):
            make_manager(client, provider).get_pod_controller_revision_hash(pod)

    def test_daemonset_hash_takes_max_revision(self, client, provider):
        ds = DaemonSetBuilder("amdgpu-driver").build(client.cluster)
        make_controller_revision(ds, "oldhash", revision=1, cluster=client.cluster)
        make_controller_revision(ds, "newhash", revision=7, cluster=client.cluster)
        mgr = make_manager(client, provider)
        assert mgr.get_daemonset_controller_revision_hash(ds) == "newhash"

    def test_daemonset_without_revisions_raises(self, client, provider):
        from k8s_operator_libs_amd.upgrade.pod_manager import (
            StaleClusterViewError,
        )

        ds = DaemonSetBuilder("amdgpu-driver").build(client.cluster)
        with pytest.raises(StaleClusterViewError):
            make_manager(client, provider).get_daemonset_controller_revision_hash(ds)


class TestPodRestart:
    def test_restart_deletes_only_listed_pods(self, client, provider):
        p1 = PodBuilder("keep", node="n1").build(client.cluster)
        p2 = PodBuilder("restart-me", node="n1").build(client.cluster)
        make_manager(client, provider).schedule_pods_restart([p2])
        assert client.get("v1", "Pod", "keep", "default")
        with pytest.raises(NotFoundError):
            client.get("v1", "Pod", "restart-me", "default")

    def test_restart_empty_is_noop(self, client, provider):
        make_manager(client, provider).schedule_pods_restart([])


class TestEviction:
    def _node_in_deletion_state(self, client):
        return (
            NodeBuilder("n1")
            .with_upgrade_state(consts.UPGRADE_STATE_POD_DELETION_REQUIRED)
            .build(client.cluster)
        )

    def test_no_matching_pods_moves_to_pod_restart(self, client, provider):
        node = self._node_in_deletion_state(client)
        PodBuilder("cpu-pod", node="n1").with_owner_reference(
            "ReplicaSet", "rs"
        ).build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_pod_eviction(
            PodManagerConfig(nodes=[node], deletion_spec=PodDeletionSpec())
        )
        mgr.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED
        # the CPU pod survived
        assert client.get("v1", "Pod", "cpu-pod", "default")

    def test_gpu_pods_evicted(self, client, provider):
        node = self._node_in_deletion_state(client)
        PodBuilder("gpu-pod", node="n1").with_owner_reference(
            "ReplicaSet", "rs"
        ).with_resource("amd.com/gpu").build(client.cluster)
        PodBuilder("cpu-pod", node="n1").with_owner_reference(
            "ReplicaSet", "rs"
        ).build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_pod_eviction(
            PodManagerConfig(nodes=[node], deletion_spec=PodDeletionSpec())
        )
        mgr.wait_idle()
        with pytest.raises(NotFoundError):
            client.get("v1", "Pod", "gpu-pod", "default")
        assert client.get("v1", "Pod", "cpu-pod", "default")
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_unforceable_pod_fails_node_when_drain_disabled(self, client, provider):
        node = self._node_in_deletion_state(client)
        # bare pod (no controller) consuming a GPU; force=false blocks it
        PodBuilder("bare-gpu", node="n1").with_resource("amd.com/gpu").build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_pod_eviction(
            PodManagerConfig(
                nodes=[node], deletion_spec=PodDeletionSpec(force=False),
                drain_enabled=False,
            )
        )
        mgr.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_FAILED
        assert client.get("v1", "Pod", "bare-gpu", "default")

    def test_unforceable_pod_goes_to_drain_when_enabled(self, client, provider):
        node = self._node_in_deletion_state(client)
        PodBuilder("bare-gpu", node="n1").with_resource("amd.com/gpu").build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_pod_eviction(
            PodManagerConfig(
                nodes=[node], deletion_spec=PodDeletionSpec(force=False),
                drain_enabled=True,
            )
        )
        mgr.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_DRAIN_REQUIRED

    def test_force_allows_bare_pod(self, client, provider):
        node = self._node_in_deletion_state(client)
        PodBuilder("bare-gpu", node="n1").with_resource("amd.com/gpu").build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_pod_eviction(
            PodManagerConfig(nodes=[node], deletion_spec=PodDeletionSpec(force=True))
        )
        mgr.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED
        with pytest.raises(NotFoundError):
            client.get("v1", "Pod", "bare-gpu", "default")

    def test_emptydir_matrix(self, client, provider):
        node = self._node_in_deletion_state(client)
        PodBuilder("gpu-ed", node="n1").with_owner_reference(
            "ReplicaSet", "rs"
        ).with_resource("amd.com/gpu").with_emptydir().build(client.cluster)
        mgr = make_manager(client, provider)
        # deleteEmptyDir=False -> blocked -> failed
        mgr.schedule_pod_eviction(
            PodManagerConfig(nodes=[node],
                             deletion_spec=PodDeletionSpec(deleteEmptyDir=False))
        )
        mgr.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_FAILED
        # deleteEmptyDir=True -> evicted
        provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_POD_DELETION_REQUIRED)
        mgr.schedule_pod_eviction(
            PodManagerConfig(nodes=[node],
                             deletion_spec=PodDeletionSpec(deleteEmptyDir=True))
        )
        mgr.wait_idle()
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_missing_spec_raises(self, client, provider):
        node = self._node_in_deletion_state(client)
        with pytest.raises(ValueError):
            make_manager(client, provider).schedule_pod_eviction(
                PodManagerConfig(nodes=[node])
            )


class TestPodCompletionWait:
    def _node_waiting(self, client):
        return (
            NodeBuilder("n1")
            .with_upgrade_state(consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED)
            .build(client.cluster)
        )

    def test_no_running_pods_moves_on(self, client, provider):
        node = self._node_waiting(client)
        PodBuilder("job", node="n1").with_labels({"app": "job"}).with_phase(
            "Succeeded"
        ).build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_check_on_pod_completion(
            PodManagerConfig(
                nodes=[node],
                wait_for_completion_spec=WaitForCompletionSpec(podSelector="app=job"),
            )
        )
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_DELETION_REQUIRED

    def test_running_pod_keeps_state_without_timeout(self, client, provider):
        node = self._node_waiting(client)
        PodBuilder("job", node="n1").with_labels({"app": "job"}).build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_check_on_pod_completion(
            PodManagerConfig(
                nodes=[node],
                wait_for_completion_spec=WaitForCompletionSpec(podSelector="app=job"),
            )
        )
        assert state_of(client, "n1") == consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED
        # no timeout -> no start-time annotation stamped
        key = util.get_wait_for_pod_completion_start_time_annotation_key()
        assert key not in client.get_node("n1")["metadata"]["annotations"]

    def test_timeout_stamps_then_forces(self, client, provider):
        node = self._node_waiting(client)
        PodBuilder("job", node="n1").with_labels({"app": "job"}).build(client.cluster)
        mgr = make_manager(client, provider)
        spec = WaitForCompletionSpec(podSelector="app=job", timeoutSecond=300)
        cfg = PodManagerConfig(nodes=[node], wait_for_completion_spec=spec)
        mgr.schedule_check_on_pod_completion(cfg)
        key = util.get_wait_for_pod_completion_start_time_annotation_key()
        assert key in client.get_node("n1")["metadata"]["annotations"]
        assert state_of(client, "n1") == consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED
        # simulate expiry by back-dating the annotation
        provider.change_node_upgrade_annotation(node, key, str(int(time.time()) - 301))
        mgr.schedule_check_on_pod_completion(cfg)
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_DELETION_REQUIRED
        assert key not in client.get_node("n1")["metadata"]["annotations"]

    def test_selector_scopes_to_node(self, client, provider):
        node = self._node_waiting(client)
        # running pod on ANOTHER node must not hold n1 back
        PodBuilder("job-elsewhere", node="n2").with_labels({"app": "job"}).build(client.cluster)
        mgr = make_manager(client, provider)
        mgr.schedule_check_on_pod_completion(
            PodManagerConfig(
                nodes=[node],
                wait_for_completion_spec=WaitForCompletionSpec(podSelector="app=job"),
            )
        )
        assert state_of(client, "n1") == consts.UPGRADE_STATE_POD_DELETION_REQUIRED
