"""Edge semantics pinned by the reference's test suite
(upgrade_state_test.go:1180-1266 orphans, 384-613 window math details,
common_manager.go:250-264/673-708 initially-unschedulable handling)."""

import pytest

from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import DRIVER_LABELS, DRIVER_NS, NodeBuilder, PodBuilder
from simenv import SimDaemonSetController
from test_state_manager import policy, setup_cluster, state_of


class TestOrphanedPods:
    def _orphan_cluster(self, client, node_state=None):
        b = NodeBuilder("node-0")
        if node_state is not None:
            b.with_upgrade_state(node_state)
        b.build(client.cluster)
        return PodBuilder("orphan-driver", node="node-0", namespace=DRIVER_NS).with_labels(
            DRIVER_LABELS
        ).build(client.cluster)

    def test_orphan_without_request_stays_done(self, client):
        """An orphaned driver pod alone does NOT trigger an upgrade — only
        the upgrade-requested annotation does (orphans have no DS revision to
        compare)."""
        manager = ClusterUpgradeStateManager(client)
        self._orphan_cluster(client, consts.UPGRADE_STATE_DONE)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE

    def test_orphan_with_upgrade_requested_flows(self, client):
        manager = ClusterUpgradeStateManager(client)
        self._orphan_cluster(client, consts.UPGRADE_STATE_DONE)
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {
                         util.get_upgrade_requested_annotation_key(): "true"}}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxUnavailable="100%"))
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UPGRADE_REQUIRED

    def test_orphan_pod_restarted_in_pod_restart_phase(self, client):
        """Orphaned pods are never 'in sync', so pod-restart deletes them
        (common_manager.go:465-472)."""
        manager = ClusterUpgradeStateManager(client)
        self._orphan_cluster(client, consts.UPGRADE_STATE_POD_RESTART_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        with pytest.raises(Exception):
            client.get("v1", "Pod", "orphan-driver", DRIVER_NS)

    def test_terminating_orphan_not_deleted_again(self, client):
        manager = ClusterUpgradeStateManager(client)
        pod = self._orphan_cluster(client, consts.UPGRADE_STATE_POD_RESTART_REQUIRED)
        # mark terminating via a finalizer + delete
        client.patch("v1", "Pod", "orphan-driver",
                     {"metadata": {"finalizers": ["test/hold"]}}, DRIVER_NS)
        client.delete("v1", "Pod", "orphan-driver", DRIVER_NS)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())  # must not raise on re-delete
        live = client.get("v1", "Pod", "orphan-driver", DRIVER_NS)
        assert "deletionTimestamp" in live["metadata"]


class TestInitiallyUnschedulable:
    def test_validation_path_ends_done_still_cordoned(self, client):
        """A node that began the upgrade cordoned skips uncordon at the end
        of the validation path and keeps its cordon."""
        manager = ClusterUpgradeStateManager(client)
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        client.patch("v1", "Node", "node-0", {"spec": {"unschedulable": True}})
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%")
        for _ in range(12):
            manager.reconcile(DRIVER_NS, DRIVER_LABELS, pol, converge=True)
            if state_of(client, "node-0") == consts.UPGRADE_STATE_DONE:
                break
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        # never uncordoned, annotation consumed
        assert client.get_node("node-0")["spec"].get("unschedulable") is True
        key = util.get_upgrade_initial_state_annotation_key()
        assert key not in client.get_node("node-0")["metadata"]["annotations"]


class TestWindowMathDetails:
    def test_max_unavailable_int_form(self, client):
        manager = ClusterUpgradeStateManager(client)
        setup_cluster(client, n_nodes=6, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=0, maxUnavailable=3))
        manager.wait_idle()
        started = sum(
            state_of(client, f"node-{i}") != consts.UPGRADE_STATE_UPGRADE_REQUIRED
            for i in range(6)
        )
        assert started == 3

    def test_max_unavailable_none_means_total(self, client):
        manager = ClusterUpgradeStateManager(client)
        setup_cluster(client, n_nodes=4, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        pol = policy(maxParallelUpgrades=0)
        pol.max_unavailable = None  # no clamp: all nodes may start
        manager.apply_state(state, pol)
        manager.wait_idle()
        for i in range(4):
            assert state_of(client, f"node-{i}") != consts.UPGRADE_STATE_UPGRADE_REQUIRED

    def test_100_percent_of_small_cluster(self, client):
        manager = ClusterUpgradeStateManager(client)
        setup_cluster(client, n_nodes=1, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        avail = manager.common.get_upgrades_available(state, 0, 1)
        assert avail == 1


class TestMultiDriverDaemonSets:
    def test_two_daemonsets_one_node_single_get(self, client):
        """Two driver DaemonSets (e.g. amdgpu + anic) with pods on the same
        node: build_state dedups the node fetch and groups one entry per
        pod."""
        manager = ClusterUpgradeStateManager(client)
        ds1, _ = setup_cluster(client, ds_name="amdgpu-driver")
        from builders import DaemonSetBuilder, driver_pod_for, make_controller_revision

        labels2 = {"app": "amdgpu-driver-daemonset", "sub": "anic"}
        ds2 = DaemonSetBuilder("anic-driver", labels=labels2).with_desired_number_scheduled(
            1
        ).build(client.cluster)
        make_controller_revision(ds2, "rev1", cluster=client.cluster)
        driver_pod_for(ds2, "node-0", hash_="rev1").build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        entries = [ns for lst in state.node_states.values() for ns in lst]
        assert len(entries) == 2
        # both entries share the same node object (fetch dedup)
        assert entries[0].node is entries[1].node


class TestSafeLoadUnblockPoints:
    def test_validation_phase_unblocks_safe_load(self, client):
        """The driver may restart after reaching validation-required and
        block on safe load again; the validation phase must unblock it
        (common_manager.go:581-586)."""
        manager = ClusterUpgradeStateManager(client).with_validation_enabled(
            "app=amd-gpu-validator"
        )
        setup_cluster(client, node_states=consts.UPGRADE_STATE_VALIDATION_REQUIRED)
        key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        # no validator pod yet -> validation does not pass, but the safe-load
        # annotation must already be gone after the phase runs
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert key not in client.get_node("node-0")["metadata"]["annotations"]
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_VALIDATION_REQUIRED


class TestInitContainerCrashLoop:
    def test_failing_init_container_fails_upgrade(self, client):
        """isDriverPodFailing also inspects initContainerStatuses
        (common_manager.go:636-648)."""
        manager = ClusterUpgradeStateManager(client)
        ds, _ = setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
                              pod_ready=False)
        pod = client.list_pods(namespace=DRIVER_NS)[0]
        client.patch("v1", "Pod", pod["metadata"]["name"],
                     {"status": {
                         "containerStatuses": [{"name": "driver", "ready": False,
                                                "restartCount": 0}],
                         "initContainerStatuses": [{"name": "safe-load-gate",
                                                    "ready": False,
                                                    "restartCount": 11}]}},
                     DRIVER_NS)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_FAILED

    def test_ready_init_container_with_restarts_is_fine(self, client):
        manager = ClusterUpgradeStateManager(client)
        setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
                      pod_ready=False)
        pod = client.list_pods(namespace=DRIVER_NS)[0]
        client.patch("v1", "Pod", pod["metadata"]["name"],
                     {"status": {
                         "containerStatuses": [{"name": "driver", "ready": False,
                                                "restartCount": 2}],
                         "initContainerStatuses": [{"name": "g", "ready": True,
                                                    "restartCount": 50}]}},
                     DRIVER_NS)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # ready init container with many restarts doesn't count as failing
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED


class TestCorruptTimeoutAnnotations:
    def test_corrupt_completion_stamp_restamps(self, client):
        from k8s_operator_libs_amd.api.upgrade.v1alpha1 import WaitForCompletionSpec
        from k8s_operator_libs_amd.upgrade.node_state_provider import (
            NodeUpgradeStateProvider,
        )
        from k8s_operator_libs_amd.upgrade.pod_manager import (
            PodManager,
            PodManagerConfig,
        )

        node = NodeBuilder("n1").with_upgrade_state(
            consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED
        ).build(client.cluster)
        PodBuilder("job", node="n1").with_labels({"app": "job"}).build(client.cluster)
        key = util.get_wait_for_pod_completion_start_time_annotation_key()
        client.patch("v1", "Node", "n1",
                     {"metadata": {"annotations": {key: "garbage"}}})
        node = client.get_node("n1")
        mgr = PodManager(client, NodeUpgradeStateProvider(client))
        mgr.schedule_check_on_pod_completion(PodManagerConfig(
            nodes=[node],
            wait_for_completion_spec=WaitForCompletionSpec(
                podSelector="app=job", timeoutSecond=300),
        ))
        stamped = client.get_node("n1")["metadata"]["annotations"][key]
        assert stamped.isdigit(), "corrupt stamp must be replaced, not crash"

    def test_corrupt_validation_stamp_restamps(self, client):
        from k8s_operator_libs_amd.upgrade.node_state_provider import (
            NodeUpgradeStateProvider,
        )
        from k8s_operator_libs_amd.upgrade.validation_manager import ValidationManager

        node = NodeBuilder("n1").build(client.cluster)
        PodBuilder("val", node="n1").with_labels(
            {"app": "amd-gpu-validator"}
        ).not_ready().build(client.cluster)
        key = util.get_validation_start_time_annotation_key()
        client.patch("v1", "Node", "n1",
                     {"metadata": {"annotations": {key: "not-a-number"}}})
        node = client.get_node("n1")
        mgr = ValidationManager(client, NodeUpgradeStateProvider(client),
                                pod_selector="app=amd-gpu-validator")
        assert mgr.validate(node) is False
        stamped = client.get_node("n1")["metadata"]["annotations"][key]
        assert stamped.isdigit()
