"""State-machine integration tests
(reference pkg/upgrade/upgrade_state_test.go — the 1865-line suite).

Covers build_state happy/edge paths, every apply_state transition, the
maxParallelUpgrades/maxUnavailable window math, pod-deletion/drain
enable/disable, safe-load, failure & recovery, uncordon semantics, and full
single-node end-to-end with a simulated DaemonSet controller."""

import pytest

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import (
        DriverUpgradePolicySpec,
)
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
from k8s_operator_libs_amd.upgrade.state_manager import (
    BuildStateError,
    ClusterUpgradeStateManager,
)

from builders import (
    DRIVER_LABELS,
    DRIVER_NS,
    DaemonSetBuilder,
    NodeBuilder,
    PodBuilder,
    driver_pod_for,
    make_controller_revision,
)
from simenv import SimDaemonSetController


def state_of(client, node_name):
    return (
        client.get_node(node_name)["metadata"]["labels"]
        .get(util.get_upgrade_state_label_key(), "")
    )


def policy(**kw):
    kw.setdefault("autoUpgrade", True)
    return DriverUpgradePolicySpec.model_validate(kw)


@pytest.fixture
def manager(client):
    return ClusterUpgradeStateManager(client)


def setup_cluster(client, n_nodes=1, pod_hash="rev1", ds_hash="rev1",
                  node_states=None, ds_name="amdgpu-driver", pod_ready=True):
    """Create a driver DaemonSet with one driver pod per node."""
    ds = (
        DaemonSetBuilder(ds_name)
        .with_desired_number_scheduled(n_nodes)
        .build(client.cluster)
    )
    make_controller_revision(ds, ds_hash, revision=2, cluster=client.cluster)
    if ds_hash != pod_hash:
        make_controller_revision(ds, pod_hash, revision=1, cluster=client.cluster)
    nodes = []
    for i in range(n_nodes):
        name = f"node-{i}"
        b = NodeBuilder(name)
        if node_states:
            b.with_upgrade_state(node_states[i] if isinstance(node_states, list) else node_states)
        nodes.append(b.build(client.cluster))
        driver_pod_for(ds, name, hash_=pod_hash, ready=pod_ready).build(client.cluster)
    return ds, nodes


class TestBuildState:
    def test_empty_cluster(self, client, manager):
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        assert state.node_states == {}

    def test_groups_by_state_label(self, client, manager):
        setup_cluster(client, n_nodes=3,
                      node_states=[consts.UPGRADE_STATE_DONE,
                                   consts.UPGRADE_STATE_DONE,
                                   consts.UPGRADE_STATE_FAILED])
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        assert len(state.nodes_in(consts.UPGRADE_STATE_DONE)) == 2
        assert len(state.nodes_in(consts.UPGRADE_STATE_FAILED)) == 1

    def test_unscheduled_ds_pods_fail_build(self, client, manager):
        ds = DaemonSetBuilder("amdgpu-driver").with_desired_number_scheduled(3).build(client.cluster)
        driver_pod_for(ds, "node-0").build(client.cluster)  # only 1 of 3
        NodeBuilder("node-0").build(client.cluster)
        with pytest.raises(BuildStateError):
            manager.build_state(DRIVER_NS, DRIVER_LABELS)

    def test_orphaned_pods_collected(self, client, manager):
        NodeBuilder("node-0").build(client.cluster)
        # driver-labeled pod with no owner: orphan
        PodBuilder("orphan", node="node-0", namespace=DRIVER_NS).with_labels(
            DRIVER_LABELS
        ).build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        assert len(state.nodes_in("")) == 1
        assert state.nodes_in("")[0].is_orphaned_pod()

    def test_pending_unscheduled_orphan_skipped(self, client, manager):
        PodBuilder("floating", node="", namespace=DRIVER_NS).with_labels(
            DRIVER_LABELS
        ).with_phase("Pending").build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        assert state.node_states == {}


class TestApplyStateGuards:
    def test_none_state_raises(self, manager):
        with pytest.raises(ValueError):
            manager.apply_state(None, policy())

    def test_auto_upgrade_disabled_noop(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, DriverUpgradePolicySpec(autoUpgrade=False))
        assert state_of(client, "node-0") == ""

    def test_none_policy_noop(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, None)
        assert state_of(client, "node-0") == ""


class TestDoneOrUnknown:
    def test_out_of_date_pod_requires_upgrade(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=0, maxUnavailable="100%"))
        # phases iterate the snapshot grouping: exactly one transition per tick
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UPGRADE_REQUIRED
        # next tick consumes upgrade-required -> cordon-required pipeline
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=0, maxUnavailable="100%"))
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_CORDON_REQUIRED

    def test_in_sync_unknown_becomes_done(self, client, manager):
        setup_cluster(client)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE

    def test_in_sync_done_stays_done(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_DONE)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE

    def test_safe_load_waiting_node_requires_upgrade(self, client, manager):
        setup_cluster(client)
        key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        # maxParallel=1 but let it start: it should leave done/unknown
        manager.apply_state(state, policy(maxUnavailable="100%"))
        assert state_of(client, "node-0") != consts.UPGRADE_STATE_DONE

    def test_upgrade_requested_annotation_triggers(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_DONE)
        key = util.get_upgrade_requested_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        pol = policy(maxParallelUpgrades=0, maxUnavailable="100%")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UPGRADE_REQUIRED
        # next tick consumes the annotation in the upgrade-required processor
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        assert key not in client.get_node("node-0")["metadata"]["annotations"]

    def test_initially_unschedulable_annotated(self, client, manager):
        ds, nodes = setup_cluster(client, pod_hash="old", ds_hash="new")
        client.patch("v1", "Node", "node-0", {"spec": {"unschedulable": True}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        key = util.get_upgrade_initial_state_annotation_key()
        assert client.get_node("node-0")["metadata"]["annotations"][key] == "true"


class TestRollingWindow:
    def _mk_upgrade_required(self, client, n):
        return setup_cluster(
            client, n_nodes=n, pod_hash="old", ds_hash="new",
            node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED,
        )

    def test_max_parallel_zero_unlimited(self, client, manager):
        self._mk_upgrade_required(client, 5)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=0, maxUnavailable="100%"))
        manager.wait_idle()
        for i in range(5):
            assert state_of(client, f"node-{i}") != consts.UPGRADE_STATE_UPGRADE_REQUIRED

    def test_max_parallel_limits(self, client, manager):
        self._mk_upgrade_required(client, 6)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=2, maxUnavailable="100%"))
        manager.wait_idle()
        started = sum(
            state_of(client, f"node-{i}") != consts.UPGRADE_STATE_UPGRADE_REQUIRED
            for i in range(6)
        )
        assert started == 2

    def test_in_progress_counts_against_limit(self, client, manager):
        ds, _ = setup_cluster(
            client, n_nodes=4, pod_hash="old", ds_hash="new",
            node_states=[
                consts.UPGRADE_STATE_UPGRADE_REQUIRED,
                consts.UPGRADE_STATE_UPGRADE_REQUIRED,
                consts.UPGRADE_STATE_POD_RESTART_REQUIRED,  # in progress
                consts.UPGRADE_STATE_POD_RESTART_REQUIRED,  # in progress
            ],
        )
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        avail = manager.common.get_upgrades_available(state, 3, 100)
        assert avail == 1  # 3 - 2 in progress

    def test_max_unavailable_percent_clamp(self, client, manager):
        # 8 nodes, 25% -> 2 slots even with maxParallel=8
        self._mk_upgrade_required(client, 8)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=8, maxUnavailable="25%"))
        manager.wait_idle()
        started = sum(
            state_of(client, f"node-{i}") != consts.UPGRADE_STATE_UPGRADE_REQUIRED
            for i in range(8)
        )
        assert started == 2

    def test_precordoned_nodes_consume_unavailability_budget(self, client, manager):
        ds, nodes = self._mk_upgrade_required(client, 8)
        # 2 nodes already cordoned (unavailable) -> 25% budget (2) exhausted
        client.patch("v1", "Node", "node-6", {"spec": {"unschedulable": True}})
        client.patch("v1", "Node", "node-7", {"spec": {"unschedulable": True}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=8, maxUnavailable="25%"))
        manager.wait_idle()
        # budget was consumed by cordoned nodes, but cordoned nodes themselves
        # bypass the limit (they're already unavailable)
        for i in range(6):
            assert state_of(client, f"node-{i}") == consts.UPGRADE_STATE_UPGRADE_REQUIRED
        for i in (6, 7):
            assert state_of(client, f"node-{i}") != consts.UPGRADE_STATE_UPGRADE_REQUIRED

    def test_not_ready_nodes_count_unavailable(self, client, manager):
        ds, _ = self._mk_upgrade_required(client, 4)
        client.patch("v1", "Node", "node-3",
                     {"status": {"conditions": [{"type": "Ready", "status": "False"}]}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        assert manager.common.get_current_unavailable_nodes(state) == 1
        avail = manager.common.get_upgrades_available(state, 4, 2)
        assert avail == 1  # maxUnavailable 2 - 1 already unavailable

    def test_skip_label_prevents_upgrade(self, client, manager):
        ds, nodes = self._mk_upgrade_required(client, 2)
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"labels": {util.get_upgrade_skip_node_label_key(): "true"}}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=0, maxUnavailable="100%"))
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UPGRADE_REQUIRED
        assert state_of(client, "node-1") != consts.UPGRADE_STATE_UPGRADE_REQUIRED


class TestDrainAndDeletionPhases:
    def test_drain_disabled_goes_to_pod_restart(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_DRAIN_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_drain_enabled_drains(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_DRAIN_REQUIRED)
        PodBuilder("w", node="node-0").with_owner_reference("ReplicaSet", "rs").build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(drainSpec={"enable": True}))
        manager.wait_idle()
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED
        with pytest.raises(Exception):
            client.get("v1", "Pod", "w", "default")

    def test_pod_deletion_disabled_skips_to_drain(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_POD_DELETION_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # pod deletion disabled (no filter configured) -> drain-required
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DRAIN_REQUIRED
        # next tick, drain disabled -> pod-restart-required
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_pod_deletion_enabled_evicts_gpu_pods(self, client):
        manager = ClusterUpgradeStateManager(client).with_pod_deletion_enabled(
            gpu_pod_deletion_filter
        )
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_POD_DELETION_REQUIRED)
        PodBuilder("gpu", node="node-0").with_owner_reference(
            "ReplicaSet", "rs"
        ).with_resource("amd.com/gpu").build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(podDeletion={}))
        manager.wait_idle()
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED
        with pytest.raises(Exception):
            client.get("v1", "Pod", "gpu", "default")

    def test_wait_for_jobs_without_selector_falls_through(self, client, manager):
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_WAIT_FOR_JOBS_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # deletion disabled -> straight to drain-required (one hop per tick)
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DRAIN_REQUIRED


class TestPodRestartPhase:
    def test_out_of_sync_pod_restarted(self, client, manager):
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new",
                              node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED)
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # old pod deleted, sim recreated with new hash
        pods = client.list_pods(namespace=DRIVER_NS)
        assert len(pods) == 1
        assert pods[0]["metadata"]["labels"]["controller-revision-hash"] == "new"
        # next tick: pod in sync & ready -> uncordon-required
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UNCORDON_REQUIRED

    def test_in_sync_ready_with_validation_enabled(self, client):
        manager = ClusterUpgradeStateManager(client).with_validation_enabled(
            "app=amd-gpu-validator"
        )
        setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_VALIDATION_REQUIRED

    def test_failing_driver_pod_fails_upgrade(self, client, manager):
        ds, _ = setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
                              pod_ready=False)
        # in-sync but not ready with >10 restarts
        pod = client.list_pods(namespace=DRIVER_NS)[0]
        client.patch("v1", "Pod", pod["metadata"]["name"],
                     {"status": {"containerStatuses": [
                         {"name": "driver", "ready": False, "restartCount": 11}]}},
                     DRIVER_NS)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_FAILED

    def test_not_failing_not_ready_waits(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
                      pod_ready=False)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_safe_load_unblocked_in_pod_restart(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED)
        key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert key not in client.get_node("node-0")["metadata"]["annotations"]


class TestFailedRecoveryAndUncordon:
    def test_failed_node_recovers_when_pod_back_in_sync(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_FAILED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # pod is in sync & ready -> uncordon-required, then uncordon runs in
        # the same tick? no: uncordon processes the snapshot's uncordon list.
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UNCORDON_REQUIRED

    def test_failed_initially_unschedulable_goes_done(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_FAILED)
        key = util.get_upgrade_initial_state_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}},
                      "spec": {"unschedulable": True}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        assert key not in client.get_node("node-0")["metadata"]["annotations"]
        # still cordoned: we never uncordon a node that started unschedulable
        assert client.get_node("node-0")["spec"].get("unschedulable") is True

    def test_uncordon_completes(self, client, manager):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_UNCORDON_REQUIRED)
        client.patch("v1", "Node", "node-0", {"spec": {"unschedulable": True}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        assert not client.get_node("node-0")["spec"].get("unschedulable")


class TestValidationPhase:
    def test_validation_passes_to_uncordon(self, client):
        manager = ClusterUpgradeStateManager(client).with_validation_enabled(
            "app=amd-gpu-validator"
        )
        setup_cluster(client, node_states=consts.UPGRADE_STATE_VALIDATION_REQUIRED)
        PodBuilder("val", node="node-0").with_labels(
            {"app": "amd-gpu-validator"}
        ).build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UNCORDON_REQUIRED

    def test_validation_not_ready_holds(self, client):
        manager = ClusterUpgradeStateManager(client).with_validation_enabled(
            "app=amd-gpu-validator"
        )
        setup_cluster(client, node_states=consts.UPGRADE_STATE_VALIDATION_REQUIRED)
        PodBuilder("val", node="node-0").with_labels(
            {"app": "amd-gpu-validator"}
        ).not_ready().build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_VALIDATION_REQUIRED


class TestEndToEnd:
    def test_single_node_full_lifecycle(self, client):
        """BASELINE config #2: upgrade-required -> ... -> upgrade-done with a
        dummy amdgpu driver DaemonSet bump."""
        manager = ClusterUpgradeStateManager(client).with_pod_deletion_enabled(
            gpu_pod_deletion_filter
        )
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        PodBuilder("gpu-workload", node="node-0").with_owner_reference(
            "ReplicaSet", "rs"
        ).with_resource("amd.com/gpu").build(client.cluster)
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%",
                     podDeletion={}, drainSpec={"enable": True})
        seen = set()
        for _ in range(12):
            state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
            manager.apply_state(state, pol)
            manager.wait_idle()
            seen.add(state_of(client, "node-0"))
            if state_of(client, "node-0") == consts.UPGRADE_STATE_DONE:
                break
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        # workload was evicted; driver pod is the new revision; node schedulable
        with pytest.raises(Exception):
            client.get("v1", "Pod", "gpu-workload", "default")
        pods = client.list_pods(namespace=DRIVER_NS)
        assert pods[0]["metadata"]["labels"]["controller-revision-hash"] == "new"
        assert not client.get_node("node-0")["spec"].get("unschedulable")
        # the node passed through the expected pipeline states
        assert consts.UPGRADE_STATE_UNCORDON_REQUIRED in seen

    def test_idempotent_reentry(self, client):
        """Running the same tick twice must not double-fire transitions."""
        manager = ClusterUpgradeStateManager(client)
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        s1 = state_of(client, "node-0")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        manager.apply_state(state, pol)  # re-apply same snapshot
        assert state_of(client, "node-0") != consts.UPGRADE_STATE_UPGRADE_REQUIRED
        assert s1 != ""


class TestCounts:
    def test_counts_exported(self, client, manager):
        setup_cluster(
            client, n_nodes=4, pod_hash="old", ds_hash="new",
            node_states=[
                consts.UPGRADE_STATE_UPGRADE_REQUIRED,
                consts.UPGRADE_STATE_DRAIN_REQUIRED,
                consts.UPGRADE_STATE_DONE,
                consts.UPGRADE_STATE_FAILED,
            ],
        )
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        counts = manager.counts(state)
        assert counts == {"total": 4, "in_progress": 2, "done": 1,
                          "failed": 1, "pending": 1}


class TestLiveReconcile:
    def test_live_mode_pipelines_multiple_transitions_per_pass(self, client):
        """Beyond-parity: with converge=True (live regrouping) a full
        single-node upgrade completes in a handful of reconcile calls, with
        every label transition still firing exactly once."""
        from k8s_operator_libs_amd.metrics import MetricsRegistry

        reg = MetricsRegistry()
        manager = ClusterUpgradeStateManager(client, metrics=reg)
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%")
        calls = 0
        for _ in range(4):
            manager.reconcile(DRIVER_NS, DRIVER_LABELS, pol, converge=True)
            calls += 1
            if state_of(client, "node-0") == consts.UPGRADE_STATE_DONE:
                break
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        assert calls <= 4
        for (frm, to), count in reg.state_transitions.items().items():
            assert count == 1, f"{frm}->{to} fired {count} times"

    def test_live_and_reference_modes_reach_same_final_state(self, client):
        ds, _ = setup_cluster(client, n_nodes=3, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        manager = ClusterUpgradeStateManager(client)
        pol = policy(maxParallelUpgrades=2, maxUnavailable="100%",
                     drainSpec={"enable": True})
        for _ in range(30):
            manager.reconcile(DRIVER_NS, DRIVER_LABELS, pol, converge=True)
            if all(state_of(client, f"node-{i}") == consts.UPGRADE_STATE_DONE
                   for i in range(3)):
                break
        for i in range(3):
            assert state_of(client, f"node-{i}") == consts.UPGRADE_STATE_DONE
            assert not client.get_node(f"node-{i}")["spec"].get("unschedulable")
