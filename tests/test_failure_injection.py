"""Failure injection and recovery (BASELINE config #5).

The reference models failure as state-machine states, not a subsystem
(SURVEY.md §5): drain/cordon errors -> upgrade-failed, eviction shortfall ->
drain-or-failed, >10 driver restarts -> failed, auto-recovery once the pod is
back in sync.  These tests inject the failures the reference only reaches
via unit state setup — node reboot mid-drain, crash-looping driver,
conflict storms, operator restart mid-flight — and assert recovery."""

import threading


from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import DRIVER_LABELS, DRIVER_NS, PodBuilder
from simenv import SimDaemonSetController
from test_state_manager import policy, setup_cluster, state_of


def run_until(client, manager, pol, target_state, node="node-0", max_ticks=15):
    for _ in range(max_ticks):
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        manager.wait_idle()
        if state_of(client, node) == target_state:
            return True
    return False


class TestNodeRebootMidDrain:
    def test_node_goes_not_ready_mid_upgrade_then_recovers(self, client):
        """Node 'reboots' (NotReady) while in drain-required; when it comes
        back the upgrade continues to done."""
        manager = ClusterUpgradeStateManager(client)
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new",
                              node_states=consts.UPGRADE_STATE_DRAIN_REQUIRED)
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        PodBuilder("w", node="node-0").with_owner_reference("ReplicaSet", "rs").build(client.cluster)
        # inject: node reboots (NotReady) mid-drain
        client.patch("v1", "Node", "node-0",
                     {"status": {"conditions": [{"type": "Ready", "status": "False"}]}})
        pol = policy(drainSpec={"enable": True})
        # drain still proceeds (API objects survive a reboot) and the machine
        # keeps walking the node forward once the driver pod restarts
        assert run_until(client, manager, pol, consts.UPGRADE_STATE_DONE)
        # node back: mark Ready again (kubelet rejoin) - state stays done
        client.patch("v1", "Node", "node-0",
                     {"status": {"conditions": [{"type": "Ready", "status": "True"}]}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE

    def test_blocked_drain_fails_then_recovers_when_pod_synced(self, client):
        """Drain blocked by an unmanaged pod -> upgrade-failed; after the
        driver pod lands in sync (e.g. manual fix + pod restart), the failed
        node auto-recovers through uncordon to done."""
        manager = ClusterUpgradeStateManager(client)
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new",
                              node_states=consts.UPGRADE_STATE_DRAIN_REQUIRED)
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        PodBuilder("bare", node="node-0").build(client.cluster)  # blocks drain
        pol = policy(drainSpec={"enable": True, "force": False})
        assert run_until(client, manager, pol, consts.UPGRADE_STATE_FAILED)
        # manual remediation: admin deletes the bare pod and restarts driver
        client.delete("v1", "Pod", "bare", "default")
        driver = client.list_pods(namespace=DRIVER_NS)[0]
        client.delete_pod(driver["metadata"]["name"], DRIVER_NS)  # sim recreates in sync
        assert run_until(client, manager, pol, consts.UPGRADE_STATE_DONE)


class TestCrashLoopingDriver:
    def test_crash_loop_fails_then_recovery(self, client):
        manager = ClusterUpgradeStateManager(client)
        ds, _ = setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
                              pod_ready=False)
        pod = client.list_pods(namespace=DRIVER_NS)[0]
        client.patch("v1", "Pod", pod["metadata"]["name"],
                     {"status": {"containerStatuses": [
                         {"name": "driver", "ready": False, "restartCount": 30}]}},
                     DRIVER_NS)
        pol = policy()
        assert run_until(client, manager, pol, consts.UPGRADE_STATE_FAILED)
        # new driver image fixes the crash loop: pod becomes ready & in sync
        client.patch("v1", "Pod", pod["metadata"]["name"],
                     {"status": {"containerStatuses": [
                         {"name": "driver", "ready": True, "restartCount": 30}]}},
                     DRIVER_NS)
        assert run_until(client, manager, pol, consts.UPGRADE_STATE_DONE)


class TestOperatorRestartMidFlight:
    def test_new_manager_resumes_from_labels(self, client):
        """Operator crash/restart = build a fresh manager; all state lives in
        node labels so the upgrade resumes exactly where it stopped
        (upgrade_state.go:46-52 design guarantee)."""
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%",
                     drainSpec={"enable": True})
        seen_states = []
        # drive each tick with a brand-new manager instance
        for _ in range(14):
            manager = ClusterUpgradeStateManager(client).with_pod_deletion_enabled(
                gpu_pod_deletion_filter
            )
            state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
            manager.apply_state(state, pol)
            manager.wait_idle()
            seen_states.append(state_of(client, "node-0"))
            if seen_states[-1] == consts.UPGRADE_STATE_DONE:
                break
        assert seen_states[-1] == consts.UPGRADE_STATE_DONE
        # monotonic forward progress, no state revisited after leaving it
        dedup = [s for i, s in enumerate(seen_states) if i == 0 or s != seen_states[i - 1]]
        assert len(dedup) == len(set(dedup)), f"state revisited: {seen_states}"


class TestConcurrentTicks:
    def test_two_managers_ticking_concurrently_converge(self, client):
        """Two operator replicas reconciling the same cluster (split-brain
        during leader-election churn) must not corrupt the state machine."""
        ds, _ = setup_cluster(client, n_nodes=4, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        pol = policy(maxParallelUpgrades=2, maxUnavailable="100%")
        managers = [ClusterUpgradeStateManager(client) for _ in range(2)]
        errors = []

        def tick(mgr):
            try:
                state = mgr.build_state(DRIVER_NS, DRIVER_LABELS)
                mgr.apply_state(state, pol)
                mgr.wait_idle()
            except Exception as exc:  # build races are tolerated, corruption is not
                errors.append(exc)

        for _ in range(20):
            threads = [threading.Thread(target=tick, args=(m,)) for m in managers]
            for t in threads:
                t.start()
            for t in threads:
                t.join()
            states = {state_of(client, f"node-{i}") for i in range(4)}
            if states == {consts.UPGRADE_STATE_DONE}:
                break
        assert {state_of(client, f"node-{i}") for i in range(4)} == {
            consts.UPGRADE_STATE_DONE
        }
        for exc in errors:
            assert "should not have unscheduled pods" in str(exc), exc


class TestAnicNicDriverFlow:
    """OFED-analogue (BASELINE config #4): xGMI/IF NIC driver upgrade using
    driver name 'anic', wait-for-jobs + validation hooks, safe-load
    handshake (docs/automatic-ofed-upgrade.md retargeted)."""

    def test_full_anic_upgrade_with_safe_load_and_validation(self, client):
        util.set_driver_name("anic")
        manager = ClusterUpgradeStateManager(client).with_validation_enabled(
            "app=anic-validator"
        )
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new",
                              ds_name="anic-driver")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        # the new driver pod's init container requests safe load
        safe_key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        assert safe_key == "amd.com/anic-driver-upgrade.driver-wait-for-safe-load"
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {safe_key: "true"}}})
        # rdma job that must complete before deletion
        PodBuilder("rdma-job", node="node-0").with_labels({"app": "rdma-job"}).with_phase(
            "Succeeded"
        ).build(client.cluster)
        # validation pod, initially not ready
        PodBuilder("anic-val", node="node-0").with_labels(
            {"app": "anic-validator"}
        ).not_ready().build(client.cluster)
        pol = policy(waitForCompletion={"podSelector": "app=rdma-job"})
        # walk to validation-required
        assert run_until(client, manager, pol,
                         consts.UPGRADE_STATE_VALIDATION_REQUIRED)
        # safe load must have been unblocked by now (pod-restart phase)
        assert safe_key not in client.get_node("node-0")["metadata"]["annotations"]
        # validator comes up healthy (e.g. xGMI link check passes)
        client.patch("v1", "Pod", "anic-val",
                     {"status": {"containerStatuses": [
                         {"name": "main", "ready": True, "restartCount": 0}]}},
                     "default")
        assert run_until(client, manager, pol, consts.UPGRADE_STATE_DONE)


class TestApiserverOutage:
    """The apiserver itself restarts mid-upgrade: the informer stack must
    reconnect (re-watch from last RV / relist), the keep-alive fast
    transport must re-establish connections, and the upgrade must complete
    without double-fired transitions.  The reference inherits this from
    client-go; here it is pinned wire-level."""

    def test_upgrade_survives_apiserver_restart(self):
        import socket
        import time

        from k8s_operator_libs_amd.core.apiserver import start_apiserver
        from k8s_operator_libs_amd.core.cache import CachedClient
        from k8s_operator_libs_amd.core.fakecluster import FakeCluster
        from k8s_operator_libs_amd.core.restclient import RestClient
        from k8s_operator_libs_amd.metrics import MetricsRegistry
        from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
        from k8s_operator_libs_amd.upgrade.state_manager import (
            BuildStateError,
            ClusterUpgradeStateManager,
        )
        from k8s_operator_libs_amd.core.errors import ApiError, NotFoundError

        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]

        cluster = FakeCluster()  # state survives the server process
        handle = start_apiserver(port=port, cluster=cluster)
        rest = RestClient(handle.url)
        cached = CachedClient(rest)

        class W:
            pass

        W.cluster = cluster
        ds, _ = setup_cluster(W, n_nodes=2, pod_hash="old", ds_hash="new")
        SimDaemonSetController(cluster, ds, current_hash="new")

        registry = MetricsRegistry()
        manager = ClusterUpgradeStateManager(cached, metrics=registry) \
            .with_pod_deletion_enabled(gpu_pod_deletion_filter)
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%",
                     drainSpec={"enable": True})

        restarted = False
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            try:
                manager.reconcile(DRIVER_NS, DRIVER_LABELS, pol)
            except (BuildStateError, NotFoundError, ApiError):
                time.sleep(0.05)
                continue
            states = {
                n["metadata"]["labels"].get(util.get_upgrade_state_label_key())
                for n in cluster.list("v1", "Node")
            }
            if not restarted and states & {"pod-restart-required",
                                           "drain-required",
                                           "pod-deletion-required"}:
                # outage in the middle of the pipeline
                handle.stop()
                time.sleep(0.2)
                handle = start_apiserver(port=port, cluster=cluster)
                restarted = True
            if states == {consts.UPGRADE_STATE_DONE}:
                break
            time.sleep(0.05)

        try:
            assert restarted, "restart point never reached"
            node_states = [
                n["metadata"]["labels"].get(util.get_upgrade_state_label_key())
                for n in cluster.list("v1", "Node")
            ]
            assert node_states == [consts.UPGRADE_STATE_DONE] * 2, node_states
            # no transition double-fired despite the outage
            for (frm, to), count in registry.state_transitions.items().items():
                assert count <= 2, f"{frm}->{to} fired {count} times"
        finally:
            manager.wait_idle()
            cached.stop()
            rest.close()
            handle.stop()
