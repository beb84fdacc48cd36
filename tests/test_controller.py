"""Event-driven UpgradeController tests."""

import threading
import time

from k8s_operator_libs_amd.upgrade import consts
from k8s_operator_libs_amd.upgrade.controller import UpgradeController
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import DRIVER_LABELS, DRIVER_NS
from simenv import SimDaemonSetController
from test_state_manager import policy, setup_cluster, state_of
import examples.upgrade_status as status_cli


def test_controller_completes_upgrade_event_driven(client):
    """Watch events (not the resync timer) drive the upgrade to done: the
    resync interval is set far beyond the test timeout."""
    ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
    SimDaemonSetController(client.cluster, ds, current_hash="new")
    manager = ClusterUpgradeStateManager(client)
    controller = UpgradeController(
        manager, DRIVER_NS, DRIVER_LABELS,
        policy(maxParallelUpgrades=1, maxUnavailable="100%"),
        resync_seconds=60.0,
    )
    done = {}

    def run():
        done["ok"] = controller.run(until_all_done=True, max_reconciles=30)

    t = threading.Thread(target=run, daemon=True)
    t.start()
    t.join(timeout=20)
    controller.stop()
    assert done.get("ok") is True
    assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
    # event-driven: far fewer reconciles than a polling loop would need
    assert controller.reconcile_count <= 10


def test_controller_resync_fallback(client):
    """With no events at all, the resync timer still ticks reconciles."""
    setup_cluster(client)  # in-sync cluster: first reconcile -> done, no events after
    manager = ClusterUpgradeStateManager(client)
    controller = UpgradeController(
        manager, DRIVER_NS, DRIVER_LABELS, policy(), resync_seconds=0.05,
    )
    t = threading.Thread(
        target=lambda: controller.run(max_reconciles=3), daemon=True
    )
    t.start()
    t.join(timeout=10)
    controller.stop()
    assert controller.reconcile_count >= 3


def test_status_cli_renders(client):
    setup_cluster(client, n_nodes=2, node_states=[
        consts.UPGRADE_STATE_DONE, consts.UPGRADE_STATE_DRAIN_REQUIRED])
    client.patch("v1", "Node", "node-1", {"spec": {"unschedulable": True}})
    out = status_cli.render(client, "amdgpu")
    assert "node-0" in out and "upgrade-done" in out
    assert "drain-required" in out and "cordoned" in out
    assert "totals:" in out


def test_requestor_controller_wakes_on_nodemaintenance_ready(client):
    """In requestor mode the controller subscribes to NodeMaintenance
    condition changes (filtered by requestor-ID + condition predicates) and
    completes the upgrade without relying on the resync timer."""
    from k8s_operator_libs_amd.upgrade.requestor import RequestorOptions
    from k8s_operator_libs_amd.upgrade.state_manager import StateOptions
    from simenv import SimMaintenanceOperator

    ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
    SimDaemonSetController(client.cluster, ds, current_hash="new")
    SimMaintenanceOperator(client.cluster)
    manager = ClusterUpgradeStateManager(
        client,
        options=StateOptions(requestor=RequestorOptions(
            use_maintenance_operator=True,
            requestor_id="amd.gpu.operator",
            namespace="default",
        )),
    )
    controller = UpgradeController(
        manager, DRIVER_NS, DRIVER_LABELS, policy(drainSpec={"enable": True}),
        resync_seconds=120.0,  # far beyond the test window: events must drive it
    )
    done = {}
    t = threading.Thread(
        target=lambda: done.update(ok=controller.run(until_all_done=True,
                                                     max_reconciles=30)),
        daemon=True,
    )
    t.start()
    t.join(timeout=25)
    controller.stop()
    assert done.get("ok") is True
    assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE


class TestControllerLeaderElection:
    """UpgradeController under Lease-based election."""

    def test_only_leader_reconciles_and_failover(self):
        import threading
        import time

        from k8s_operator_libs_amd.core.client import FakeClient
        from k8s_operator_libs_amd.upgrade.controller import UpgradeController
        from k8s_operator_libs_amd.upgrade.state_manager import (
            ClusterUpgradeStateManager,
        )

        client = FakeClient()
        setup_cluster(client, n_nodes=1)

        def make(identity):
            manager = ClusterUpgradeStateManager(FakeClient(client.cluster))
            return UpgradeController(
                manager, DRIVER_NS, DRIVER_LABELS,
                policy(drainSpec={"enable": True}), resync_seconds=0.05,
            ), identity

        c1, _ = make("replica-1")
        c2, _ = make("replica-2")
        threads = []
        for ctrl, ident in ((c1, "replica-1"), (c2, "replica-2")):
            t = threading.Thread(
                target=ctrl.run_with_leader_election,
                kwargs=dict(lease_name="test-upgrade-lease",
                            identity=ident,
                            lease_duration=0.5, retry_period=0.05),
                daemon=True,
            )
            t.start()
            threads.append(t)

        # exactly one replica wins and reconciles
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            if c1.reconcile_count + c2.reconcile_count > 2:
                break
            time.sleep(0.05)
        leader, standby = (c1, c2) if c1.reconcile_count else (c2, c1)
        assert leader.reconcile_count > 0
        assert standby.reconcile_count == 0, "standby replica reconciled"

        # leader steps down -> standby takes over
        leader.stop()
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            if standby.reconcile_count > 0:
                break
            time.sleep(0.05)
        assert standby.reconcile_count > 0, "no failover to standby"
        standby.stop()
        for t in threads:
            t.join(timeout=5)
