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
