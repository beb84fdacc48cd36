"""Capstone integration: the full production topology in one test.

Two driver stacks (amdgpu GPU driver + anic NIC driver) on the same nodes,
each managed by an operator replica under Lease leader election, running
event-driven UpgradeControllers over the HTTP apiserver through informer
caches, in requestor mode against ONE simulated maintenance operator — the
reference's marquee deployment shape (GPU Operator + Network Operator
sharing maintenance windows, docs/automatic-ofed-upgrade.md:117-135)
exercised end-to-end on the wire, with TWO contending replicas per stack.

Writing this test exposed a real bug: a leader renewing its Lease through
an informer-backed client could read its own just-written Lease STALE (the
watch event not yet landed), renew against the stale resourceVersion, hit
the optimistic-lock 409 and fake-demote itself — bistably, depending on
propagation timing.  Fixed in core/leaderelection.py (renew against the
last-written response, not a fresh read).

Because the upgrade-state label key is driver-name-scoped
(amd.com/<driver>-driver-upgrade-state) while the library's driver-name
registry is process-global, the two operators run in SEPARATE PROCESSES —
exactly like production — coordinating purely through the apiserver.
"""

import multiprocessing
import time

import pytest

from k8s_operator_libs_amd.core.apiserver import start_apiserver
from k8s_operator_libs_amd.testing import (
    DRIVER_LABELS,
    DRIVER_NS,
    DaemonSetBuilder,
    NodeBuilder,
    SimDaemonSetController,
    SimMaintenanceOperator,
    driver_pod_for,
    make_controller_revision,
)

N_NODES = 3
NIC_LABELS = {"app": "anic-driver-daemonset"}


def _operator_process(driver_name, url, labels, replica, result_q):
    """One operator replica: own process, own driver-name registry."""
    from k8s_operator_libs_amd.api.upgrade.v1alpha1 import DriverUpgradePolicySpec
    from k8s_operator_libs_amd.core.cache import CachedClient
    from k8s_operator_libs_amd.core.restclient import RestClient
    from k8s_operator_libs_amd.upgrade import consts, util
    from k8s_operator_libs_amd.upgrade.controller import UpgradeController
    from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
    from k8s_operator_libs_amd.upgrade.requestor import RequestorOptions
    from k8s_operator_libs_amd.upgrade.state_manager import (
        ClusterUpgradeStateManager,
        StateOptions,
    )

    util.set_driver_name(driver_name)
    rest = RestClient(url)
    cached = CachedClient(rest)
    manager = ClusterUpgradeStateManager(
        cached,
        options=StateOptions(requestor=RequestorOptions(
            use_maintenance_operator=True,
            requestor_id=f"amd.{driver_name}.operator",
            namespace="default",
        )),
    ).with_pod_deletion_enabled(gpu_pod_deletion_filter)
    policy = DriverUpgradePolicySpec.model_validate({
        "autoUpgrade": True, "maxParallelUpgrades": 0,
        "maxUnavailable": "100%",
        "podDeletion": {"force": True, "deleteEmptyDir": True},
    })
    controller = UpgradeController(
        manager, DRIVER_NS, labels, policy, resync_seconds=0.1,
    )

    state_key = util.get_upgrade_state_label_key()

    def run_until_done():
        import traceback

        deadline = time.monotonic() + 100
        last_exc = ["<none>"]
        controller.start_watches()
        try:
            while time.monotonic() < deadline and not controller._stop.is_set():
                try:
                    controller.reconcile_once()
                except Exception:
                    last_exc[0] = traceback.format_exc(limit=4)
                states = [
                    n["metadata"].get("labels", {}).get(state_key)
                    for n in cached.list_nodes()
                ]
                if states and all(s == consts.UPGRADE_STATE_DONE
                                  for s in states):
                    result_q.put((driver_name, replica, "done",
                                  manager.metrics.reconcile_duration.count))
                    controller.stop()  # also ends the campaign loop
                    return
                controller._wake.wait(0.1)
                controller._wake.clear()
        finally:
            controller.stop_watches()
        result_q.put((driver_name, replica, f"timeout: {last_exc[0]}", 0))
        controller.stop()  # end the campaign loop too

    controller.run = run_until_done  # election invokes the bounded loop
    try:
        # re-campaign after a lost stint (a real Deployment replica restarts
        # its campaign too); generous lease so CPU contention between the
        # four replica processes cannot fake-expire a healthy leader
        deadline = time.monotonic() + 110
        while time.monotonic() < deadline and not controller._stop.is_set():
            controller.run_with_leader_election(
                lease_name=f"{driver_name}-upgrade-lease",
                lease_namespace="default",
                identity=f"{driver_name}-replica-{replica}",
                lease_duration=15.0, retry_period=0.5,
            )
    finally:
        manager.wait_idle()
        cached.stop()
        rest.close()


@pytest.mark.timeout(180)
def test_two_driver_stacks_two_replicas_each_over_http():
    handle = start_apiserver()
    cluster = handle.cluster
    try:
        # --- cluster fixtures: nodes + two out-of-date driver stacks -----
        for i in range(N_NODES):
            NodeBuilder(f"node-{i}").build(cluster)

        gpu_ds = DaemonSetBuilder("amdgpu-driver", labels=dict(DRIVER_LABELS)) \
            .with_desired_number_scheduled(N_NODES).build(cluster)
        make_controller_revision(gpu_ds, "new", revision=2, cluster=cluster)
        make_controller_revision(gpu_ds, "old", revision=1, cluster=cluster)
        nic_ds = DaemonSetBuilder("anic-driver", labels=dict(NIC_LABELS)) \
            .with_desired_number_scheduled(N_NODES).build(cluster)
        make_controller_revision(nic_ds, "new", revision=2, cluster=cluster)
        make_controller_revision(nic_ds, "old", revision=1, cluster=cluster)
        for i in range(N_NODES):
            driver_pod_for(gpu_ds, f"node-{i}", hash_="old").build(cluster)
            driver_pod_for(nic_ds, f"node-{i}", hash_="old").build(cluster)

        SimDaemonSetController(cluster, gpu_ds, current_hash="new")
        SimDaemonSetController(cluster, nic_ds, current_hash="new")
        SimMaintenanceOperator(cluster)

        # --- four operator replicas in separate processes ----------------
        ctx = multiprocessing.get_context("spawn")
        result_q = ctx.Queue()
        procs = []
        for driver_name, labels in (("amdgpu", DRIVER_LABELS),
                                    ("anic", NIC_LABELS)):
            for replica in (1, 2):
                p = ctx.Process(
                    target=_operator_process,
                    args=(driver_name, handle.url, labels, replica, result_q),
                    daemon=True,
                )
                p.start()
                procs.append(p)

        # --- one "done" per driver stack (the elected leaders) -----------
        done = {}
        deadline = time.monotonic() + 120
        while time.monotonic() < deadline and len(done) < 2:
            try:
                driver_name, replica, status, ticks = result_q.get(timeout=5)
            except Exception:
                continue
            if status == "done":
                done[driver_name] = (replica, ticks)
        assert set(done) == {"amdgpu", "anic"}, f"completed: {done}"

        # --- authoritative end state --------------------------------------
        for key_fmt_driver in ("amdgpu", "anic"):
            key = f"amd.com/{key_fmt_driver}-driver-upgrade-state"
            states = [n["metadata"].get("labels", {}).get(key)
                      for n in cluster.list("v1", "Node")]
            assert states == ["upgrade-done"] * N_NODES, (key_fmt_driver, states)
        for labels, ns_ in ((DRIVER_LABELS, DRIVER_NS), (NIC_LABELS, DRIVER_NS)):
            sel = ",".join(f"{k}={v}" for k, v in labels.items())
            pods = cluster.list("v1", "Pod", namespace=ns_, label_selector=sel)
            assert pods and all(
                p["metadata"]["labels"]["controller-revision-hash"] == "new"
                for p in pods
            )
        # nodes schedulable again, no NodeMaintenance objects left behind
        assert all(not n["spec"].get("unschedulable")
                   for n in cluster.list("v1", "Node"))
        assert cluster.list("maintenance.amd.com/v1alpha1",
                            "NodeMaintenance") == []

        for p in procs:
            p.terminate()
            p.join(timeout=10)
    finally:
        for p in procs:
            if p.is_alive():
                p.terminate()
        handle.stop()
