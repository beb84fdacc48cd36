"""Chaos property: randomized fault injection during rolling upgrades.

For seed-driven schedules of faults injected *between* reconcile ticks —
nodes flapping NotReady, driver pods crash-looping then recovering, validator
pods deleted, timeout annotations corrupted, spurious cordons — the state
machine must still converge every node to upgrade-done once faults stop, with
labels never leaving the 13-state alphabet."""

import random

from hypothesis import given, settings, strategies as st

# Scale example counts for deep campaigns: HYPOTHESIS_SCALE=20 multiplies
# every property's max_examples (default 1).
import os as _os

_SCALE = max(1, int(_os.environ.get("HYPOTHESIS_SCALE", "1")))

from k8s_operator_libs_amd.core import FakeClient
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import PodBuilder
from simenv import SimDaemonSetController, SimMaintenanceOperator
from test_state_manager import policy, setup_cluster


def _inject_fault(rng, client, n_nodes):
    node = f"node-{rng.randrange(n_nodes)}"
    fault = rng.randrange(6)
    try:
        if fault == 0:  # node goes NotReady briefly
            client.patch("v1", "Node", node,
                         {"status": {"conditions": [{"type": "Ready",
                                                     "status": "False"}]}})
        elif fault == 1:  # node comes back Ready
            client.patch("v1", "Node", node,
                         {"status": {"conditions": [{"type": "Ready",
                                                     "status": "True"}]}})
        elif fault == 2:  # spurious manual cordon
            client.patch("v1", "Node", node, {"spec": {"unschedulable": True}})
        elif fault == 3:  # corrupt a timeout annotation
            key = util.get_validation_start_time_annotation_key()
            client.patch("v1", "Node", node,
                         {"metadata": {"annotations": {key: "garbage"}}})
        elif fault == 4:  # validator pod deleted (recreated not-ready)
            client.delete("v1", "Pod", f"validator-{node}", "amd-gpu-operator")
        elif fault == 5:  # a stray workload pod lands on the node
            PodBuilder(f"stray-{node}-{rng.randrange(1000)}", node=node) \
                .with_owner_reference("ReplicaSet", "rs").build(client.cluster)
    except Exception:
        pass  # faults racing each other is part of the chaos


@settings(max_examples=12 * _SCALE, deadline=None)
@given(seed=st.integers(min_value=0, max_value=10_000))
def test_converges_despite_random_faults(seed):
    rng = random.Random(seed)
    n_nodes = rng.randrange(2, 8)
    client = FakeClient()
    ds, _ = setup_cluster(client, n_nodes=n_nodes, pod_hash="old", ds_hash="new")
    SimDaemonSetController(client.cluster, ds, current_hash="new")
    options = None
    if rng.randrange(2):  # half the schedules run in requestor mode
        from k8s_operator_libs_amd.upgrade.requestor import RequestorOptions
        from k8s_operator_libs_amd.upgrade.state_manager import StateOptions

        SimMaintenanceOperator(
            client.cluster,
            evict_filter=lambda pod: pod["metadata"].get("labels", {})
            .get("app") != "amd-gpu-validator",
        )
        options = StateOptions(requestor=RequestorOptions(
            use_maintenance_operator=True, requestor_id="amd.gpu.operator",
            namespace="default"))
    manager = ClusterUpgradeStateManager(client, options=options)
    pol = policy(maxParallelUpgrades=rng.randrange(0, 3),
                 maxUnavailable="100%",
                 drainSpec={"enable": bool(rng.randrange(2))})
    state_key = util.get_upgrade_state_label_key()

    # chaos phase: faults between ticks
    for _ in range(rng.randrange(5, 15)):
        manager.reconcile("amd-gpu-operator", {"app": "amdgpu-driver-daemonset"},
                          pol, converge=bool(rng.randrange(2)))
        for _ in range(rng.randrange(0, 3)):
            _inject_fault(rng, client, n_nodes)
        for n in client.list_nodes():
            assert n["metadata"]["labels"].get(state_key, "") in consts.ALL_STATES

    # recovery phase: faults stop, nodes made Ready, machine must converge
    for i in range(n_nodes):
        client.patch("v1", "Node", f"node-{i}",
                     {"status": {"conditions": [{"type": "Ready",
                                                 "status": "True"}]}})
    for _ in range(40 * n_nodes):
        manager.reconcile("amd-gpu-operator", {"app": "amdgpu-driver-daemonset"},
                          pol, converge=True)
        states = [n["metadata"]["labels"].get(state_key, "")
                  for n in client.list_nodes()]
        if all(s == consts.UPGRADE_STATE_DONE for s in states):
            break
    assert all(
        n["metadata"]["labels"].get(state_key) == consts.UPGRADE_STATE_DONE
        for n in client.list_nodes()
    ), f"seed {seed} did not converge: {states}"
    # every driver pod ended on the new revision
    for p in client.list_pods(namespace="amd-gpu-operator",
                              label_selector="app=amdgpu-driver-daemonset"):
        assert p["metadata"]["labels"]["controller-revision-hash"] == "new"


@settings(max_examples=4 * _SCALE, deadline=None)
@given(seed=st.integers(min_value=0, max_value=10_000))
def test_converges_despite_faults_over_the_wire(seed):
    """The same chaos campaign through the FULL production stack — state
    machine -> CachedClient informers -> REST -> HTTP apiserver — with an
    extra wire-only fault: the apiserver itself restarting mid-campaign.
    Transient staleness (BuildStateError / NotFound / missing revisions) is
    requeued exactly like a Reconcile error."""
    import socket
    import time as _time

    from k8s_operator_libs_amd.core.apiserver import start_apiserver
    from k8s_operator_libs_amd.core.cache import CachedClient
    from k8s_operator_libs_amd.core.errors import ApiError, NotFoundError
    from k8s_operator_libs_amd.core.fakecluster import FakeCluster
    from k8s_operator_libs_amd.core.restclient import RestClient
    from k8s_operator_libs_amd.upgrade.pod_manager import StaleClusterViewError
    from k8s_operator_libs_amd.upgrade.state_manager import BuildStateError

    rng = random.Random(seed)
    n_nodes = rng.randrange(2, 5)
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cluster = FakeCluster()
    handle = start_apiserver(port=port, cluster=cluster)
    rest = RestClient(handle.url)
    cached = CachedClient(rest)

    class W:
        pass

    W.cluster = cluster
    ds, _ = setup_cluster(W, n_nodes=n_nodes, pod_hash="old", ds_hash="new")
    SimDaemonSetController(cluster, ds, current_hash="new")
    sim_client = FakeClient(cluster)  # faults act directly on the store
    manager = ClusterUpgradeStateManager(cached)
    pol = policy(maxParallelUpgrades=rng.randrange(0, 3),
                 maxUnavailable="100%",
                 drainSpec={"enable": bool(rng.randrange(2))})
    state_key = util.get_upgrade_state_label_key()

    def tick():
        try:
            manager.reconcile("amd-gpu-operator",
                              {"app": "amdgpu-driver-daemonset"}, pol,
                              converge=True)
        except (BuildStateError, NotFoundError, StaleClusterViewError,
                ApiError):
            _time.sleep(0.02)

    try:
        restarts = 0
        for _ in range(rng.randrange(4, 10)):
            tick()
            if restarts < 2 and rng.randrange(3) == 0:
                handle.stop()
                _time.sleep(0.05)
                handle = start_apiserver(port=port, cluster=cluster)
                restarts += 1
            for _ in range(rng.randrange(0, 2)):
                _inject_fault(rng, sim_client, n_nodes)

        for i in range(n_nodes):
            cluster.patch("v1", "Node", f"node-{i}",
                          {"status": {"conditions": [{"type": "Ready",
                                                      "status": "True"}]}})
        deadline = _time.monotonic() + 60
        while _time.monotonic() < deadline:
            tick()
            states = [n["metadata"].get("labels", {}).get(state_key, "")
                      for n in cluster.list("v1", "Node")]
            if all(s == consts.UPGRADE_STATE_DONE for s in states):
                break
            _time.sleep(0.02)
        assert all(
            n["metadata"].get("labels", {}).get(state_key)
            == consts.UPGRADE_STATE_DONE
            for n in cluster.list("v1", "Node")
        ), f"seed {seed} did not converge over the wire: {states}"
    finally:
        manager.common.drain_manager.wait_idle()
        manager.common.pod_manager.wait_idle()
        cached.stop()
        rest.close()
        handle.stop()
