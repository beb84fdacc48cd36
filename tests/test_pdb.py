"""PodDisruptionBudget enforcement: the apiserver substrate blocks evictions
that would violate a budget (429), and the drain retries until replicas
recover or the timeout expires — kubectl-drain semantics."""

import threading
import time

import pytest

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import DrainSpec
from k8s_operator_libs_amd.core.errors import ApiError, NotFoundError
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.drain import drain_node
from k8s_operator_libs_amd.upgrade.drain_manager import DrainConfiguration, DrainManager
from k8s_operator_libs_amd.upgrade.node_state_provider import NodeUpgradeStateProvider

from builders import NodeBuilder, PodBuilder


def mk_pdb(cluster, name, match, min_available=None, max_unavailable=None,
           namespace="default", expected_pods=None):
    spec = {"selector": {"matchLabels": match}}
    if min_available is not None:
        spec["minAvailable"] = min_available
    if max_unavailable is not None:
        spec["maxUnavailable"] = max_unavailable
    obj = {
        "apiVersion": "policy/v1", "kind": "PodDisruptionBudget",
        "metadata": {"name": name, "namespace": namespace},
        "spec": spec,
    }
    if expected_pods is not None:
        obj["status"] = {"expectedPods": expected_pods}
    return cluster.create(obj)


def mk_replicas(cluster, n, node_of, labels):
    pods = []
    for i in range(n):
        pods.append(
            PodBuilder(f"web-{i}", node=node_of(i)).with_labels(labels)
            .with_owner_reference("ReplicaSet", "web-rs").build(cluster)
        )
    return pods


class TestEvictionApi:
    def test_eviction_blocked_at_min_available(self, client):
        mk_replicas(client.cluster, 2, lambda i: f"n{i}", {"app": "web"})
        mk_pdb(client.cluster, "web-pdb", {"app": "web"}, min_available=2)
        with pytest.raises(ApiError) as exc:
            client.evict_pod("web-0", "default")
        assert exc.value.code == 429
        # pod survived
        assert client.get("v1", "Pod", "web-0", "default")

    def test_eviction_allowed_above_budget(self, client):
        mk_replicas(client.cluster, 3, lambda i: f"n{i}", {"app": "web"})
        mk_pdb(client.cluster, "web-pdb", {"app": "web"}, min_available=2)
        client.evict_pod("web-0", "default")
        with pytest.raises(NotFoundError):
            client.get("v1", "Pod", "web-0", "default")
        # next eviction would breach: blocked
        with pytest.raises(ApiError):
            client.evict_pod("web-1", "default")

    def test_max_unavailable_budget(self, client):
        mk_replicas(client.cluster, 4, lambda i: f"n{i}", {"app": "web"})
        mk_pdb(client.cluster, "web-pdb", {"app": "web"}, max_unavailable=1,
               expected_pods=4)
        client.evict_pod("web-0", "default")
        with pytest.raises(ApiError):
            client.evict_pod("web-1", "default")

    def test_percent_min_available(self, client):
        mk_replicas(client.cluster, 4, lambda i: f"n{i}", {"app": "web"})
        mk_pdb(client.cluster, "web-pdb", {"app": "web"}, min_available="50%",
               expected_pods=4)
        client.evict_pod("web-0", "default")
        client.evict_pod("web-1", "default")
        with pytest.raises(ApiError):
            client.evict_pod("web-2", "default")

    def test_unrelated_pdb_ignored(self, client):
        mk_replicas(client.cluster, 1, lambda i: "n0", {"app": "web"})
        mk_pdb(client.cluster, "other", {"app": "db"}, min_available=1)
        client.evict_pod("web-0", "default")  # no matching budget: allowed


class TestDrainWithPdb:
    def test_drain_waits_for_replacement_then_completes(self, client):
        """PDB blocks the eviction until a replacement replica becomes
        healthy elsewhere; the drain retries and then finishes."""
        NodeBuilder("n0").build(client.cluster)
        mk_replicas(client.cluster, 2, lambda i: "n0" if i == 0 else "n1",
                    {"app": "web"})
        mk_pdb(client.cluster, "web-pdb", {"app": "web"}, min_available=2)

        def bring_up_replacement():
            time.sleep(0.25)
            PodBuilder("web-new", node="n2").with_labels({"app": "web"}) \
                .with_owner_reference("ReplicaSet", "web-rs").build(client.cluster)

        t = threading.Thread(target=bring_up_replacement, daemon=True)
        t.start()
        drain_node(client, "n0", DrainSpec(enable=True, timeoutSeconds=10))
        t.join()
        with pytest.raises(NotFoundError):
            client.get("v1", "Pod", "web-0", "default")

    def test_drain_times_out_on_permanently_blocked_pdb(self, client):
        node = NodeBuilder("n0").with_upgrade_state(
            consts.UPGRADE_STATE_DRAIN_REQUIRED
        ).build(client.cluster)
        mk_replicas(client.cluster, 1, lambda i: "n0", {"app": "web"})
        mk_pdb(client.cluster, "web-pdb", {"app": "web"}, min_available=1)
        manager = DrainManager(client, NodeUpgradeStateProvider(client))
        manager.schedule_nodes_drain(DrainConfiguration(
            spec=DrainSpec(enable=True, timeoutSeconds=1), nodes=[node]))
        manager.wait_idle()
        key = util.get_upgrade_state_label_key()
        assert client.get_node("n0")["metadata"]["labels"][key] == \
            consts.UPGRADE_STATE_FAILED
        # the budgeted pod was never deleted
        assert client.get("v1", "Pod", "web-0", "default")
