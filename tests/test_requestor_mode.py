"""Requestor (maintenance-operator) mode tests
(reference pkg/upgrade/upgrade_state_test.go:1296-1768 and
upgrade_requestor.go)."""

import pytest

from k8s_operator_libs_amd.core.errors import NotFoundError
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.requestor import (
    DEFAULT_NODE_MAINTENANCE_NAME_PREFIX,
    NodeMaintenanceUpgradeDisabledError,
    RequestorNodeStateManager,
    RequestorOptions,
    condition_changed_predicate,
    get_requestor_opts_from_envs,
    requestor_id_predicate,
)
from k8s_operator_libs_amd.upgrade.state_manager import (
    ClusterUpgradeStateManager,
    StateOptions,
)
from k8s_operator_libs_amd.upgrade.common_manager import CommonUpgradeManager

from builders import (
    DRIVER_LABELS,
    DRIVER_NS,
    NodeMaintenanceBuilder,
)
from simenv import SimDaemonSetController, SimMaintenanceOperator
from test_state_manager import policy, setup_cluster, state_of

NM_API = "maintenance.amd.com/v1alpha1"


def requestor_opts(**kw):
    kw.setdefault("use_maintenance_operator", True)
    kw.setdefault("requestor_id", "amd.gpu.operator")
    kw.setdefault("namespace", "default")
    return RequestorOptions(**kw)


def make_manager(client, **opts_kw):
    return ClusterUpgradeStateManager(
        client, options=StateOptions(requestor=requestor_opts(**opts_kw))
    )


class TestOptions:
    def test_disabled_opts_raise(self, client):
        common = CommonUpgradeManager(client)
        with pytest.raises(NodeMaintenanceUpgradeDisabledError):
            RequestorNodeStateManager(common, RequestorOptions())

    def test_env_opts(self, monkeypatch):
        monkeypatch.setenv("MAINTENANCE_OPERATOR_ENABLED", "true")
        monkeypatch.setenv("MAINTENANCE_OPERATOR_REQUESTOR_ID", "amd.net.operator")
        monkeypatch.setenv("MAINTENANCE_OPERATOR_REQUESTOR_NAMESPACE", "amd-system")
        opts = get_requestor_opts_from_envs()
        assert opts.use_maintenance_operator is True
        assert opts.requestor_id == "amd.net.operator"
        assert opts.namespace == "amd-system"
        assert opts.name_prefix == DEFAULT_NODE_MAINTENANCE_NAME_PREFIX

    def test_env_opts_defaults(self, monkeypatch):
        monkeypatch.delenv("MAINTENANCE_OPERATOR_ENABLED", raising=False)
        opts = get_requestor_opts_from_envs()
        assert opts.use_maintenance_operator is False
        assert opts.namespace == "default"


class TestUpgradeRequired:
    def test_creates_node_maintenance_and_annotates(self, client):
        manager = make_manager(client)
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED
        nm = client.get(NM_API, "NodeMaintenance",
                        f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0", "default")
        assert nm["spec"]["requestorID"] == "amd.gpu.operator"
        assert nm["spec"]["nodeName"] == "node-0"
        anno = client.get_node("node-0")["metadata"]["annotations"]
        assert anno[util.get_upgrade_requestor_mode_annotation_key()] == "true"

    def test_no_max_parallel_gating(self, client):
        # requestor mode intentionally starts ALL upgrade-required nodes
        manager = make_manager(client)
        setup_cluster(client, n_nodes=5, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(maxParallelUpgrades=1))
        for i in range(5):
            assert state_of(client, f"node-{i}") == consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED

    def test_policy_mapped_into_node_maintenance(self, client):
        manager = make_manager(client)
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy(
            drainSpec={"enable": True, "force": True, "timeoutSeconds": 120},
            waitForCompletion={"podSelector": "app=job", "timeoutSecond": 60},
        ))
        nm = client.get(NM_API, "NodeMaintenance",
                        f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0", "default")
        assert nm["spec"]["drainSpec"]["force"] is True
        assert nm["spec"]["drainSpec"]["timeoutSeconds"] == 120
        assert nm["spec"]["waitForPodCompletion"]["podSelector"] == "app=job"


class TestSharedRequestor:
    def test_second_operator_joins_additional_requestors(self, client):
        # an object with the default prefix already exists, owned by the NIC op
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        NodeMaintenanceBuilder(
            f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        ).with_node("node-0").with_requestor("amd.network.operator").build(client.cluster)
        manager = make_manager(client)  # we are amd.gpu.operator
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        nm = client.get(NM_API, "NodeMaintenance",
                        f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0", "default")
        assert nm["spec"]["requestorID"] == "amd.network.operator"
        assert nm["spec"]["additionalRequestors"] == ["amd.gpu.operator"]

    def test_join_is_idempotent(self, client):
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        NodeMaintenanceBuilder(
            f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        ).with_node("node-0").with_requestor("amd.network.operator").build(client.cluster)
        manager = make_manager(client)
        for _ in range(2):
            state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
            manager.apply_state(state, policy())
            # force back to upgrade-required to re-run the processor
            node = client.get_node("node-0")
            manager.common.node_state_provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_UPGRADE_REQUIRED
            )
        nm = client.get(NM_API, "NodeMaintenance",
                        f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0", "default")
        assert nm["spec"]["additionalRequestors"] == ["amd.gpu.operator"]

    def test_custom_prefix_does_not_share(self, client):
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        manager = make_manager(client, name_prefix="gpu-op")
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        nm = client.get(NM_API, "NodeMaintenance", "gpu-op-node-0", "default")
        assert nm["spec"]["requestorID"] == "amd.gpu.operator"

    def test_non_owner_release_removes_from_additional_requestors(self, client):
        setup_cluster(client, node_states=consts.UPGRADE_STATE_UNCORDON_REQUIRED)
        key = util.get_upgrade_requestor_mode_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        nm = NodeMaintenanceBuilder(
            f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        ).with_node("node-0").with_requestor("amd.network.operator").build(client.cluster)
        client.patch(NM_API, "NodeMaintenance", nm["metadata"]["name"],
                     {"spec": {"additionalRequestors": ["amd.gpu.operator"]}},
                     "default")
        manager = make_manager(client)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        live = client.get(NM_API, "NodeMaintenance", nm["metadata"]["name"], "default")
        assert "amd.gpu.operator" not in (live["spec"].get("additionalRequestors") or [])
        # object itself survives: the NIC operator still owns it
        assert live["spec"]["requestorID"] == "amd.network.operator"


class TestMaintenanceLifecycle:
    def test_ready_condition_moves_to_pod_restart(self, client):
        manager = make_manager(client)
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED)
        key = util.get_upgrade_requestor_mode_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        NodeMaintenanceBuilder(
            f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        ).with_node("node-0").with_requestor("amd.gpu.operator").with_conditions(
            "Ready", "True", reason="Ready"
        ).build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_POD_RESTART_REQUIRED

    def test_not_ready_condition_holds(self, client):
        manager = make_manager(client)
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED)
        NodeMaintenanceBuilder(
            f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        ).with_node("node-0").with_requestor("amd.gpu.operator").with_conditions(
            "Pending", "True", reason="Scheduled"
        ).build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED

    def test_missing_object_recovers_to_upgrade_required(self, client):
        manager = make_manager(client)
        setup_cluster(client, pod_hash="old", ds_hash="new",
                      node_states=consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UPGRADE_REQUIRED

    def test_uncordon_completes_and_deletes_owned_object(self, client):
        manager = make_manager(client)
        setup_cluster(client, node_states=consts.UPGRADE_STATE_UNCORDON_REQUIRED)
        key = util.get_upgrade_requestor_mode_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}}})
        NodeMaintenanceBuilder(
            f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        ).with_node("node-0").with_requestor("amd.gpu.operator").build(client.cluster)
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        assert key not in client.get_node("node-0")["metadata"]["annotations"]
        with pytest.raises(NotFoundError):
            client.get(NM_API, "NodeMaintenance",
                       f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0", "default")

    def test_inplace_uncordon_skips_requestor_nodes(self, client):
        # a requestor-mode node in uncordon-required must NOT be uncordoned
        # by the inplace flow (upgrade_inplace.go:124-147)
        manager = ClusterUpgradeStateManager(client)  # requestor disabled
        setup_cluster(client, node_states=consts.UPGRADE_STATE_UNCORDON_REQUIRED)
        key = util.get_upgrade_requestor_mode_annotation_key()
        client.patch("v1", "Node", "node-0",
                     {"metadata": {"annotations": {key: "true"}},
                      "spec": {"unschedulable": True}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # untouched: still uncordon-required and still cordoned
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_UNCORDON_REQUIRED
        assert client.get_node("node-0")["spec"].get("unschedulable") is True


class TestEndToEndRequestor:
    def test_full_lifecycle_with_simulated_maintenance_operator(self, client):
        manager = make_manager(client)
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        SimMaintenanceOperator(client.cluster)
        pol = policy(drainSpec={"enable": True})
        for _ in range(12):
            state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
            manager.apply_state(state, pol)
            manager.wait_idle()
            if state_of(client, "node-0") == consts.UPGRADE_STATE_DONE:
                break
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        # object deleted (maintenance operator honoured the finalizer flow)
        with pytest.raises(NotFoundError):
            client.get(NM_API, "NodeMaintenance",
                       f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0", "default")
        # maintenance operator uncordoned the node on release
        assert not client.get_node("node-0")["spec"].get("unschedulable")
        # driver pod is the new revision
        pods = client.list_pods(namespace=DRIVER_NS, label_selector="app=amdgpu-driver-daemonset")
        assert pods[0]["metadata"]["labels"]["controller-revision-hash"] == "new"

    def test_inplace_and_requestor_coexistence(self, client):
        """Nodes mid-in-place-upgrade complete in-place even after requestor
        mode is enabled (upgrade_state.go:311-325)."""
        manager = make_manager(client)
        setup_cluster(client, n_nodes=2, pod_hash="old", ds_hash="new",
                      node_states=[consts.UPGRADE_STATE_UNCORDON_REQUIRED,
                                   consts.UPGRADE_STATE_UPGRADE_REQUIRED])
        # node-0 was upgraded in-place (no requestor annotation), cordoned
        client.patch("v1", "Node", "node-0", {"spec": {"unschedulable": True}})
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, policy())
        # in-place path uncordoned and completed node-0
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        assert not client.get_node("node-0")["spec"].get("unschedulable")
        # node-1 entered the requestor flow
        assert state_of(client, "node-1") == consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED


class TestPredicates:
    def test_requestor_id_predicate(self):
        pred = requestor_id_predicate("amd.gpu.operator")
        owned = {"spec": {"requestorID": "amd.gpu.operator"}}
        shared = {"spec": {"requestorID": "x", "additionalRequestors": ["amd.gpu.operator"]}}
        other = {"spec": {"requestorID": "x"}}
        assert pred(owned) and pred(shared) and not pred(other)

    def test_condition_changed_predicate(self):
        a = {"metadata": {}, "status": {"conditions": [
            {"type": "Ready", "status": "False", "reason": "Pending"}]}}
        b = {"metadata": {}, "status": {"conditions": [
            {"type": "Ready", "status": "True", "reason": "Ready"}]}}
        assert condition_changed_predicate(a, b)
        assert not condition_changed_predicate(a, a)
        # condition order must not matter
        c1 = {"metadata": {}, "status": {"conditions": [
            {"type": "A", "status": "True"}, {"type": "B", "status": "True"}]}}
        c2 = {"metadata": {}, "status": {"conditions": [
            {"type": "B", "status": "True"}, {"type": "A", "status": "True"}]}}
        assert not condition_changed_predicate(c1, c2)

    def test_finalizer_removal_during_deletion_fires(self):
        old = {"metadata": {"finalizers": ["x"], "deletionTimestamp": "t"},
               "status": {"conditions": []}}
        new = {"metadata": {"finalizers": [], "deletionTimestamp": "t"},
               "status": {"conditions": []}}
        assert condition_changed_predicate(old, new)


class TestSharedWindowEndToEnd:
    def test_gpu_and_nic_operators_share_one_maintenance_window(self, client):
        """The reference's marquee multi-operator flow
        (docs/automatic-ofed-upgrade.md:117-135): GPU and NIC operators with
        the default NodeMaintenance prefix share ONE maintenance window per
        node — the second operator joins additionalRequestors instead of
        creating a duplicate, each removes itself on completion, and the
        owner's deletion releases the node."""
        from simenv import SimMaintenanceOperator

        nm_name = f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
        SimMaintenanceOperator(client.cluster)

        # amdgpu driver DS (out of date) under driver name "amdgpu"
        util.set_driver_name("amdgpu")
        gpu_ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new",
                                  ds_name="amdgpu-driver")
        SimDaemonSetController(client.cluster, gpu_ds, current_hash="new")
        gpu_mgr = make_manager(client, requestor_id="amd.gpu.operator")

        # anic driver DS (out of date) on the SAME node, driver name "anic"
        util.set_driver_name("anic")
        from builders import DaemonSetBuilder, driver_pod_for, make_controller_revision

        nic_labels = {"app": "anic-driver-daemonset"}
        nic_ds = DaemonSetBuilder("anic-driver", labels=nic_labels) \
            .with_desired_number_scheduled(1).build(client.cluster)
        make_controller_revision(nic_ds, "new", revision=2, cluster=client.cluster)
        make_controller_revision(nic_ds, "old", revision=1, cluster=client.cluster)
        driver_pod_for(nic_ds, "node-0", hash_="old").build(client.cluster)
        SimDaemonSetController(client.cluster, nic_ds, current_hash="new")
        nic_mgr = make_manager(client, requestor_id="amd.network.operator")

        pol = policy(drainSpec={"enable": True})
        saw_shared = False
        for _ in range(15):
            util.set_driver_name("amdgpu")
            gpu_mgr.reconcile(DRIVER_NS, DRIVER_LABELS, pol)
            util.set_driver_name("anic")
            nic_mgr.reconcile(DRIVER_NS, nic_labels, pol)
            try:
                nm = client.get(NM_API, "NodeMaintenance", nm_name, "default")
                if nm["spec"].get("additionalRequestors"):
                    saw_shared = True
                    # one object, two requestors — never two objects
                    assert nm["spec"]["requestorID"] in (
                        "amd.gpu.operator", "amd.network.operator")
            except NotFoundError:
                pass
            util.set_driver_name("amdgpu")
            gpu_done = state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
            util.set_driver_name("anic")
            nic_done = state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
            if gpu_done and nic_done:
                break
        assert gpu_done and nic_done, (gpu_done, nic_done)
        assert saw_shared, "operators never shared the maintenance window"
        # window fully released: object gone, node schedulable
        with pytest.raises(NotFoundError):
            client.get(NM_API, "NodeMaintenance", nm_name, "default")
        assert not client.get_node("node-0")["spec"].get("unschedulable")
        # both drivers on their new revisions
        for sel, ns_ in (("app=amdgpu-driver-daemonset", DRIVER_NS),
                         ("app=anic-driver-daemonset", DRIVER_NS)):
            pods = client.list_pods(namespace=ns_, label_selector=sel)
            assert pods and all(
                p["metadata"]["labels"]["controller-revision-hash"] == "new"
                for p in pods)


class TestSharedWindowOverWire:
    def test_shared_maintenance_window_over_http(self):
        """The shared-requestor optimistic-lock protocol at the wire level:
        two operators (amdgpu + anic) through separate RestClients against
        the HTTP apiserver share one NodeMaintenance per node; the
        resourceVersion-locked joins/releases travel real HTTP (409s
        included).  In-process variant: TestSharedWindowEndToEnd."""
        from k8s_operator_libs_amd.core.apiserver import start_apiserver
        from k8s_operator_libs_amd.core.restclient import RestClient
        from builders import DaemonSetBuilder, driver_pod_for, make_controller_revision
        from simenv import SimMaintenanceOperator

        handle = start_apiserver()
        gpu_rest = RestClient(handle.url)
        nic_rest = RestClient(handle.url)
        try:
            nm_name = f"{DEFAULT_NODE_MAINTENANCE_NAME_PREFIX}-node-0"
            SimMaintenanceOperator(handle.cluster)

            class W:
                cluster = handle.cluster

            util.set_driver_name("amdgpu")
            gpu_ds, _ = setup_cluster(W, pod_hash="old", ds_hash="new",
                                      ds_name="amdgpu-driver")
            SimDaemonSetController(handle.cluster, gpu_ds, current_hash="new")
            gpu_mgr = make_manager(gpu_rest, requestor_id="amd.gpu.operator")

            util.set_driver_name("anic")
            nic_labels = {"app": "anic-driver-daemonset"}
            nic_ds = DaemonSetBuilder("anic-driver", labels=nic_labels) \
                .with_desired_number_scheduled(1).build(handle.cluster)
            make_controller_revision(nic_ds, "new", revision=2,
                                     cluster=handle.cluster)
            make_controller_revision(nic_ds, "old", revision=1,
                                     cluster=handle.cluster)
            driver_pod_for(nic_ds, "node-0", hash_="old").build(handle.cluster)
            SimDaemonSetController(handle.cluster, nic_ds, current_hash="new")
            nic_mgr = make_manager(nic_rest, requestor_id="amd.network.operator")

            pol = policy(drainSpec={"enable": True})
            saw_shared = False
            gpu_done = nic_done = False
            for _ in range(25):
                util.set_driver_name("amdgpu")
                gpu_mgr.reconcile(DRIVER_NS, DRIVER_LABELS, pol)
                util.set_driver_name("anic")
                nic_mgr.reconcile(DRIVER_NS, nic_labels, pol)
                try:
                    nm = gpu_rest.get(NM_API, "NodeMaintenance", nm_name,
                                      "default")
                    if nm["spec"].get("additionalRequestors"):
                        saw_shared = True
                except NotFoundError:
                    pass
                util.set_driver_name("amdgpu")
                gpu_done = state_of(gpu_rest, "node-0") == consts.UPGRADE_STATE_DONE
                util.set_driver_name("anic")
                nic_done = state_of(nic_rest, "node-0") == consts.UPGRADE_STATE_DONE
                if gpu_done and nic_done:
                    break
            assert gpu_done and nic_done, (gpu_done, nic_done)
            assert saw_shared, "operators never shared the window over HTTP"
            with pytest.raises(NotFoundError):
                gpu_rest.get(NM_API, "NodeMaintenance", nm_name, "default")
            assert not gpu_rest.get_node("node-0")["spec"].get("unschedulable")
        finally:
            gpu_rest.close()
            nic_rest.close()
            handle.stop()
