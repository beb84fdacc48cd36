"""Example operator demo test: full rolling upgrade over REST with metrics."""

import httpx

import examples.amdgpu_upgrade_operator as operator


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_demo_completes_and_serves_metrics():
    rc = operator.main([
        "--demo", "--demo-nodes", "3", "--interval", "0.02",
        "--metrics-port", str(_free_port()),
    ])
    assert rc == 0


def test_metrics_endpoint_serves_prometheus(client):
    from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

    manager = ClusterUpgradeStateManager(client)
    port = _free_port()
    server = operator.serve_metrics(manager, port)
    try:
        resp = httpx.get(f"http://127.0.0.1:{port}/metrics")
        assert resp.status_code == 200
        assert "amd_upgrade_reconcile_duration_seconds" in resp.text
        assert httpx.get(f"http://127.0.0.1:{port}/other").status_code == 404
    finally:
        server.shutdown()


class TestPolicyWebhook:
    def _review(self, policy):
        return {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
                "request": {"uid": "u1", "object": {
                    "spec": {"driverUpgradePolicy": policy}}}}

    def test_valid_policy_allowed(self):
        import examples.policy_webhook as wh

        out = wh.review_response(self._review(
            {"autoUpgrade": True, "maxParallelUpgrades": 2,
             "maxUnavailable": "25%"}))
        assert out["response"]["allowed"] is True
        assert out["response"]["uid"] == "u1"

    def test_invalid_policy_rejected_with_fields(self):
        import examples.policy_webhook as wh

        out = wh.review_response(self._review(
            {"maxParallelUpgrades": -3, "maxUnavailable": "junk"}))
        resp = out["response"]
        assert resp["allowed"] is False
        assert "maxParallelUpgrades" in resp["status"]["message"]

    def test_missing_policy_allowed(self):
        import examples.policy_webhook as wh

        out = wh.review_response({"request": {"uid": "u2", "object": {"spec": {}}}})
        assert out["response"]["allowed"] is True

    def test_over_http(self):
        import threading

        import httpx
        import uvicorn

        import examples.policy_webhook as wh

        app = wh.create_app()
        config = uvicorn.Config(app, host="127.0.0.1", port=18912, log_level="error")
        server = uvicorn.Server(config)
        t = threading.Thread(target=server.run, daemon=True)
        t.start()
        import time
        deadline = time.monotonic() + 10
        while not server.started and time.monotonic() < deadline:
            time.sleep(0.02)
        try:
            resp = httpx.post("http://127.0.0.1:18912/validate",
                              json=self._review({"maxParallelUpgrades": -1}))
            assert resp.status_code == 200
            assert resp.json()["response"]["allowed"] is False
        finally:
            server.should_exit = True
            t.join(timeout=5)


def test_demo_requestor_mode_completes():
    rc = operator.main([
        "--demo", "--demo-requestor", "--demo-nodes", "2",
        "--interval", "0.02", "--metrics-port", str(_free_port()),
    ])
    assert rc == 0


def test_production_entrypoint_with_leader_election(monkeypatch):
    """The REAL production path, end-to-end: `main()` (no --demo) resolves
    the cluster from $KUBERNETES_MASTER, campaigns for the Lease, wires the
    event-driven controller, and completes a rolling upgrade over HTTP."""
    import threading
    import time

    from k8s_operator_libs_amd.core.apiserver import start_apiserver
    from k8s_operator_libs_amd.testing import (
        DRIVER_NS,
        SimDaemonSetController,
    )
    from k8s_operator_libs_amd.upgrade import consts, util
    from test_state_manager import setup_cluster

    handle = start_apiserver()
    monkeypatch.setenv("KUBERNETES_MASTER", handle.url)
    monkeypatch.delenv("KUBECONFIG", raising=False)

    class W:
        cluster = handle.cluster

    ds, _ = setup_cluster(W, n_nodes=2, pod_hash="old", ds_hash="new")
    SimDaemonSetController(handle.cluster, ds, current_hash="new")
    # ready validator pods: main() enables the validation phase
    for i in range(2):
        handle.cluster.create({
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": f"validator-node-{i}", "namespace": DRIVER_NS,
                         "labels": {"app": "amd-gpu-validator"}},
            "spec": {"nodeName": f"node-{i}",
                     "containers": [{"name": "v", "image": "validator"}]},
            "status": {"phase": "Running",
                       "containerStatuses": [{"name": "v", "ready": True,
                                              "restartCount": 0}]},
        })

    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        metrics_port = s.getsockname()[1]
    rc = {}

    def run():
        rc["v"] = operator.main([
            "--namespace", DRIVER_NS,
            "--interval", "0.05", "--metrics-port", str(metrics_port),
        ])

    t = threading.Thread(target=run, daemon=True)
    t.start()
    try:
        state_key = util.get_upgrade_state_label_key()
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            states = [
                n["metadata"].get("labels", {}).get(state_key)
                for n in handle.cluster.list("v1", "Node")
            ]
            if states and all(s == consts.UPGRADE_STATE_DONE for s in states):
                break
            time.sleep(0.1)
        assert states == [consts.UPGRADE_STATE_DONE] * 2, states
        # the operator holds the Lease (leader election actually ran)
        leases = handle.cluster.list("coordination.k8s.io/v1", "Lease")
        assert any(
            l["metadata"]["name"] == "amd-gpu-operator-upgrade" for l in leases
        ), leases
        # metrics endpoint live on the production path too
        resp = httpx.get(f"http://127.0.0.1:{metrics_port}/metrics", timeout=5)
        assert "amd_upgrade_reconcile_duration_seconds" in resp.text
    finally:
        handle.stop()
        t.join(timeout=10)


def test_sigterm_releases_lease(monkeypatch):
    """Graceful shutdown: SIGTERM stops the controller and the elector
    releases the Lease, so a standby replica can take over immediately."""
    import os
    import signal
    import threading
    import time

    from k8s_operator_libs_amd.core.apiserver import start_apiserver

    handle = start_apiserver()
    monkeypatch.setenv("KUBERNETES_MASTER", handle.url)
    monkeypatch.delenv("KUBECONFIG", raising=False)

    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]

    # run main() on the MAIN thread so signal handlers install; drive the
    # SIGTERM from a helper thread once the lease is held
    def killer():
        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            leases = handle.cluster.list("coordination.k8s.io/v1", "Lease")
            if leases:
                os.kill(os.getpid(), signal.SIGTERM)
                return
            time.sleep(0.05)

    t = threading.Thread(target=killer, daemon=True)
    t.start()
    old_term = signal.getsignal(signal.SIGTERM)
    old_int = signal.getsignal(signal.SIGINT)
    try:
        rc = operator.main([
            "--namespace", "amd-gpu-operator",
            "--interval", "0.05", "--metrics-port", str(port),
        ])
        assert rc == 0
        lease = handle.cluster.list("coordination.k8s.io/v1", "Lease")[0]
        assert not lease["spec"].get("holderIdentity"), lease["spec"]
    finally:
        signal.signal(signal.SIGTERM, old_term)
        signal.signal(signal.SIGINT, old_int)
        handle.stop()
