"""Example operator demo test: full rolling upgrade over REST with metrics."""

import httpx

import examples.amdgpu_upgrade_operator as operator


def test_demo_completes_and_serves_metrics():
    rc = operator.main([
        "--demo", "--demo-nodes", "3", "--interval", "0.02",
        "--metrics-port", "18877",
    ])
    assert rc == 0


def test_metrics_endpoint_serves_prometheus(client):
    from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

    manager = ClusterUpgradeStateManager(client)
    server = operator.serve_metrics(manager, 18878)
    try:
        resp = httpx.get("http://127.0.0.1:18878/metrics")
        assert resp.status_code == 200
        assert "amd_upgrade_reconcile_duration_seconds" in resp.text
        assert httpx.get("http://127.0.0.1:18878/other").status_code == 404
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
        "--interval", "0.02", "--metrics-port", "18879",
    ])
    assert rc == 0
