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
