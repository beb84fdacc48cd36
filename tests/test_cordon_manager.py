"""CordonManager tests (reference pkg/upgrade/cordon_manager_test.go:27)."""

from k8s_operator_libs_amd.upgrade.cordon_manager import CordonManager

from builders import NodeBuilder


def test_cordon_uncordon_roundtrip(client):
    node = NodeBuilder("n1").build(client.cluster)
    mgr = CordonManager(client)
    mgr.cordon(node)
    assert client.get_node("n1")["spec"].get("unschedulable") is True
    assert node["spec"]["unschedulable"] is True
    mgr.uncordon(node)
    assert not client.get_node("n1")["spec"].get("unschedulable")
    assert "unschedulable" not in node["spec"]


def test_cordon_idempotent(client):
    node = NodeBuilder("n1").unschedulable().build(client.cluster)
    rv = node["metadata"]["resourceVersion"]
    CordonManager(client).cordon(node)
    assert client.get_node("n1")["metadata"]["resourceVersion"] == rv
