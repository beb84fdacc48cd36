"""NodeUpgradeStateProvider tests
(reference pkg/upgrade/node_upgrade_state_provider_test.go:37-69)."""

import pytest

from k8s_operator_libs_amd.core.events import FakeRecorder
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.node_state_provider import NodeUpgradeStateProvider

from builders import NodeBuilder


@pytest.fixture
def provider(client):
    return NodeUpgradeStateProvider(client, FakeRecorder())


def test_change_state_label_roundtrip(client, provider):
    node = NodeBuilder("n1").build(client.cluster)
    provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_UPGRADE_REQUIRED)
    key = util.get_upgrade_state_label_key()
    # both the live object and the caller's snapshot are updated
    assert client.get_node("n1")["metadata"]["labels"][key] == "upgrade-required"
    assert node["metadata"]["labels"][key] == "upgrade-required"
    provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_DONE)
    assert client.get_node("n1")["metadata"]["labels"][key] == "upgrade-done"


def test_change_state_noop_when_same(client, provider):
    node = (
        NodeBuilder("n1")
        .with_upgrade_state(consts.UPGRADE_STATE_DONE)
        .build(client.cluster)
    )
    rv = node["metadata"]["resourceVersion"]
    provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_DONE)
    assert client.get_node("n1")["metadata"]["resourceVersion"] == rv


def test_annotation_set_and_delete(client, provider):
    node = NodeBuilder("n1").build(client.cluster)
    key = util.get_upgrade_requested_annotation_key()
    provider.change_node_upgrade_annotation(node, key, "true")
    assert client.get_node("n1")["metadata"]["annotations"][key] == "true"
    assert node["metadata"]["annotations"][key] == "true"
    # the value "null" deletes the annotation (provider semantics)
    provider.change_node_upgrade_annotation(node, key, consts.NULL_STRING)
    assert key not in client.get_node("n1")["metadata"]["annotations"]
    assert key not in node["metadata"]["annotations"]


def test_get_node(client, provider):
    NodeBuilder("n9").build(client.cluster)
    assert provider.get_node("n9")["metadata"]["name"] == "n9"


def test_events_recorded(client):
    rec = FakeRecorder()
    provider = NodeUpgradeStateProvider(client, rec)
    node = NodeBuilder("n1").build(client.cluster)
    provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_UPGRADE_REQUIRED)
    assert any("Successfully updated node state label" in e for e in rec.events)


def test_patch_failure_records_warning_event_and_raises(client):
    class FailingClient:
        cluster = client.cluster

        def __getattr__(self, name):
            return getattr(client, name)

        def patch(self, *a, **kw):
            raise RuntimeError("apiserver unavailable")

    rec = FakeRecorder()
    provider = NodeUpgradeStateProvider(FailingClient(), rec)
    node = NodeBuilder("n1").build(client.cluster)
    with pytest.raises(RuntimeError):
        provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_UPGRADE_REQUIRED)
    assert any("Failed to update node state label" in e for e in rec.events)
    # the in-memory snapshot was NOT mutated on failure
    assert util.get_upgrade_state_label_key() not in node["metadata"]["labels"]


def test_barrier_timeout_when_cache_never_converges(client, monkeypatch):
    """A cache that never reflects the patch must raise after the deadline
    rather than return (double-fire protection)."""
    from k8s_operator_libs_amd.upgrade import node_state_provider as nsp

    class FrozenCacheClient:
        cluster = client.cluster

        def __getattr__(self, name):
            return getattr(client, name)

        def patch(self, *a, **kw):
            return client.patch(*a, **kw)

        def get_node(self, name):
            # always serve the ORIGINAL (stale) object
            return {"metadata": {"name": name, "labels": {}, "annotations": {}}}

    monkeypatch.setattr(nsp, "_BARRIER_TIMEOUT_S", 0.2)
    provider = NodeUpgradeStateProvider(FrozenCacheClient())
    node = NodeBuilder("n1").build(client.cluster)
    with pytest.raises(nsp.StateChangeTimeoutError):
        provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_UPGRADE_REQUIRED)
