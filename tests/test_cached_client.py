"""Informer-cache tests: staleness, barrier convergence, end-to-end over a
lagging cache — the scenario the reference's patch-then-poll exists for
(node_upgrade_state_provider.go:92-117)."""

import time

import pytest

from k8s_operator_libs_amd.core.cache import CachedClient
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.node_state_provider import NodeUpgradeStateProvider
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import DRIVER_LABELS, DRIVER_NS, NodeBuilder
from simenv import SimDaemonSetController
from test_state_manager import policy, setup_cluster, state_of


@pytest.fixture
def cached(client):
    c = CachedClient(client)
    yield c
    c.stop()


def test_reads_served_from_cache(client, cached):
    NodeBuilder("n1").build(client.cluster)
    node = cached.get_node("n1")
    assert node["metadata"]["name"] == "n1"
    # a second node created after informer start appears via the watch
    NodeBuilder("n2").build(client.cluster)
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        try:
            cached.get_node("n2")
            break
        except Exception:
            time.sleep(0.005)
    assert cached.get_node("n2")["metadata"]["name"] == "n2"


def test_cache_lags_writes(client):
    cached = CachedClient(client, sync_delay=0.15)
    try:
        NodeBuilder("n1").build(client.cluster)
        cached.get_node("n1")  # prime informer
        cached.patch("v1", "Node", "n1", {"metadata": {"labels": {"x": "1"}}})
        # immediately after the write the cache is still stale
        assert "x" not in cached.get_node("n1")["metadata"]["labels"]
        time.sleep(0.4)
        assert cached.get_node("n1")["metadata"]["labels"]["x"] == "1"
    finally:
        cached.stop()


def test_provider_barrier_converges_over_stale_cache(client):
    """The core hard part: the provider must not return until
    the cache reflects the new state label, else the next reconcile
    double-fires the transition."""
    cached = CachedClient(client, sync_delay=0.1)
    try:
        node_live = NodeBuilder("n1").build(client.cluster)
        provider = NodeUpgradeStateProvider(cached)
        node = cached.get_node("n1")
        t0 = time.monotonic()
        provider.change_node_upgrade_state(node, consts.UPGRADE_STATE_UPGRADE_REQUIRED)
        waited = time.monotonic() - t0
        # the call blocked for roughly the cache lag, then returned with the
        # cache coherent
        assert waited >= 0.05
        key = util.get_upgrade_state_label_key()
        assert cached.get_node("n1")["metadata"]["labels"][key] == "upgrade-required"
    finally:
        cached.stop()


def test_full_upgrade_over_lagging_cache(client):
    """End-to-end single-node upgrade with a 30ms-lag informer cache: every
    transition must fire exactly once."""
    cached = CachedClient(client, sync_delay=0.03)
    try:
        ds, _ = setup_cluster(client, pod_hash="old", ds_hash="new")
        SimDaemonSetController(client.cluster, ds, current_hash="new")
        manager = ClusterUpgradeStateManager(cached)
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%")
        transitions = manager.metrics.state_transitions
        # generous budget: under parallel test load the watch thread gets
        # scheduled late and each tick may deliver nothing yet
        for _ in range(30):
            state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
            manager.apply_state(state, pol)
            manager.wait_idle()
            if state_of(client, "node-0") == consts.UPGRADE_STATE_DONE:
                break
            # model the operator's requeue interval: give the watch stream
            # time to deliver pod events (labels are barrier-protected,
            # pod/DS caches are eventually consistent like controller-runtime)
            time.sleep(0.1)
        assert state_of(client, "node-0") == consts.UPGRADE_STATE_DONE
        # no transition fired twice
        for (frm, to), count in transitions.items().items():
            assert count == 1, f"transition {frm}->{to} fired {count} times"
    finally:
        cached.stop()


def test_informer_reconnects_after_watch_drop(client):
    """A dropped watch stream must not silently freeze the cache: the
    informer reconnects and relists (resync semantics)."""
    cached = CachedClient(client)
    try:
        NodeBuilder("n1").build(client.cluster)
        cached.get_node("n1")  # start informer
        inf = cached._informers[("v1", "Node")]
        # simulate a server-side stream drop
        inf._watch.stop()
        # mutate while the stream is down
        client.patch("v1", "Node", "n1", {"metadata": {"labels": {"x": "1"}}})
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            if cached.get_node("n1")["metadata"].get("labels", {}).get("x") == "1":
                break
            time.sleep(0.02)
        assert cached.get_node("n1")["metadata"]["labels"]["x"] == "1"
    finally:
        cached.stop()


def test_rv_barrier_event_driven():
    """The provider's patch-then-confirm barrier over a cached client uses
    the event-driven RV wait (no polling): after patch, wait_for_resource_version
    returns once the informer sees the patch's RV."""
    import time

    from k8s_operator_libs_amd.core.cache import CachedClient
    from k8s_operator_libs_amd.core.client import FakeClient
    from k8s_operator_libs_amd.core.fakecluster import FakeCluster

    cluster = FakeCluster()
    cluster.create({"apiVersion": "v1", "kind": "Node",
                    "metadata": {"name": "n1"}, "spec": {}})
    cached = CachedClient(FakeClient(cluster), sync_delay=0.05)
    try:
        cached.get("v1", "Node", "n1")
        resp = cached.patch("v1", "Node", "n1",
                            {"metadata": {"labels": {"k": "v"}}})
        rv = resp["metadata"]["resourceVersion"]
        t0 = time.monotonic()
        assert cached.wait_for_resource_version("v1", "Node", "n1", "", rv, 5.0)
        # the 50ms artificial informer lag bounds the wait from below;
        # event-driven wakeup bounds it from above (well under a poll sweep)
        assert cached.get("v1", "Node", "n1")["metadata"]["labels"]["k"] == "v"
        # timeout path: an RV the cluster will never reach
        assert not cached.wait_for_resource_version(
            "v1", "Node", "n1", "", str(int(rv) + 1000), 0.2)
    finally:
        cached.stop()


def test_provider_barrier_uses_rv_fast_path():
    """End-to-end: state transitions through the provider over a laggy
    informer cache still fire exactly once, via the RV fast path."""
    from k8s_operator_libs_amd.core.cache import CachedClient
    from k8s_operator_libs_amd.core.client import FakeClient
    from k8s_operator_libs_amd.core.fakecluster import FakeCluster
    from k8s_operator_libs_amd.upgrade.node_state_provider import (
        NodeUpgradeStateProvider,
    )

    cluster = FakeCluster()
    cluster.create({"apiVersion": "v1", "kind": "Node",
                    "metadata": {"name": "n1", "labels": {}}, "spec": {}})
    cached = CachedClient(FakeClient(cluster), sync_delay=0.02)
    try:
        provider = NodeUpgradeStateProvider(cached)
        node = provider.get_node("n1")
        provider.change_node_upgrade_state(node, "cordon-required")
        assert provider.get_node("n1")["metadata"]["labels"][
            util.get_upgrade_state_label_key()] == "cordon-required"
    finally:
        cached.stop()
