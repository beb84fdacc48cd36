"""Metrics tests: reconcile histograms, transition counters, exposition."""

import pytest

from k8s_operator_libs_amd.metrics import (
    Counter,
    Gauge,
    Histogram,
    MetricsRegistry,
)
from k8s_operator_libs_amd.upgrade import consts
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import DRIVER_LABELS, DRIVER_NS
from test_state_manager import policy, setup_cluster


def test_histogram_quantiles():
    h = Histogram("h")
    for v in [0.001 * i for i in range(1, 101)]:
        h.observe(v)
    assert h.count == 100
    assert abs(h.quantile(0.5) - 0.05) < 0.005
    assert h.quantile(0.99) >= 0.095
    snap = h.snapshot()
    assert snap["count"] == 100 and snap["sum"] > 0


def test_counter_and_gauge_labels():
    c = Counter("c")
    c.inc("a", "b")
    c.inc("a", "b")
    c.inc("x", "y")
    assert c.value("a", "b") == 2
    g = Gauge("g")
    g.set(3, "done")
    assert g.value("done") == 3


def test_state_manager_records_metrics(client):
    reg = MetricsRegistry()
    manager = ClusterUpgradeStateManager(client, metrics=reg)
    setup_cluster(client, pod_hash="old", ds_hash="new")
    state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
    manager.apply_state(state, policy())
    assert reg.build_state_duration.count == 1
    assert reg.reconcile_duration.count == 1
    assert reg.reconcile_duration.quantile(0.5) > 0
    # "" -> upgrade-required transition counted
    assert reg.state_transitions.value("", consts.UPGRADE_STATE_UPGRADE_REQUIRED) == 1
    # gauge recorded the snapshot grouping
    assert reg.node_states.value("") == 1


def test_failure_counter(client):
    reg = MetricsRegistry()
    manager = ClusterUpgradeStateManager(client, metrics=reg)
    setup_cluster(client, node_states=consts.UPGRADE_STATE_POD_RESTART_REQUIRED,
                  pod_ready=False)
    pod = client.list_pods(namespace=DRIVER_NS)[0]
    client.patch("v1", "Pod", pod["metadata"]["name"],
                 {"status": {"containerStatuses": [
                     {"name": "driver", "ready": False, "restartCount": 11}]}},
                 DRIVER_NS)
    state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
    manager.apply_state(state, policy())
    assert reg.upgrade_failures.value() == 1


def test_prometheus_text_exposition(client):
    reg = MetricsRegistry()
    manager = ClusterUpgradeStateManager(client, metrics=reg)
    setup_cluster(client, pod_hash="old", ds_hash="new")
    state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
    manager.apply_state(state, policy())
    text = reg.render_text()
    assert "amd_upgrade_reconcile_duration_seconds_bucket" in text
    assert 'le="+Inf"' in text
    assert "amd_upgrade_state_transitions_total" in text
    # parseable by the official client if present
    try:
        from prometheus_client.parser import text_string_to_metric_families
    except ImportError:
        pytest.skip("prometheus_client not installed")
    families = list(text_string_to_metric_families(text))
    names = {f.name for f in families}
    assert "amd_upgrade_reconcile_duration_seconds" in names


def test_node_state_gauge_zeroes_on_transition(client):
    from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

    reg = MetricsRegistry()
    manager = ClusterUpgradeStateManager(client, metrics=reg)
    setup_cluster(client, node_states=consts.UPGRADE_STATE_DRAIN_REQUIRED)
    manager.build_state(DRIVER_NS, DRIVER_LABELS)
    assert reg.node_states.value(consts.UPGRADE_STATE_DRAIN_REQUIRED) == 1
    # node moves on; the old state's gauge must drop to 0, not linger
    node = client.get_node("node-0")
    manager.common.node_state_provider.change_node_upgrade_state(
        node, consts.UPGRADE_STATE_DONE
    )
    manager.build_state(DRIVER_NS, DRIVER_LABELS)
    assert reg.node_states.value(consts.UPGRADE_STATE_DRAIN_REQUIRED) == 0
    assert reg.node_states.value(consts.UPGRADE_STATE_DONE) == 1


def test_histogram_quantiles_are_run_global():
    """quantiles must reflect the WHOLE run, not the
    last window.  A run whose large outliers all arrive in the first half
    (pre-overflow) must still show them in p99 after 100k later small
    observations displace the window a truncating reservoir would keep."""
    from k8s_operator_libs_amd.metrics import Histogram

    h = Histogram("t")
    # 5% of the first 10k observations are 1.0s outliers...
    for i in range(10_000):
        h.observe(1.0 if i % 20 == 0 else 0.001)
    # ...then 100k fast ones (a tail-window reservoir now holds only these)
    for _ in range(100_000):
        h.observe(0.001)
    # run-global outlier fraction is 500/110000 ~ 0.45% (expected ~18.6 of
    # the 4096 reservoir slots): the early outliers must still be visible
    assert max(h.samples()) == 1.0
    assert h.quantile(0.999) == 1.0
    # and p50 stays at the common value
    assert h.quantile(0.5) == 0.001
    assert h.count == 110_000


def test_histogram_exact_below_reservoir_capacity():
    from k8s_operator_libs_amd.metrics import Histogram

    h = Histogram("t2")
    for v in [0.001, 0.002, 0.003, 0.004, 0.005]:
        h.observe(v)
    assert sorted(h.samples()) == [0.001, 0.002, 0.003, 0.004, 0.005]
    assert h.quantile(0.5) == 0.003
