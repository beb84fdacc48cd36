"""Property-based invariants (hypothesis) for the rolling-upgrade window.

For arbitrary cluster sizes and policies, starting from a healthy all-Ready
cluster, the state machine must:

- complete every node to upgrade-done,
- never exceed maxParallelUpgrades nodes in progress at any observation,
- never exceed the maxUnavailable budget of cordoned nodes,
- fire every (from,to) label transition at most once per node lifecycle.
"""

import math

from hypothesis import given, settings, strategies as st

# Scale example counts for deep campaigns: HYPOTHESIS_SCALE=20 multiplies
# every property's max_examples (default 1).
import os as _os

_SCALE = max(1, int(_os.environ.get("HYPOTHESIS_SCALE", "1")))

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import IntOrString
from k8s_operator_libs_amd.core import FakeClient
from k8s_operator_libs_amd.metrics import MetricsRegistry
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from simenv import SimDaemonSetController
from test_state_manager import policy, setup_cluster


@settings(max_examples=25 * _SCALE, deadline=None)
@given(
    n_nodes=st.integers(min_value=1, max_value=8),
    max_parallel=st.integers(min_value=0, max_value=4),
    max_unavailable_pct=st.sampled_from([25, 50, 100]),
    drain=st.booleans(),
    live=st.booleans(),
)
def test_rolling_window_invariants(n_nodes, max_parallel, max_unavailable_pct,
                                   drain, live):
    client = FakeClient()
    reg = MetricsRegistry()
    ds, _ = setup_cluster(client, n_nodes=n_nodes, pod_hash="old", ds_hash="new")
    SimDaemonSetController(client.cluster, ds, current_hash="new")
    manager = ClusterUpgradeStateManager(client, metrics=reg)
    pol = policy(
        maxParallelUpgrades=max_parallel,
        maxUnavailable=f"{max_unavailable_pct}%",
        drainSpec={"enable": drain},
    )
    state_key = util.get_upgrade_state_label_key()
    budget = IntOrString.scaled_value(f"{max_unavailable_pct}%", n_nodes, True)

    in_progress_states = set(consts.ALL_STATES) - {
        consts.UPGRADE_STATE_UNKNOWN,
        consts.UPGRADE_STATE_DONE,
        consts.UPGRADE_STATE_UPGRADE_REQUIRED,
    }

    for tick in range(40 * n_nodes):
        state = manager.reconcile(
            "amd-gpu-operator", {"app": "amdgpu-driver-daemonset"}, pol,
            converge=live,
        )
        nodes = client.list_nodes()
        states = [n["metadata"]["labels"].get(state_key, "") for n in nodes]
        in_progress = sum(s in in_progress_states for s in states)
        cordoned = sum(bool(n["spec"].get("unschedulable")) for n in nodes)
        if max_parallel > 0:
            assert in_progress <= max_parallel, (
                f"window exceeded: {in_progress} > {max_parallel} at tick {tick}"
            )
        assert cordoned <= max(budget, 1), (
            f"unavailability budget exceeded: {cordoned} cordoned > {budget}"
        )
        for s in states:
            assert s in consts.ALL_STATES
        if all(s == consts.UPGRADE_STATE_DONE for s in states):
            break
    assert all(
        n["metadata"]["labels"].get(state_key) == consts.UPGRADE_STATE_DONE
        for n in client.list_nodes()
    ), f"did not converge: {states}"
    # every node uncordoned at the end
    assert not any(n["spec"].get("unschedulable") for n in client.list_nodes())
    # each transition fired exactly n_nodes times at most (once per node)
    for (frm, to), count in reg.state_transitions.items().items():
        assert count <= n_nodes, f"{frm}->{to} fired {count} times for {n_nodes} nodes"


@settings(max_examples=40 * _SCALE, deadline=None)
@given(
    total=st.integers(min_value=0, max_value=200),
    pct=st.integers(min_value=0, max_value=100),
)
def test_int_or_percent_bounds(total, pct):
    up = IntOrString.scaled_value(f"{pct}%", total, True)
    down = IntOrString.scaled_value(f"{pct}%", total, False)
    assert down <= up <= down + 1
    assert up == math.ceil(pct * total / 100)
    assert 0 <= up <= total or pct > 100


_key = st.text(alphabet="abcz-._/", min_size=1, max_size=10).filter(
    lambda s: s[0].isalnum() or s[0] in "abcz"
)
_val = st.text(alphabet="abcz123-", min_size=0, max_size=8)


@settings(max_examples=80 * _SCALE, deadline=None)
@given(
    reqs=st.lists(
        st.one_of(
            st.tuples(st.just("eq"), _key, _val),
            st.tuples(st.just("neq"), _key, _val),
            st.tuples(st.just("exists"), _key, st.just("")),
            st.tuples(st.just("notexists"), _key, st.just("")),
            st.tuples(st.just("in"), _key,
                      st.lists(_val.filter(bool), min_size=1, max_size=3)),
        ),
        min_size=1, max_size=4,
    ),
    labels=st.dictionaries(_key, _val, max_size=4),
)
def test_label_selector_roundtrip_matches_composed_predicate(reqs, labels):
    """Build a selector string from structured requirements, parse it, and
    check matches() against the independently-composed predicate."""
    from k8s_operator_libs_amd.core.meta import LabelSelector

    parts, preds = [], []
    for op, key, val in reqs:
        if op == "eq":
            parts.append(f"{key}={val}")
            preds.append(lambda l, k=key, v=val: l.get(k) == v)
        elif op == "neq":
            parts.append(f"{key}!={val}")
            preds.append(lambda l, k=key, v=val: not (k in l and l[k] == v))
        elif op == "exists":
            parts.append(key)
            preds.append(lambda l, k=key: k in l)
        elif op == "notexists":
            parts.append(f"!{key}")
            preds.append(lambda l, k=key: k not in l)
        elif op == "in":
            parts.append(f"{key} in ({','.join(val)})")
            preds.append(lambda l, k=key, vs=set(val): l.get(k) in vs)
    selector = ",".join(parts)
    sel = LabelSelector(selector)
    expected = all(p(labels) for p in preds)
    assert sel.matches(labels) == expected, (selector, labels)


@settings(max_examples=60 * _SCALE, deadline=None)
@given(junk=st.text(max_size=40))
def test_label_selector_never_crashes_on_junk(junk):
    from k8s_operator_libs_amd.core.meta import LabelSelector

    sel = LabelSelector(junk)
    sel.matches({"a": "b"})
    sel.matches({})


_json_scalar = st.one_of(st.none(), st.booleans(), st.integers(-9, 9),
                         st.text(alphabet="xyz", max_size=3))
_json = st.recursive(
    _json_scalar,
    lambda children: st.one_of(
        st.lists(children, max_size=3),
        st.dictionaries(st.text(alphabet="abc", min_size=1, max_size=2),
                        children, max_size=3),
    ),
    max_leaves=12,
)


def _spec_merge_patch(target, patch):
    """RFC 7386 pseudocode, transcribed independently of the implementation."""
    if isinstance(patch, dict):
        if not isinstance(target, dict):
            target = {}
        result = dict(target)
        for k, v in patch.items():
            if v is None:
                result.pop(k, None)
            else:
                result[k] = _spec_merge_patch(result.get(k), v)
        return result
    return patch


@settings(max_examples=120 * _SCALE, deadline=None)
@given(target=_json, patch=_json)
def test_merge_patch_matches_rfc_pseudocode(target, patch):
    import copy as _copy

    from k8s_operator_libs_amd.core.meta import json_merge_patch

    expected = _spec_merge_patch(_copy.deepcopy(target), patch)
    got = json_merge_patch(_copy.deepcopy(target), patch)
    assert got == expected


@settings(max_examples=120 * _SCALE, deadline=None)
@given(tree=_json)
def test_native_deep_copy_matches_stdlib(tree):
    """The C++ _jsonops.deep_copy must agree with copy.deepcopy on arbitrary
    JSON trees and produce fully independent containers."""
    import copy as _copy

    from k8s_operator_libs_amd.core import meta

    got = meta.deep_copy(tree)
    assert got == _copy.deepcopy(tree)
    # independence: mutate every dict/list in the copy; original unchanged
    snapshot = _copy.deepcopy(tree)

    def mutate(node):
        if isinstance(node, dict):
            node["__mut__"] = 1
            for v in list(node.values()):
                mutate(v)
        elif isinstance(node, list):
            node.append("__mut__")
            for v in node[:-1]:
                mutate(v)

    mutate(got)
    assert tree == snapshot
