"""Property-based invariants (hypothesis) for the watch protocol.

For arbitrary mutation sequences against the cluster:

1. **RV-anchored replay is exact**: a watch opened at any historical
   list-RV delivers precisely the events with RV > anchor, in RV order,
   and replaying them over the anchored snapshot reconstructs the current
   authoritative state (the client-go reflector correctness contract).
2. **Informer convergence**: a live informer cache converges to the
   authoritative store after any mutation burst, including across a forced
   watch drop (reconnect-from-lastRV path).
"""

import os as _os
import time

from hypothesis import given, settings, strategies as st

from k8s_operator_libs_amd.core import meta
from k8s_operator_libs_amd.core.cache import CachedClient
from k8s_operator_libs_amd.core.client import FakeClient
from k8s_operator_libs_amd.core.fakecluster import FakeCluster

_SCALE = max(1, int(_os.environ.get("HYPOTHESIS_SCALE", "1")))

# an op is (kind, name_idx, op_type) applied to a small node population
_OPS = st.lists(
    st.tuples(
        st.integers(min_value=0, max_value=4),           # which node
        st.sampled_from(["create", "label", "delete"]),  # what to do
    ),
    min_size=1, max_size=40,
)


def _apply(cluster, idx, op, counter):
    name = f"prop-{idx}"
    try:
        if op == "create":
            cluster.create({"apiVersion": "v1", "kind": "Node",
                            "metadata": {"name": name}, "spec": {}})
        elif op == "label":
            counter[0] += 1
            cluster.patch("v1", "Node", name,
                          {"metadata": {"labels": {"v": str(counter[0])}}})
        else:
            cluster.delete("v1", "Node", name)
    except Exception:
        pass  # op invalid for current state (patch/delete of absent, dup create)


def _snapshot(cluster):
    return {
        meta.name(o): meta.resource_version(o)
        for o in cluster.list("v1", "Node")
    }


@settings(max_examples=40 * _SCALE, deadline=None)
@given(ops=_OPS, anchor_at=st.integers(min_value=0, max_value=39))
def test_rv_anchored_replay_reconstructs_state(ops, anchor_at):
    cluster = FakeCluster()
    cluster.watch("v1", "Node").stop()  # turn history on from RV 0
    counter = [0]
    anchor_rv = None
    anchored = None
    for i, (idx, op) in enumerate(ops):
        if i == min(anchor_at, len(ops) - 1):
            anchored, anchor_rv = cluster.list_with_meta("v1", "Node")
        _apply(cluster, idx, op, counter)
    if anchor_rv is None:
        anchored, anchor_rv = cluster.list_with_meta("v1", "Node")

    w = cluster.watch("v1", "Node", resource_version=anchor_rv)
    # replay over the anchored snapshot
    store = {meta.name(o): o for o in anchored}
    last_rv = int(anchor_rv)
    while True:
        ev = w.next(timeout=0)
        if ev is None:
            break
        etype, obj = ev
        rv = int(meta.resource_version(obj))
        assert rv > last_rv, "replayed event at or before the anchor"
        last_rv = rv
        if etype == "DELETED":
            store.pop(meta.name(obj), None)
        else:
            store.setdefault(meta.name(obj), obj)
            store[meta.name(obj)] = obj
    w.stop()
    reconstructed = {n: meta.resource_version(o) for n, o in store.items()}
    assert reconstructed == _snapshot(cluster)


@settings(max_examples=15 * _SCALE, deadline=None)
@given(ops=_OPS, drop_at=st.integers(min_value=0, max_value=39))
def test_informer_converges_across_watch_drop(ops, drop_at):
    cluster = FakeCluster()
    cached = CachedClient(FakeClient(cluster))
    try:
        cached.list("v1", "Node")  # start the informer
        inf = cached._informers[("v1", "Node")]
        counter = [0]
        for i, (idx, op) in enumerate(ops):
            if i == min(drop_at, len(ops) - 1):
                inf._watch.stop()  # force a reconnect mid-burst
            _apply(cluster, idx, op, counter)
        want = _snapshot(cluster)
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            got = {meta.name(o): meta.resource_version(o)
                   for o in cached.list("v1", "Node")}
            if got == want:
                return
            time.sleep(0.01)
        assert got == want, f"cache never converged: {got} != {want}"
    finally:
        cached.stop()
