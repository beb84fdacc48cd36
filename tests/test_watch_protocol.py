"""Kubernetes watch-protocol semantics.

Pins the real-apiserver behaviors the informer stack depends on:
resourceVersion-anchored LIST-then-WATCH (client-go reflector contract,
node_upgrade_state_provider.go:92-117 staleness rationale), 410-Gone relist,
namespace-scoped watch paths, label-selector watches with the
stops-matching -> DELETED transform, BOOKMARK frames, and deletion events
carrying fresh resourceVersions.  Exercised at both the FakeCluster layer and
over the wire (RestClient -> HTTP mini-apiserver)."""

import json
import threading
import time

import httpx
import pytest

from k8s_operator_libs_amd.core.apiserver import start_apiserver
from k8s_operator_libs_amd.core.cache import CachedClient
from k8s_operator_libs_amd.core.client import FakeClient
from k8s_operator_libs_amd.core.errors import GoneError
from k8s_operator_libs_amd.core.fakecluster import FakeCluster
from k8s_operator_libs_amd.core.restclient import RestClient


def node(name, labels=None):
    return {"apiVersion": "v1", "kind": "Node",
            "metadata": {"name": name, "labels": labels or {}}, "spec": {}}


def pod(name, ns="default", labels=None):
    return {"apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": name, "namespace": ns, "labels": labels or {}},
            "spec": {"nodeName": "n1"}}


# ---------------------------------------------------------------- FakeCluster


class TestClusterWatchSemantics:
    def test_list_with_meta_returns_anchorable_rv(self):
        c = FakeCluster()
        c.create(node("n1"))
        items, rv = c.list_with_meta("v1", "Node")
        assert len(items) == 1 and rv == c.current_rv()
        # nothing happened since the LIST: a watch from rv sees only new events
        w = c.watch("v1", "Node", resource_version=rv)
        assert w.next(0.05) is None
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"x": "1"}}})
        ev = w.next(0.5)
        assert ev[0] == "MODIFIED" and ev[1]["metadata"]["labels"] == {"x": "1"}

    def test_resume_replays_events_after_rv(self):
        c = FakeCluster()
        c.create(node("n1"))
        _, rv = c.list_with_meta("v1", "Node")
        c.watch("v1", "Node")  # turns history recording on
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": "1"}}})
        c.delete("v1", "Node", "n1")
        w = c.watch("v1", "Node", resource_version=rv)
        assert w.next(0.5)[0] == "MODIFIED"
        assert w.next(0.5)[0] == "DELETED"
        assert w.next(0.05) is None

    def test_deletion_allocates_fresh_rv(self):
        # without this, an RV-anchored resume would skip deletes whose
        # object RV predates the anchor
        c = FakeCluster()
        c.create(node("n1"))
        rv_before = int(c.current_rv())
        c.delete("v1", "Node", "n1")
        assert int(c.current_rv()) > rv_before

    def test_resume_before_history_window_raises_gone(self):
        c = FakeCluster()
        c.create(node("n1"))
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": "1"}}})
        # history starts only when the first watch opens; rv=1 predates it
        c.watch("v1", "Node")
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": "2"}}})
        with pytest.raises(GoneError):
            c.watch("v1", "Node", resource_version="1")

    def test_history_ring_eviction_raises_gone(self):
        c = FakeCluster()
        c.WATCH_HISTORY = 4  # shrink the ring for the test
        c._history.clear(), c._history_start.clear()
        c.create(node("n1"))
        _, rv = c.list_with_meta("v1", "Node")
        c.watch("v1", "Node")
        for i in range(10):  # overflow the 4-event ring
            c.patch("v1", "Node", "n1", {"metadata": {"labels": {"i": str(i)}}})
        with pytest.raises(GoneError):
            c.watch("v1", "Node", resource_version=rv)

    def test_rv_zero_sends_synthetic_addeds(self):
        c = FakeCluster()
        c.create(node("n1"))
        c.create(node("n2"))
        w = c.watch("v1", "Node", resource_version="0")
        got = {w.next(0.5)[1]["metadata"]["name"] for _ in range(2)}
        assert got == {"n1", "n2"}

    def test_namespace_scoped_watch(self):
        c = FakeCluster()
        w = c.watch("v1", "Pod", namespace="ns-a")
        c.create(pod("p1", ns="ns-a"))
        c.create(pod("p2", ns="ns-b"))
        ev = w.next(0.5)
        assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "p1"
        assert w.next(0.05) is None  # ns-b event filtered out

    def test_selector_watch_stops_matching_becomes_deleted(self):
        c = FakeCluster()
        w = c.watch("v1", "Node", label_selector="role=gpu")
        c.create(node("n1", {"role": "gpu"}))
        assert w.next(0.5)[0] == "ADDED"
        # label removed -> object stops matching -> DELETED (client-go contract)
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"role": "cpu"}}})
        assert w.next(0.5)[0] == "DELETED"
        # starts matching again -> ADDED, not MODIFIED
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"role": "gpu"}}})
        assert w.next(0.5)[0] == "ADDED"

    def test_bookmark_rv_only_when_drained(self):
        c = FakeCluster()
        w = c.watch("v1", "Node")
        c.create(node("n1"))
        assert w.bookmark_rv() is None  # event pending: no safe bookmark
        w.next(0.5)
        assert w.bookmark_rv() == c.current_rv()


# ------------------------------------------------------------------ informer


class TestInformerReflector:
    def test_informer_resumes_from_list_rv(self):
        cluster = FakeCluster()
        cluster.create(node("n1"))
        cached = CachedClient(FakeClient(cluster))
        assert cached.get("v1", "Node", "n1")["metadata"]["name"] == "n1"
        inf = cached._informers[("v1", "Node")]
        assert inf._last_rv is not None

    def test_informer_survives_gone_with_relist(self):
        cluster = FakeCluster()
        cluster.create(node("n1"))
        cached = CachedClient(FakeClient(cluster))
        cached.get("v1", "Node", "n1")
        inf = cached._informers[("v1", "Node")]
        # kill the watch and poison the resume point so reconnect gets 410
        inf._watch.stop()
        inf._last_rv = "0"
        inf._last_rv = None  # force the legacy relist path too
        cluster.create(node("n2"))
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            try:
                if cached.get("v1", "Node", "n2"):
                    break
            except Exception:
                time.sleep(0.05)
        assert cached.get("v1", "Node", "n2")["metadata"]["name"] == "n2"
        cached.stop()

    def test_informer_no_events_lost_across_reconnect(self):
        cluster = FakeCluster()
        cluster.create(node("n1"))
        cached = CachedClient(FakeClient(cluster))
        cached.get("v1", "Node", "n1")
        inf = cached._informers[("v1", "Node")]
        # drop the stream, then mutate BEFORE the informer reconnects:
        # the RV anchor must replay the missed event
        inf._watch.stop()
        cluster.patch("v1", "Node", "n1", {"metadata": {"labels": {"k": "v"}}})
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            if cached.get("v1", "Node", "n1")["metadata"].get("labels", {}).get("k") == "v":
                break
            time.sleep(0.02)
        assert cached.get("v1", "Node", "n1")["metadata"]["labels"]["k"] == "v"
        cached.stop()


# ------------------------------------------------------------------ wire level


@pytest.fixture(scope="module")
def server():
    handle = start_apiserver()
    yield handle
    handle.stop()


@pytest.fixture
def rest(server):
    client = RestClient(server.url)
    yield client
    with server.cluster._lock:
        server.cluster._store.clear()
    client.close()


class TestWireProtocol:
    def test_list_response_carries_rv(self, rest, server):
        rest.create(node("w1"))
        items, rv = rest.list_with_meta("v1", "Node")
        assert len(items) == 1
        assert rv == server.cluster.current_rv()

    def test_watch_anchored_at_list_rv_over_http(self, rest, server):
        rest.create(node("w1"))
        _, rv = rest.list_with_meta("v1", "Node")
        w = rest.watch("v1", "Node", resource_version=rv)
        try:
            rest.patch("v1", "Node", "w1", {"metadata": {"labels": {"a": "1"}}})
            ev = w.next(5.0)
            assert ev[0] == "MODIFIED" and ev[1]["metadata"]["labels"] == {"a": "1"}
        finally:
            w.stop()

    def test_watch_connects_without_handshake_frames(self, rest, server):
        # raw wire check: no frame arrives until a real event happens —
        # the round-1 custom BOOKMARK handshake is gone
        rest.create(node("w1"))
        _, rv = rest.list_with_meta("v1", "Node")
        got = []

        def read():
            with httpx.stream(
                "GET", f"{server.url}/api/v1/nodes",
                params={"watch": "true", "resourceVersion": rv}, timeout=5,
            ) as resp:
                for line in resp.iter_lines():
                    if line.strip():
                        got.append(json.loads(line))
                        return

        t = threading.Thread(target=read, daemon=True)
        t.start()
        time.sleep(0.3)
        assert got == []  # nothing until an actual event
        rest.patch("v1", "Node", "w1", {"metadata": {"labels": {"z": "9"}}})
        t.join(5)
        assert got and got[0]["type"] == "MODIFIED"

    def test_expired_rv_streams_error_410(self, server, rest):
        rest.create(node("w1"))
        # turn on history, then ask for an RV before it began
        rest.watch("v1", "Node").stop()
        rest.patch("v1", "Node", "w1", {"metadata": {"labels": {"a": "1"}}})
        with httpx.stream(
            "GET", f"{server.url}/api/v1/nodes",
            params={"watch": "true", "resourceVersion": "1"}, timeout=5,
        ) as resp:
            assert resp.status_code == 200  # real apiservers answer 200...
            line = next(l for l in resp.iter_lines() if l.strip())
            ev = json.loads(line)
        # ...and stream the 410 as an in-band ERROR Status event
        assert ev["type"] == "ERROR"
        assert ev["object"]["code"] == 410
        assert ev["object"]["reason"] == "Expired"

    def test_bookmarks_when_requested(self, rest, server):
        rest.create(node("w1"))
        _, rv = rest.list_with_meta("v1", "Node")
        with httpx.stream(
            "GET", f"{server.url}/api/v1/nodes",
            params={"watch": "true", "resourceVersion": rv,
                    "allowWatchBookmarks": "true"}, timeout=5,
        ) as resp:
            line = next(l for l in resp.iter_lines() if l.strip())
            ev = json.loads(line)
        assert ev["type"] == "BOOKMARK"
        assert ev["object"]["metadata"]["resourceVersion"] == rv

    def test_namespace_scoped_watch_over_http(self, rest, server):
        w = rest.watch("v1", "Pod", namespace="ns-a")
        try:
            rest.create(pod("p1", ns="ns-a"))
            rest.create(pod("p2", ns="ns-b"))
            ev = w.next(5.0)
            assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "p1"
            assert w.next(0.3) is None
        finally:
            w.stop()

    def test_selector_watch_over_http(self, rest, server):
        w = rest.watch("v1", "Node", label_selector="role=gpu")
        try:
            rest.create(node("g1", {"role": "gpu"}))
            rest.create(node("c1", {"role": "cpu"}))
            ev = w.next(5.0)
            assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "g1"
            rest.patch("v1", "Node", "g1", {"metadata": {"labels": {"role": "cpu"}}})
            assert w.next(5.0)[0] == "DELETED"
        finally:
            w.stop()

    def test_watch_timeout_seconds_closes_stream(self, server, rest):
        rest.create(node("w1"))
        t0 = time.monotonic()
        with httpx.stream(
            "GET", f"{server.url}/api/v1/nodes",
            params={"watch": "true", "timeoutSeconds": "1"}, timeout=10,
        ) as resp:
            for _ in resp.iter_lines():
                pass
        assert time.monotonic() - t0 < 5

    def test_informer_stack_over_http_survives_reconnect(self, rest, server):
        rest.create(node("w1"))
        cached = CachedClient(rest)
        try:
            assert cached.get("v1", "Node", "w1")
            inf = cached._informers[("v1", "Node")]
            inf._watch.stop()
            rest.patch("v1", "Node", "w1", {"metadata": {"labels": {"r": "1"}}})
            deadline = time.monotonic() + 10
            while time.monotonic() < deadline:
                if cached.get("v1", "Node", "w1")["metadata"].get(
                        "labels", {}).get("r") == "1":
                    break
                time.sleep(0.05)
            assert cached.get("v1", "Node", "w1")["metadata"]["labels"]["r"] == "1"
        finally:
            cached.stop()

    def test_plural_resolution(self, rest):
        from k8s_operator_libs_amd.core.restclient import _lower_plural

        assert _lower_plural("NetworkPolicy") == "networkpolicies"
        assert _lower_plural("Ingress") == "ingresses"
        assert _lower_plural("Endpoints") == "endpoints"
        assert _lower_plural("Gateway") == "gateways"  # vowel+y
        assert _lower_plural("Node") == "nodes"

    def test_unregistered_kind_resolved_via_discovery(self, rest, server):
        # PodDisruptionBudget is served by the cluster but absent from the
        # RestClient's static registry: discovery must resolve its plural
        # (naive 's'-append would give 'poddisruptionbudgets' here too, so
        # assert the discovery result is marked namespaced as served)
        assert ("policy/v1", "PodDisruptionBudget") not in rest._kinds
        rest.create({
            "apiVersion": "policy/v1", "kind": "PodDisruptionBudget",
            "metadata": {"name": "pdb1", "namespace": "default"},
            "spec": {"minAvailable": 1, "selector": {"matchLabels": {}}},
        })
        got = rest.get("policy/v1", "PodDisruptionBudget", "pdb1", "default")
        assert got["metadata"]["name"] == "pdb1"
        assert rest._kinds[("policy/v1", "PodDisruptionBudget")] == (
            "poddisruptionbudgets", True)


def test_field_selector_watch():
    """Field-selector watches (spec.nodeName) with the same
    transform semantics as label selectors."""
    c = FakeCluster()
    w = c.watch("v1", "Pod", field_selector="spec.nodeName=n1")
    c.create(pod("p1"))          # nodeName n1 (pod() default)
    ev = w.next(0.5)
    assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "p1"
    c.create({"apiVersion": "v1", "kind": "Pod",
              "metadata": {"name": "p2", "namespace": "default"},
              "spec": {"nodeName": "other"}})
    assert w.next(0.1) is None  # other node filtered out
    w.stop()


def test_field_selector_watch_over_http(rest, server):
    w = rest.watch("v1", "Pod", field_selector="spec.nodeName=n1")
    try:
        rest.create(pod("f1"))
        ev = w.next(5.0)
        assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "f1"
    finally:
        w.stop()


class TestUvicornEngineParity:
    """The FastAPI/uvicorn engine stays wire-compatible with the default
    threaded engine (both serve the same protocol; the threaded one is the
    benchmark substrate)."""

    @pytest.fixture(scope="class")
    def uv(self):
        handle = start_apiserver(engine="uvicorn")
        client = RestClient(handle.url)
        yield client, handle
        client.close()
        handle.stop()

    def test_crud_and_rv_watch(self, uv):
        rest, server = uv
        rest.create(node("uv1"))
        items, rv = rest.list_with_meta("v1", "Node")
        assert rv and len(items) == 1
        w = rest.watch("v1", "Node", resource_version=rv)
        try:
            rest.patch("v1", "Node", "uv1", {"metadata": {"labels": {"u": "1"}}})
            ev = w.next(5.0)
            while ev and ev[0] in ("BOOKMARK",):
                ev = w.next(5.0)
            assert ev[0] == "MODIFIED"
        finally:
            w.stop()

    def test_pagination_and_eviction(self, uv):
        rest, server = uv
        for i in range(3):
            rest.create({"apiVersion": "v1", "kind": "Pod",
                         "metadata": {"name": f"uvp-{i}", "namespace": "default"},
                         "spec": {"containers": [{"name": "c", "image": "x"}]}})
        old = rest.LIST_PAGE_SIZE
        try:
            rest.LIST_PAGE_SIZE = 2
            items, _ = rest.list_with_meta("v1", "Pod", namespace="default")
            assert len(items) == 3
        finally:
            rest.LIST_PAGE_SIZE = old
        rest.evict_pod("uvp-0", "default")
        with pytest.raises(Exception):
            rest.get("v1", "Pod", "uvp-0", "default")


def test_informer_recovers_from_in_stream_410_over_http(rest, server):
    """Over the wire: the informer's watch drops, the resume window has
    been evicted (tiny history ring), reconnect gets the in-stream
    ERROR-410 — the informer must relist and converge."""
    cluster = server.cluster
    old_hist = cluster.WATCH_HISTORY
    rest.create(node("w410"))
    cached = CachedClient(rest)
    try:
        assert cached.get("v1", "Node", "w410")
        inf = cached._informers[("v1", "Node")]
        # shrink the ring, then push enough events to evict the resume point
        with cluster._lock:
            import collections as _c

            key = ("v1", "Node")
            cluster._history[key] = _c.deque(cluster._history.get(key, ()),
                                             maxlen=4)
        inf._watch.stop()
        for i in range(10):
            rest.patch("v1", "Node", "w410",
                       {"metadata": {"labels": {"i": str(i)}}})
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            if cached.get("v1", "Node", "w410")["metadata"].get(
                    "labels", {}).get("i") == "9":
                break
            time.sleep(0.05)
        assert cached.get("v1", "Node", "w410")["metadata"]["labels"]["i"] == "9"
    finally:
        cached.stop()
        cluster.WATCH_HISTORY = old_hist


class TestWatchListProtocol:
    """KEP-3157 WatchList (sendInitialEvents, beta in K8s 1.32): initial
    state streams through the watch, ending with a BOOKMARK annotated
    k8s.io/initial-events-end; informers bootstrap without a LIST."""

    def test_cluster_watchlist_semantics(self):
        c = FakeCluster()
        c.create(node("n1", {"a": "1"}))
        c.create(node("n2"))
        w = c.watch("v1", "Node", send_initial_events=True)
        got = {}
        end_rv = None
        for _ in range(10):
            etype, obj = w.next(0.5)
            if etype == "BOOKMARK":
                ann = obj["metadata"].get("annotations") or {}
                assert ann.get("k8s.io/initial-events-end") == "true"
                end_rv = obj["metadata"]["resourceVersion"]
                break
            assert etype == "ADDED"
            got[obj["metadata"]["name"]] = obj
        assert set(got) == {"n1", "n2"}
        assert end_rv == c.current_rv()
        # stream continues live past the marker, losslessly
        c.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": "2"}}})
        ev = w.next(0.5)
        assert ev[0] == "MODIFIED" and ev[1]["metadata"]["labels"]["a"] == "2"
        w.stop()

    def test_watchlist_over_http(self, rest, server):
        rest.create(node("wl1"))
        w = rest.watch("v1", "Node", send_initial_events=True)
        try:
            seen_added = False
            for _ in range(10):
                etype, obj = w.next(5.0)
                if etype == "ADDED" and obj["metadata"]["name"] == "wl1":
                    seen_added = True
                if etype == "BOOKMARK" and (obj["metadata"].get("annotations")
                                            or {}).get("k8s.io/initial-events-end"):
                    break
            assert seen_added
        finally:
            w.stop()

    def test_informer_bootstraps_via_watchlist(self, rest, server):
        rest.create(node("wl2"))
        calls = {"list": 0}
        orig = rest.list_with_meta

        def counting(*a, **kw):
            calls["list"] += 1
            return orig(*a, **kw)

        rest.list_with_meta = counting
        cached = CachedClient(rest)
        try:
            assert cached.get("v1", "Node", "wl2")["metadata"]["name"] == "wl2"
            assert calls["list"] == 0, "informer LISTed despite WatchList"
            inf = cached._informers[("v1", "Node")]
            assert inf._last_rv is not None
            # live updates flow on the same stream
            rest.patch("v1", "Node", "wl2", {"metadata": {"labels": {"x": "1"}}})
            deadline = time.monotonic() + 5
            while time.monotonic() < deadline:
                if cached.get("v1", "Node", "wl2")["metadata"].get(
                        "labels", {}).get("x") == "1":
                    break
                time.sleep(0.02)
            assert cached.get("v1", "Node", "wl2")["metadata"]["labels"]["x"] == "1"
        finally:
            rest.list_with_meta = orig
            cached.stop()

    def test_informer_falls_back_when_server_lacks_watchlist(self):
        """A delegate that rejects sendInitialEvents (pre-1.27 apiserver)
        still syncs via LIST-then-WATCH."""
        cluster = FakeCluster()
        cluster.create(node("fb1"))
        inner = FakeClient(cluster)

        class NoWatchList(FakeClient):
            def __init__(self):
                self.cluster = cluster

            def watch(self, api_version, kind, namespace=None,
                      resource_version=None, label_selector="",
                      field_selector="", send_initial_events=False):
                if send_initial_events:
                    raise RuntimeError("sendInitialEvents is not allowed")
                return inner.watch(api_version, kind, namespace=namespace,
                                   resource_version=resource_version,
                                   label_selector=label_selector,
                                   field_selector=field_selector)

        cached = CachedClient(NoWatchList())
        try:
            assert cached.get("v1", "Node", "fb1")["metadata"]["name"] == "fb1"
        finally:
            cached.stop()
