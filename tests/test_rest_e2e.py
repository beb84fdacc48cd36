"""Wire-level e2e: RestClient over HTTP against the mini-apiserver.

This is the envtest analogue at full fidelity: the same upgrade state machine
that runs against the in-process FakeClient here talks real REST (httpx ->
uvicorn -> FakeCluster), covering path construction, selectors, merge
patches, optimistic locking, the eviction subresource and discovery."""

import pytest

from k8s_operator_libs_amd.core.apiserver import start_apiserver
from k8s_operator_libs_amd.core.errors import ConflictError, NotFoundError
from k8s_operator_libs_amd.core.restclient import RestClient
from k8s_operator_libs_amd.crdutil import CRD_OPERATION_APPLY, process_crds
from k8s_operator_libs_amd.upgrade import consts
from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
from k8s_operator_libs_amd.upgrade.state_manager import ClusterUpgradeStateManager

from builders import DRIVER_LABELS, DRIVER_NS
from simenv import SimDaemonSetController
from test_state_manager import policy, setup_cluster, state_of


@pytest.fixture(scope="module")
def server():
    handle = start_apiserver()
    yield handle
    handle.stop()


@pytest.fixture
def rest(server):
    client = RestClient(server.url)
    yield client
    # wipe all objects between tests
    with server.cluster._lock:
        server.cluster._store.clear()
    client.close()


class FakeBackedClients:
    """Both views of the same cluster: REST and direct."""


def test_crud_roundtrip(rest):
    rest.create({"apiVersion": "v1", "kind": "Node",
                 "metadata": {"name": "n1", "labels": {"a": "1"}}, "spec": {}})
    node = rest.get_node("n1")
    assert node["metadata"]["labels"]["a"] == "1"
    rest.patch("v1", "Node", "n1", {"metadata": {"labels": {"b": "2"}}})
    assert rest.get_node("n1")["metadata"]["labels"] == {"a": "1", "b": "2"}
    node = rest.get_node("n1")
    node["metadata"]["labels"]["c"] = "3"
    rest.update(node)
    assert rest.get_node("n1")["metadata"]["labels"]["c"] == "3"
    rest.delete("v1", "Node", "n1")
    with pytest.raises(NotFoundError):
        rest.get_node("n1")


def test_conflict_on_stale_update(rest):
    rest.create({"apiVersion": "v1", "kind": "Node",
                 "metadata": {"name": "n1"}, "spec": {}})
    a = rest.get_node("n1")
    b = rest.get_node("n1")
    a["metadata"]["labels"] = {"x": "1"}
    rest.update(a)
    b["metadata"]["labels"] = {"x": "2"}
    with pytest.raises(ConflictError):
        rest.update(b)


def test_selectors_over_wire(rest):
    for i, app in enumerate(["train", "serve"]):
        rest.create({"apiVersion": "v1", "kind": "Pod",
                     "metadata": {"name": f"p{i}", "namespace": "default",
                                  "labels": {"app": app}},
                     "spec": {"nodeName": f"n{i}"}})
    assert len(rest.list_pods(label_selector="app=train")) == 1
    assert len(rest.list_pods(field_selector="spec.nodeName=n1")) == 1
    assert len(rest.list_pods(namespace="default")) == 2


def test_eviction_subresource(rest):
    rest.create({"apiVersion": "v1", "kind": "Pod",
                 "metadata": {"name": "victim", "namespace": "default"},
                 "spec": {}})
    rest.evict_pod("victim", "default")
    with pytest.raises(NotFoundError):
        rest.get("v1", "Pod", "victim", "default")


def test_crdutil_over_wire(rest, tmp_path):
    crd = tmp_path / "crd.yaml"
    crd.write_text(
        "apiVersion: apiextensions.k8s.io/v1\n"
        "kind: CustomResourceDefinition\n"
        "metadata:\n  name: things.amd.com\n"
        "spec:\n  group: amd.com\n  scope: Namespaced\n"
        "  names: {kind: Thing, plural: things}\n"
        "  versions:\n    - {name: v1, served: true, storage: true}\n"
    )
    n = process_crds(rest, [str(crd)], CRD_OPERATION_APPLY)
    assert n == 1
    # discovery now serves the new resource and CRUD works over REST
    assert rest.discover_resource("amd.com/v1", "things")
    rest.create({"apiVersion": "amd.com/v1", "kind": "Thing",
                 "metadata": {"name": "t1", "namespace": "default"}})
    assert rest.get("amd.com/v1", "Thing", "t1", "default")


def test_full_upgrade_lifecycle_over_rest(rest, server):
    """The flagship path, wire-level: the whole state machine driven through
    HTTP only (BASELINE config #2 over REST)."""
    # build the synthetic cluster through the REST client itself
    class RestWrapper:
        cluster = server.cluster  # setup_cluster uses client.cluster.create

    ds, _ = setup_cluster(RestWrapper, pod_hash="old", ds_hash="new")
    SimDaemonSetController(server.cluster, ds, current_hash="new")
    manager = ClusterUpgradeStateManager(rest).with_pod_deletion_enabled(
        gpu_pod_deletion_filter
    )
    pol = policy(maxParallelUpgrades=1, maxUnavailable="100%",
                 drainSpec={"enable": True})
    for _ in range(12):
        state = manager.build_state(DRIVER_NS, DRIVER_LABELS)
        manager.apply_state(state, pol)
        manager.wait_idle()
        if state_of(rest, "node-0") == consts.UPGRADE_STATE_DONE:
            break
    assert state_of(rest, "node-0") == consts.UPGRADE_STATE_DONE
    pods = rest.list_pods(namespace=DRIVER_NS)
    assert pods[0]["metadata"]["labels"]["controller-revision-hash"] == "new"


def test_http_watch_stream(rest, server):
    w = rest.watch("v1", "Node")
    import time
    time.sleep(0.2)  # let the stream connect
    server.cluster.create({"apiVersion": "v1", "kind": "Node",
                           "metadata": {"name": "wn1"}, "spec": {}})
    server.cluster.patch("v1", "Node", "wn1", {"metadata": {"labels": {"s": "1"}}})
    events = []
    for _ in range(2):
        item = w.next(timeout=5)
        assert item is not None, "watch event not delivered"
        events.append(item[0])
    assert events == ["ADDED", "MODIFIED"]
    w.stop()


def test_cached_client_over_rest(rest, server):
    """Informer cache fed by the HTTP watch stream: the production
    architecture end-to-end."""
    import time

    from k8s_operator_libs_amd.core.cache import CachedClient

    server.cluster.create({"apiVersion": "v1", "kind": "Node",
                           "metadata": {"name": "cn1"}, "spec": {}})
    cached = CachedClient(rest)
    try:
        assert cached.get_node("cn1")["metadata"]["name"] == "cn1"
        server.cluster.patch("v1", "Node", "cn1",
                             {"metadata": {"labels": {"x": "1"}}})
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            if cached.get_node("cn1")["metadata"].get("labels", {}).get("x") == "1":
                break
            time.sleep(0.01)
        assert cached.get_node("cn1")["metadata"]["labels"]["x"] == "1"
    finally:
        cached.stop()


def test_concurrent_rest_clients(rest, server):
    """Thread-safety of the HTTP apiserver under concurrent writers."""
    import threading

    errors = []

    def worker(i):
        try:
            c = RestClient(server.url)
            for j in range(20):
                name = f"cw-{i}-{j}"
                c.create({"apiVersion": "v1", "kind": "Pod",
                          "metadata": {"name": name, "namespace": "default"},
                          "spec": {"nodeName": f"n{i}"}})
                c.patch("v1", "Pod", name,
                        {"metadata": {"labels": {"round": str(j)}}}, "default")
            c.close()
        except Exception as exc:
            errors.append(exc)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors
    assert len(rest.list_pods(namespace="default")) == 80
    # per-node index stayed consistent under concurrency
    assert len(rest.list_pods(field_selector="spec.nodeName=n2")) == 20


def test_requestor_mode_over_rest(rest, server):
    """Full requestor-mode lifecycle through HTTP: NodeMaintenance CRUD on
    the named API group, shared finalizer flow, node completion."""
    from k8s_operator_libs_amd.upgrade.requestor import RequestorOptions
    from k8s_operator_libs_amd.upgrade.state_manager import StateOptions
    from simenv import SimMaintenanceOperator

    class W:
        cluster = server.cluster

    ds, _ = setup_cluster(W, pod_hash="old", ds_hash="new")
    SimDaemonSetController(server.cluster, ds, current_hash="new")
    SimMaintenanceOperator(server.cluster)
    manager = ClusterUpgradeStateManager(
        rest,
        options=StateOptions(requestor=RequestorOptions(
            use_maintenance_operator=True,
            requestor_id="amd.gpu.operator",
            namespace="default",
        )),
    )
    pol = policy(drainSpec={"enable": True})
    for _ in range(12):
        manager.reconcile(DRIVER_NS, DRIVER_LABELS, pol)
        if state_of(rest, "node-0") == consts.UPGRADE_STATE_DONE:
            break
    assert state_of(rest, "node-0") == consts.UPGRADE_STATE_DONE
    with pytest.raises(NotFoundError):
        rest.get("maintenance.amd.com/v1alpha1", "NodeMaintenance",
                 "amd-operator-node-0", "default")


def test_status_subresource(rest, server):
    rest.create({"apiVersion": "maintenance.amd.com/v1alpha1",
                 "kind": "NodeMaintenance",
                 "metadata": {"name": "nm1", "namespace": "default"},
                 "spec": {"nodeName": "n1", "requestorID": "op"}})
    rest.patch_status("maintenance.amd.com/v1alpha1", "NodeMaintenance", "nm1",
                      {"conditions": [{"type": "Ready", "status": "True",
                                       "reason": "Ready"}]}, "default")
    live = rest.get("maintenance.amd.com/v1alpha1", "NodeMaintenance", "nm1", "default")
    assert live["status"]["conditions"][0]["reason"] == "Ready"
    # spec untouched by the status patch
    assert live["spec"]["requestorID"] == "op"


def test_from_environment_kubernetes_master(server, monkeypatch):
    monkeypatch.delenv("KUBECONFIG", raising=False)
    monkeypatch.setenv("KUBERNETES_MASTER", server.url)
    c = RestClient.from_environment()
    try:
        c.create({"apiVersion": "v1", "kind": "Node",
                  "metadata": {"name": "envn"}, "spec": {}})
        assert c.get_node("envn")["metadata"]["name"] == "envn"
    finally:
        c.close()


def test_from_environment_kubeconfig(server, tmp_path, monkeypatch):
    kc = tmp_path / "kubeconfig"
    kc.write_text(f"""
apiVersion: v1
kind: Config
current-context: test
contexts:
  - name: test
    context: {{cluster: c1, user: u1}}
clusters:
  - name: c1
    cluster: {{server: "{server.url}", insecure-skip-tls-verify: true}}
users:
  - name: u1
    user: {{token: dummy-token}}
""")
    monkeypatch.setenv("KUBECONFIG", str(kc))
    monkeypatch.delenv("KUBERNETES_MASTER", raising=False)
    c = RestClient.from_environment()
    try:
        assert c._http.headers["Authorization"] == "Bearer dummy-token"
        c.create({"apiVersion": "v1", "kind": "Node",
                  "metadata": {"name": "kcn"}, "spec": {}})
        assert c.get_node("kcn")["metadata"]["name"] == "kcn"
    finally:
        c.close()


def test_from_environment_nothing_configured(monkeypatch):
    monkeypatch.delenv("KUBECONFIG", raising=False)
    monkeypatch.delenv("KUBERNETES_MASTER", raising=False)
    with pytest.raises(RuntimeError):
        RestClient.from_environment()


def test_full_upgrade_through_informer_stack(rest, server):
    """The complete production architecture in one test: state machine ->
    CachedClient (watch-fed informers) -> RestClient -> HTTP -> apiserver.
    Reads come from the cache, writes go through REST, the provider barrier
    holds transitions until the informers converge."""
    import time

    from k8s_operator_libs_amd.core.cache import CachedClient

    class W:
        cluster = server.cluster

    ds, _ = setup_cluster(W, pod_hash="old", ds_hash="new")
    SimDaemonSetController(server.cluster, ds, current_hash="new")
    cached = CachedClient(rest)
    try:
        manager = ClusterUpgradeStateManager(cached).with_pod_deletion_enabled(
            gpu_pod_deletion_filter
        )
        pol = policy(maxParallelUpgrades=1, maxUnavailable="100%",
                     drainSpec={"enable": True})
        for _ in range(25):
            manager.reconcile(DRIVER_NS, DRIVER_LABELS, pol)
            if state_of(rest, "node-0") == consts.UPGRADE_STATE_DONE:
                break
            time.sleep(0.1)  # informer requeue interval
        assert state_of(rest, "node-0") == consts.UPGRADE_STATE_DONE
        # every label transition fired exactly once despite the async cache
        for (frm, to), count in manager.metrics.state_transitions.items().items():
            assert count == 1, f"{frm}->{to} fired {count} times"
    finally:
        cached.stop()


def test_from_environment_service_account(server, tmp_path, monkeypatch):
    """In-cluster resolution: service-account token + KUBERNETES_SERVICE_*
    (restclient.py from_environment path 2).  The URL is https (in-cluster
    is always TLS) so this pins resolution, not connectivity."""
    import k8s_operator_libs_amd.core.restclient as rc

    sa = tmp_path / "serviceaccount"
    sa.mkdir()
    (sa / "token").write_text("sa-token-123\n")
    # no ca.crt: a PEM-invalid CA would fail httpx's SSL context load;
    # the resolver falls back to verify=True
    monkeypatch.setattr(rc, "SA_DIR", str(sa))
    monkeypatch.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    monkeypatch.setenv("KUBERNETES_SERVICE_PORT", "6443")
    client = rc.RestClient.from_environment()
    try:
        assert client.base_url == "https://10.0.0.1:6443"
        assert client._headers["Authorization"] == "Bearer sa-token-123"
        assert client._fast_netloc is None  # TLS: stays on httpx
    finally:
        client.close()


def test_transient_5xx_retried(monkeypatch):
    """client-go-style retry: 503s from a briefly-unhealthy apiserver are
    retried with backoff; the request ultimately succeeds (restclient.py
    _request).  Pinned with a raw socket server so both transports (fast
    http.client and httpx) see identical wire bytes."""
    import http.server
    import json as _json
    import socketserver
    import threading

    state = {"fails": 2, "requests": 0}

    class Handler(http.server.BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, *a):
            pass

        def do_GET(self):
            state["requests"] += 1
            if state["fails"] > 0:
                state["fails"] -= 1
                body = b'{"kind":"Status","message":"apiserver warming up"}'
                self.send_response(503)
            else:
                body = _json.dumps({"apiVersion": "v1", "kind": "Node",
                                    "metadata": {"name": "n1"}}).encode()
                self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    class Srv(socketserver.ThreadingMixIn, http.server.HTTPServer):
        daemon_threads = True
        allow_reuse_address = True

    srv = Srv(("127.0.0.1", 0), Handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        client = RestClient(f"http://127.0.0.1:{srv.server_address[1]}")
        node = client.get_node("n1")
        assert node["metadata"]["name"] == "n1"
        assert state["requests"] == 3  # 2 failures + 1 success
        client.close()
    finally:
        srv.shutdown()
        srv.server_close()


def test_retries_exhausted_surface_api_error(monkeypatch):
    import http.server
    import socketserver
    import threading

    import pytest as _pytest

    from k8s_operator_libs_amd.core.errors import ApiError

    class Handler(http.server.BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, *a):
            pass

        def do_GET(self):
            body = b'{"kind":"Status","message":"still down"}'
            self.send_response(503)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    class Srv(socketserver.ThreadingMixIn, http.server.HTTPServer):
        daemon_threads = True
        allow_reuse_address = True

    srv = Srv(("127.0.0.1", 0), Handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        client = RestClient(f"http://127.0.0.1:{srv.server_address[1]}",
                            retries=1)
        with _pytest.raises(ApiError) as exc:
            client.get_node("n1")
        assert exc.value.code == 503
        client.close()
    finally:
        srv.shutdown()
        srv.server_close()
