"""Mini kube-apiserver: serve a FakeCluster over HTTP.

The wire-level test substrate (our envtest's apiserver): a FastAPI app
exposing the Kubernetes REST surface the library uses — typed CRUD, JSON
merge patch, label/field selectors, the pod eviction subresource, and
API-group discovery — backed by a
:class:`~k8s_operator_libs_amd.core.fakecluster.FakeCluster`.  The
:class:`~k8s_operator_libs_amd.core.restclient.RestClient` is tested
end-to-end against it, and the example operator can run against it for local
development.
"""

import json
import threading
import time
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from .errors import ApiError
from .fakecluster import FakeCluster


def _status_body(exc: ApiError) -> dict:
    reason = {
        404: "NotFound",
        409: "Conflict",
        400: "BadRequest",
        429: "TooManyRequests",
    }.get(exc.code, "InternalError")
    if exc.code == 409 and "already exists" in exc.message:
        reason = "AlreadyExists"
    return {
        "kind": "Status",
        "apiVersion": "v1",
        "status": "Failure",
        "message": exc.message,
        "reason": reason,
        "code": exc.code,
    }


def create_app(cluster: Optional[FakeCluster] = None):
    """Build the FastAPI app; returns (app, cluster)."""
    cluster = cluster or FakeCluster()
    app = FastAPI(title="amd-k8s-mini-apiserver")

    def _resolve(group: str, version: str, plural: str):
        api_version = version if not group else f"{group}/{version}"
        kind = cluster.lookup_by_plural(api_version, plural)
        if kind is None:
            raise ApiError(f"resource {plural} not served in {api_version}")
        return api_version, kind

    def _handle(fn):
        try:
            return fn()
        except ApiError as exc:
            return JSONResponse(_status_body(exc), status_code=exc.code)

    # -- discovery -----------------------------------------------------------

    @app.get("/api/v1")
    def discovery_core():
        return _discovery("", "v1")

    @app.get("/apis/{group}/{version}")
    def discovery_group(group: str, version: str):
        return _discovery(group, version)

    def _discovery(group: str, version: str):
        api_version = version if not group else f"{group}/{version}"
        resources = []
        for (av, kind), (plural, namespaced) in cluster._kinds.items():
            if av == api_version:
                resources.append(
                    {"name": plural, "kind": kind, "namespaced": namespaced}
                )
        return {"kind": "APIResourceList", "groupVersion": api_version,
                "resources": resources}

    # -- collection routes ---------------------------------------------------

    async def _list(group, version, plural, request: Request, namespace: str = ""):
        if request.query_params.get("watch") in ("true", "1"):
            return _watch_stream(group, version, plural, request, namespace)

        def run():
            api_version, kind = _resolve(group, version, plural)
            qp = request.query_params
            try:
                limit = int(qp.get("limit", "0"))
            except ValueError:
                limit = 0
            items, rv, next_token = cluster.list_paged(
                api_version, kind,
                namespace=namespace or None,
                label_selector=qp.get("labelSelector", ""),
                field_selector=qp.get("fieldSelector", ""),
                limit=limit,
                continue_token=qp.get("continue", ""),
            )
            md = {"resourceVersion": rv}
            if next_token:
                md["continue"] = next_token
            return {
                "kind": f"{kind}List",
                "apiVersion": api_version,
                "metadata": md,
                "items": items,
            }
        return _handle(run)

    def _watch_stream(group, version, plural, request: Request, namespace: str = ""):
        """Kubernetes watch protocol: newline-delimited WatchEvent JSON.

        Semantics match a real apiserver (no custom handshake frames):

        - ``resourceVersion`` anchors the stream; an expired RV is signalled
          as an in-stream ``ERROR`` event carrying a 410 Status (the wire
          shape client-go expects), after which the stream closes;
        - ``allowWatchBookmarks=true`` opts into periodic ``BOOKMARK``
          events whose object carries only ``metadata.resourceVersion``;
        - ``timeoutSeconds`` closes the stream server-side when it elapses;
        - ``labelSelector`` filters events (stops-matching -> DELETED);
        - namespace-scoped paths watch only that namespace.
        """
        from fastapi.responses import StreamingResponse

        from .errors import GoneError

        qp = request.query_params
        try:
            api_version, kind = _resolve(group, version, plural)
        except ApiError as exc:
            return JSONResponse(_status_body(exc), status_code=exc.code)

        resource_version = qp.get("resourceVersion") or None
        label_selector = qp.get("labelSelector", "")
        bookmarks = qp.get("allowWatchBookmarks") in ("true", "1")
        try:
            timeout_s = float(qp.get("timeoutSeconds", "0")) or None
        except ValueError:
            timeout_s = None

        try:
            watch = cluster.watch(
                api_version, kind,
                namespace=namespace or None,
                resource_version=resource_version,
                label_selector=label_selector,
                field_selector=qp.get("fieldSelector", ""),
                send_initial_events=qp.get("sendInitialEvents")
                in ("true", "1"),
            )
        except GoneError as exc:
            # expired before the stream even opened: stream a single ERROR
            # event (a real apiserver answers 200 + in-stream ERROR Status)
            gone_message = exc.message

            def gone_gen():
                yield json.dumps({
                    "type": "ERROR",
                    "object": {
                        "kind": "Status", "apiVersion": "v1",
                        "status": "Failure", "reason": "Expired",
                        "message": gone_message, "code": 410,
                    },
                }) + "\n"
            return StreamingResponse(gone_gen(), media_type="application/json")
        except ApiError as exc:
            return JSONResponse(_status_body(exc), status_code=exc.code)

        def gen():
            deadline = time.monotonic() + timeout_s if timeout_s else None
            last_bookmark = None
            try:
                while True:
                    if deadline is not None and time.monotonic() >= deadline:
                        return
                    item = watch.next(timeout=0.5)
                    if item is None:
                        if bookmarks:
                            rv = watch.bookmark_rv()
                            if rv is not None and rv != last_bookmark:
                                last_bookmark = rv
                                yield json.dumps({
                                    "type": "BOOKMARK",
                                    "object": {
                                        "kind": kind,
                                        "apiVersion": api_version,
                                        "metadata": {"resourceVersion": rv},
                                    },
                                }) + "\n"
                                continue
                        yield ""  # keep-alive; also surfaces disconnects
                        continue
                    event_type, obj = item
                    yield json.dumps({"type": event_type, "object": obj}) + "\n"
            finally:
                watch.stop()

        return StreamingResponse(gen(), media_type="application/json")

    async def _create(group, version, plural, request: Request, namespace: str = ""):
        body = json.loads(await request.body())

        def run():
            _resolve(group, version, plural)
            if namespace:
                body.setdefault("metadata", {})["namespace"] = namespace
            created = cluster.create(body)
            return JSONResponse(created, status_code=201)
        return _handle(run)

    async def _get(group, version, plural, name, namespace: str = ""):
        def run():
            api_version, kind = _resolve(group, version, plural)
            return cluster.get(api_version, kind, name, namespace)
        return _handle(run)

    async def _put(group, version, plural, name, request: Request, namespace: str = ""):
        body = json.loads(await request.body())

        def run():
            api_version, kind = _resolve(group, version, plural)
            if namespace:
                body.setdefault("metadata", {})["namespace"] = namespace
            return cluster.update(body)
        return _handle(run)

    async def _patch(group, version, plural, name, request: Request, namespace: str = ""):
        body = json.loads(await request.body())

        def run():
            api_version, kind = _resolve(group, version, plural)
            return cluster.patch(api_version, kind, name, body, namespace)
        return _handle(run)

    async def _delete(group, version, plural, name, namespace: str = ""):
        def run():
            api_version, kind = _resolve(group, version, plural)
            cluster.delete(api_version, kind, name, namespace)
            return {"kind": "Status", "status": "Success"}
        return _handle(run)

    # core group, cluster-scoped + namespaced
    app.add_api_route("/api/{version}/{plural}", _wrap_nogroup(_list), methods=["GET"])
    app.add_api_route("/api/{version}/{plural}", _wrap_nogroup(_create), methods=["POST"])
    app.add_api_route("/api/{version}/{plural}/{name}", _wrap_nogroup(_get), methods=["GET"])
    app.add_api_route("/api/{version}/{plural}/{name}", _wrap_nogroup(_put), methods=["PUT"])
    app.add_api_route("/api/{version}/{plural}/{name}", _wrap_nogroup(_patch), methods=["PATCH"])
    app.add_api_route("/api/{version}/{plural}/{name}", _wrap_nogroup(_delete), methods=["DELETE"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}",
                      _wrap_nogroup(_list), methods=["GET"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}",
                      _wrap_nogroup(_create), methods=["POST"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}/{name}",
                      _wrap_nogroup(_get), methods=["GET"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}/{name}",
                      _wrap_nogroup(_put), methods=["PUT"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}/{name}",
                      _wrap_nogroup(_patch), methods=["PATCH"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}/{name}",
                      _wrap_nogroup(_delete), methods=["DELETE"])

    # named groups
    app.add_api_route("/apis/{group}/{version}/{plural}", _list, methods=["GET"])
    app.add_api_route("/apis/{group}/{version}/{plural}", _create, methods=["POST"])
    app.add_api_route("/apis/{group}/{version}/{plural}/{name}", _get, methods=["GET"])
    app.add_api_route("/apis/{group}/{version}/{plural}/{name}", _put, methods=["PUT"])
    app.add_api_route("/apis/{group}/{version}/{plural}/{name}", _patch, methods=["PATCH"])
    app.add_api_route("/apis/{group}/{version}/{plural}/{name}", _delete, methods=["DELETE"])
    app.add_api_route("/apis/{group}/{version}/namespaces/{namespace}/{plural}",
                      _list, methods=["GET"])
    app.add_api_route("/apis/{group}/{version}/namespaces/{namespace}/{plural}",
                      _create, methods=["POST"])
    app.add_api_route("/apis/{group}/{version}/namespaces/{namespace}/{plural}/{name}",
                      _get, methods=["GET"])
    app.add_api_route("/apis/{group}/{version}/namespaces/{namespace}/{plural}/{name}",
                      _put, methods=["PUT"])
    app.add_api_route("/apis/{group}/{version}/namespaces/{namespace}/{plural}/{name}",
                      _patch, methods=["PATCH"])
    app.add_api_route("/apis/{group}/{version}/namespaces/{namespace}/{plural}/{name}",
                      _delete, methods=["DELETE"])

    # status subresource: merge only the status field
    async def _patch_status(group, version, plural, name, request: Request,
                            namespace: str = ""):
        body = json.loads(await request.body())

        def run():
            api_version, kind = _resolve(group, version, plural)
            return cluster.patch_status(
                api_version, kind, name, body.get("status", body), namespace
            )
        return _handle(run)

    app.add_api_route("/api/{version}/{plural}/{name}/status",
                      _wrap_nogroup_status(_patch_status), methods=["PATCH", "PUT"])
    app.add_api_route("/api/{version}/namespaces/{namespace}/{plural}/{name}/status",
                      _wrap_nogroup_status(_patch_status), methods=["PATCH", "PUT"])
    app.add_api_route("/apis/{group}/{version}/{plural}/{name}/status",
                      _patch_status, methods=["PATCH", "PUT"])
    app.add_api_route(
        "/apis/{group}/{version}/namespaces/{namespace}/{plural}/{name}/status",
        _patch_status, methods=["PATCH", "PUT"])

    # pod eviction subresource
    @app.post("/api/v1/namespaces/{namespace}/pods/{name}/eviction")
    async def evict(namespace: str, name: str):
        def run():
            cluster.evict_pod(name, namespace)
            return {"kind": "Status", "status": "Success"}
        return _handle(run)

    return app, cluster


def _wrap_nogroup_status(fn):
    async def handler(version: str, plural: str, name: str, request: Request,
                      namespace: str = ""):
        return await fn("", version, plural, name, request, namespace)
    return handler


def _wrap_nogroup(fn):
    """Adapt the group-style handlers to core-group routes (group='')."""
    import inspect

    params = list(inspect.signature(fn).parameters)

    if "request" in params and "name" in params:
        async def handler(version: str, plural: str, name: str, request: Request,
                          namespace: str = ""):
            return await fn("", version, plural, name, request, namespace)
    elif "name" in params:
        async def handler(version: str, plural: str, name: str, namespace: str = ""):
            return await fn("", version, plural, name, namespace)
    elif "request" in params:
        async def handler(version: str, plural: str, request: Request,
                          namespace: str = ""):
            return await fn("", version, plural, request, namespace)
    else:
        async def handler(version: str, plural: str, namespace: str = ""):
            return await fn("", version, plural, namespace)
    return handler


class ApiServerHandle:
    """A running mini-apiserver on a background thread."""

    def __init__(self, server, thread, cluster, url, engine="thread"):
        self._server = server
        self._thread = thread
        self.cluster = cluster
        self.url = url
        self.engine = engine

    def stop(self) -> None:
        if self.engine == "uvicorn":
            self._server.should_exit = True
        else:
            self._server._shutting_down = True
            self._server.shutdown()
            self._server.server_close()
        self._thread.join(timeout=10)


def start_apiserver(
    host: str = "127.0.0.1", port: int = 0,
    cluster: Optional[FakeCluster] = None,
    engine: str = "thread",
) -> ApiServerHandle:
    """Start the mini-apiserver on a background thread; returns a handle with
    the bound URL (port=0 picks a free port).

    ``engine="thread"`` (default) uses the lean stdlib threaded server —
    the benchmark substrate; ``engine="uvicorn"`` keeps the FastAPI/asyncio
    implementation for cross-checking the wire surface against a second
    independent stack (tests/test_rest_e2e.py runs a matrix)."""
    cluster = cluster or FakeCluster()

    if engine == "thread":
        server = _make_threaded_server(host, port, cluster)
        thread = threading.Thread(target=server.serve_forever,
                                  kwargs={"poll_interval": 0.05}, daemon=True)
        thread.start()
        url = f"http://{host}:{server.server_address[1]}"
        return ApiServerHandle(server, thread, cluster, url, engine="thread")

    import socket

    import uvicorn

    if port == 0:
        with socket.socket() as s:
            s.bind((host, 0))
            port = s.getsockname()[1]

    app, cluster = create_app(cluster)
    config = uvicorn.Config(app, host=host, port=port, log_level="error")
    server = uvicorn.Server(config)
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    url = f"http://{host}:{port}"
    deadline = time.monotonic() + 15
    while not server.started:
        if time.monotonic() > deadline:
            raise RuntimeError("mini-apiserver failed to start in 15s")
        time.sleep(0.01)
    return ApiServerHandle(server, thread, cluster, url, engine="uvicorn")


# ---------------------------------------------------------------------------
# threaded engine: a lean stdlib HTTP server for the same API surface
# ---------------------------------------------------------------------------

def _make_threaded_server(host, port, cluster):
    """A hand-rolled threaded HTTP apiserver over the same FakeCluster.

    Serves the identical wire surface as the FastAPI app (CRUD, merge
    patch, selectors, pagination, status/eviction subresources, discovery,
    RV-anchored watches with bookmarks/ERROR-410) but through
    ``http.server`` with HTTP/1.1 keep-alive and no asyncio: per-request
    overhead is several times lower than the uvicorn engine, which matters
    because this server is the benchmark substrate (the reconcile loop's
    wire cost is dominated by it).  Watch streams use read-until-close
    framing on a dedicated connection."""
    import http.server
    import socketserver
    import urllib.parse

    from .errors import ApiError, GoneError

    class Handler(http.server.BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"
        # TCP_NODELAY: without it, Nagle + delayed-ACK interplay stalls
        # small request/response pairs by ~40ms each
        disable_nagle_algorithm = True

        # -- plumbing --------------------------------------------------------

        def log_message(self, fmt, *args):  # quiet
            pass

        def _send_json(self, obj, code=200):
            body = json.dumps(obj).encode()
            self.send_response(code)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def _read_body(self):
            length = int(self.headers.get("Content-Length") or 0)
            if not length:
                return {}
            return json.loads(self.rfile.read(length) or b"{}")

        def _route(self):
            """Parse the request path into
            (api_version, plural, namespace, name, subresource, query)."""
            parsed = urllib.parse.urlsplit(self.path)
            qp = dict(urllib.parse.parse_qsl(parsed.query))
            from .errors import NotFoundError

            parts = [p for p in parsed.path.split("/") if p]
            if not parts:
                raise NotFoundError("not found")
            if parts[0] == "api":
                api_version = parts[1] if len(parts) > 1 else ""
                rest = parts[2:]
            elif parts[0] == "apis":
                if len(parts) < 3:
                    raise NotFoundError("not found")
                api_version = f"{parts[1]}/{parts[2]}"
                rest = parts[3:]
            else:
                raise NotFoundError("not found")
            namespace = ""
            if rest[:1] == ["namespaces"] and len(rest) >= 2:
                # bare namespace-object ops go through the core route
                if len(rest) == 2 and api_version == "v1":
                    return api_version, "namespaces", "", rest[1], "", qp
                namespace = rest[1]
                rest = rest[2:]
            plural = rest[0] if rest else ""
            name = rest[1] if len(rest) > 1 else ""
            sub = rest[2] if len(rest) > 2 else ""
            return api_version, plural, namespace, name, sub, qp

        def _resolve(self, api_version, plural):
            kind = cluster.lookup_by_plural(api_version, plural)
            if kind is None:
                raise ApiError(f"resource {plural} not served in {api_version}")
            return kind

        def _handle(self, fn):
            try:
                fn()
            except ApiError as exc:
                self._send_json(_status_body(exc), code=exc.code)
            except BrokenPipeError:
                pass

        # -- methods ---------------------------------------------------------

        def do_GET(self):
            def run():
                api_version, plural, namespace, name, sub, qp = self._route()
                if not plural:  # discovery
                    group_version = api_version
                    resources = []
                    for (av, kind), (pl, namespaced) in cluster._kinds.items():
                        if av == group_version:
                            resources.append({"name": pl, "kind": kind,
                                              "namespaced": namespaced})
                    self._send_json({"kind": "APIResourceList",
                                     "groupVersion": group_version,
                                     "resources": resources})
                    return
                kind = self._resolve(api_version, plural)
                if name:
                    self._send_json(cluster.get(api_version, kind, name, namespace))
                    return
                if qp.get("watch") in ("true", "1"):
                    self._stream_watch(api_version, kind, namespace, qp)
                    return
                try:
                    limit = int(qp.get("limit", "0"))
                except ValueError:
                    limit = 0
                items, rv, cont = cluster.list_paged(
                    api_version, kind, namespace=namespace or None,
                    label_selector=qp.get("labelSelector", ""),
                    field_selector=qp.get("fieldSelector", ""),
                    limit=limit, continue_token=qp.get("continue", ""),
                )
                md = {"resourceVersion": rv}
                if cont:
                    md["continue"] = cont
                self._send_json({"kind": f"{kind}List",
                                 "apiVersion": api_version,
                                 "metadata": md, "items": items})
            self._handle(run)

        def _stream_watch(self, api_version, kind, namespace, qp):
            from .errors import GoneError as _Gone

            bookmarks = qp.get("allowWatchBookmarks") in ("true", "1")
            try:
                timeout_s = float(qp.get("timeoutSeconds", "0")) or None
            except ValueError:
                timeout_s = None
            try:
                watch = cluster.watch(
                    api_version, kind, namespace=namespace or None,
                    resource_version=qp.get("resourceVersion") or None,
                    label_selector=qp.get("labelSelector", ""),
                    field_selector=qp.get("fieldSelector", ""),
                    send_initial_events=qp.get("sendInitialEvents")
                    in ("true", "1"),
                )
            except _Gone as exc:
                # 200 + in-stream ERROR Status, the real apiserver shape
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Connection", "close")
                self.end_headers()
                self.wfile.write((json.dumps({
                    "type": "ERROR",
                    "object": {"kind": "Status", "apiVersion": "v1",
                               "status": "Failure", "reason": "Expired",
                               "message": exc.message, "code": 410},
                }) + "\n").encode())
                self.close_connection = True
                return
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Connection", "close")  # read-until-close framing
            self.end_headers()
            self.close_connection = True
            deadline = time.monotonic() + timeout_s if timeout_s else None
            last_bookmark = None
            try:
                while not getattr(server, "_shutting_down", False):
                    if deadline is not None and time.monotonic() >= deadline:
                        return
                    item = watch.next(timeout=0.5)
                    if item is None:
                        if bookmarks:
                            rv = watch.bookmark_rv()
                            if rv is not None and rv != last_bookmark:
                                last_bookmark = rv
                                self.wfile.write((json.dumps({
                                    "type": "BOOKMARK",
                                    "object": {"kind": kind,
                                               "apiVersion": api_version,
                                               "metadata": {"resourceVersion": rv}},
                                }) + "\n").encode())
                                self.wfile.flush()
                                continue
                        # keep-alive probe surfaces dead clients
                        self.wfile.write(b"\n")
                        self.wfile.flush()
                        continue
                    event_type, obj = item
                    self.wfile.write((json.dumps(
                        {"type": event_type, "object": obj}) + "\n").encode())
                    self.wfile.flush()
            except (BrokenPipeError, ConnectionResetError, OSError):
                pass
            finally:
                watch.stop()

        def do_POST(self):
            def run():
                api_version, plural, namespace, name, sub, qp = self._route()
                kind = self._resolve(api_version, plural)
                body = self._read_body()
                if sub == "eviction" and kind == "Pod":
                    cluster.evict_pod(name, namespace)
                    self._send_json({"kind": "Status", "status": "Success"})
                    return
                if namespace:
                    body.setdefault("metadata", {})["namespace"] = namespace
                self._send_json(cluster.create(body), code=201)
            self._handle(run)

        def do_PUT(self):
            def run():
                api_version, plural, namespace, name, sub, qp = self._route()
                kind = self._resolve(api_version, plural)
                body = self._read_body()
                if sub == "status":
                    self._send_json(cluster.patch_status(
                        api_version, kind, name,
                        body.get("status", body), namespace))
                    return
                if namespace:
                    body.setdefault("metadata", {})["namespace"] = namespace
                self._send_json(cluster.update(body))
            self._handle(run)

        def do_PATCH(self):
            def run():
                api_version, plural, namespace, name, sub, qp = self._route()
                kind = self._resolve(api_version, plural)
                body = self._read_body()
                if sub == "status":
                    self._send_json(cluster.patch_status(
                        api_version, kind, name,
                        body.get("status", body), namespace))
                    return
                self._send_json(cluster.patch(api_version, kind, name, body,
                                              namespace))
            self._handle(run)

        def do_DELETE(self):
            def run():
                api_version, plural, namespace, name, sub, qp = self._route()
                kind = self._resolve(api_version, plural)
                self._read_body()  # DeleteOptions accepted, grace ignored here
                cluster.delete(api_version, kind, name, namespace)
                self._send_json({"kind": "Status", "status": "Success"})
            self._handle(run)

    class Server(socketserver.ThreadingMixIn, http.server.HTTPServer):
        daemon_threads = True
        allow_reuse_address = True

    server = Server((host, port), Handler)
    return server
