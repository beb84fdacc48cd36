"""In-memory kube-apiserver: the envtest equivalent.

The reference tests against envtest — a real kube-apiserver + etcd with no
kubelet/scheduler/controllers.  This module provides the same
substrate natively: typed CRUD with resourceVersions and optimistic
concurrency, label/field-selector LISTs, JSON merge patches (with
optimistic-lock support), finalizer-aware deletion, pod eviction, and watch
streams.  Nothing "runs" pods: like envtest, tests and benchmarks force pod /
DaemonSet statuses directly.

It is also the storage engine of :mod:`k8s_operator_libs_amd.core.apiserver`,
which serves it over HTTP so the REST client can be tested wire-level.
"""

from __future__ import annotations

import collections
import json
import logging
import queue
import threading
import time
import uuid
from typing import Callable, Dict, List, Optional, Tuple

from . import meta
from .errors import (
    AlreadyExistsError,
    ApiError,
    BadRequestError,
    ConflictError,
    NotFoundError,
)
from .meta import K8sObject

# (apiVersion, kind) -> (plural, namespaced)
_BUILTIN_KINDS: Dict[Tuple[str, str], Tuple[str, bool]] = {
    ("v1", "Node"): ("nodes", False),
    ("v1", "Pod"): ("pods", True),
    ("v1", "Event"): ("events", True),
    ("v1", "Namespace"): ("namespaces", False),
    ("apps/v1", "DaemonSet"): ("daemonsets", True),
    ("apps/v1", "ControllerRevision"): ("controllerrevisions", True),
    ("apiextensions.k8s.io/v1", "CustomResourceDefinition"): (
        "customresourcedefinitions",
        False,
    ),
    # AMD maintenance-operator API (requestor mode; the reference's analogue
    # is maintenance.nvidia.com/v1alpha1 NodeMaintenance).
    ("maintenance.amd.com/v1alpha1", "NodeMaintenance"): ("nodemaintenances", True),
    ("coordination.k8s.io/v1", "Lease"): ("leases", True),
    ("policy/v1", "PodDisruptionBudget"): ("poddisruptionbudgets", True),
}


class Watch:
    """A watch stream: a queue of ``(event_type, object)`` tuples where
    event_type is ``ADDED`` / ``MODIFIED`` / ``DELETED`` (or ``ERROR`` with a
    Status object — the Kubernetes wire shape for an expired watch).

    Supports the real apiserver's filtering semantics: an optional namespace
    scope and label selector, where an object that stops matching the
    selector is delivered as ``DELETED`` and one that starts matching as
    ``ADDED`` (client-go reflector contract)."""

    def __init__(
        self,
        cluster: "FakeCluster",
        key: Tuple[str, str],
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
    ) -> None:
        self._cluster = cluster
        self._key = key
        self._namespace = namespace or None
        selectors = []
        if label_selector:
            selectors.append(meta.parse_label_selector(label_selector))
        if field_selector:
            selectors.append(meta.parse_field_selector(field_selector))
        self._selectors = selectors or None
        # object keys currently matching the selectors (for the
        # stops-matching -> DELETED transform); only used with selectors
        self._matched: set = set()
        self.events: "queue.Queue[Tuple[str, K8sObject]]" = queue.Queue()
        self._stopped = False

    def _matches(self, snapshot: K8sObject) -> bool:
        return all(s.matches_object(snapshot) for s in self._selectors)

    def _deliver(self, event_type: str, snapshot: K8sObject) -> None:
        """Apply namespace/selector filtering; called under the cluster lock
        so delivery order matches resourceVersion order."""
        if self._namespace is not None and meta.namespace(snapshot) != self._namespace:
            return
        if self._selectors is not None:
            okey = (meta.namespace(snapshot), meta.name(snapshot))
            matches = event_type != "DELETED" and self._matches(snapshot)
            was_matched = okey in self._matched
            if event_type == "DELETED":
                # deliver if this watch saw the object OR the final snapshot
                # matches (a resumed watch replaying a delete it never saw
                # the ADDED for must still get the DELETED)
                if not was_matched and not self._matches(snapshot):
                    return
                self._matched.discard(okey)
            elif matches and not was_matched:
                self._matched.add(okey)
                event_type = "ADDED"
            elif matches:
                pass  # stays MODIFIED
            elif was_matched:
                self._matched.discard(okey)
                event_type = "DELETED"
            else:
                return
        self.events.put((event_type, snapshot))

    def next(self, timeout: Optional[float] = None) -> Optional[Tuple[str, K8sObject]]:
        try:
            return self.events.get(timeout=timeout)
        except queue.Empty:
            return None

    def bookmark_rv(self) -> Optional[str]:
        """Current cluster resourceVersion, but only when this watch's queue
        is drained — taken under the cluster lock so no event at or below the
        returned RV can still be pending delivery.  This makes the value safe
        to hand out as a BOOKMARK a client may resume from."""
        with self._cluster._lock:
            if not self.events.empty():
                return None
            return str(self._cluster._rv_counter)

    def stop(self) -> None:
        self._stopped = True
        self._cluster._remove_watch(self)

    def alive(self) -> bool:
        return not self._stopped


class FakeCluster:
    """Thread-safe in-memory Kubernetes object store."""

    #: events retained per kind for resourceVersion-anchored watch resume;
    #: resuming below the window -> 410 Gone (matches the apiserver's
    #: bounded etcd watch cache)
    WATCH_HISTORY = 2048

    def __init__(self) -> None:
        self._lock = threading.RLock()
        # (apiVersion, kind) -> {(namespace, name): object}
        self._store: Dict[Tuple[str, str], Dict[Tuple[str, str], K8sObject]] = {}
        # starts at 1: a real apiserver never serves resourceVersion "0"
        # (it is the magic watch value "any revision"), so neither may an
        # empty cluster's LIST here — found by tests/test_property_watch.py
        self._rv_counter = 1
        self._kinds = dict(_BUILTIN_KINDS)
        self._watches: Dict[Tuple[str, str], List[Watch]] = {}
        self._change_hooks: List[Callable[[str, K8sObject], None]] = []
        # Watch-resume cache: per kind, a bounded deque of
        # (rv_int, event_type, snapshot).  Recording starts when the first
        # watch on the kind opens (before that, any RV-anchored resume gets
        # 410 Gone, which is the correct "too old" answer).
        self._history: Dict[Tuple[str, str], "collections.deque"] = {}
        # RV from which each kind's history is complete
        self._history_start: Dict[Tuple[str, str], int] = {}
        # structural schemas of CRD-registered kinds, enforced on CR writes
        self._cr_schemas: Dict[Tuple[str, str], dict] = {}
        # CRD kinds declaring subresources.status: main-resource writes
        # cannot touch .status and /status cannot touch the rest (real
        # apiserver isolation; builtin-registered kinds are exempt so
        # fixtures keep the envtest Status().Update convenience)
        self._status_subresource: set = set()
        # spec.nodeName index for pods: nodeName -> {(ns, name)}; keeps
        # per-node pod LISTs (the reconcile loop's hottest query) O(pods on
        # node) instead of O(all pods)
        self._pod_node_index: Dict[str, set] = {}

    # -- kind registry -------------------------------------------------------

    def register_kind(self, api_version: str, kind: str, plural: str, namespaced: bool) -> None:
        with self._lock:
            self._kinds[(api_version, kind)] = (plural, namespaced)

    def register_crd(self, crd: K8sObject) -> None:
        """Make a created CustomResourceDefinition's kind servable and
        record its structural schema — subsequent CRs of the kind are
        validated against it (422 Invalid), like a real apiserver."""
        spec = crd.get("spec", {})
        group = spec.get("group", "")
        names = spec.get("names", {})
        kind = names.get("kind", "")
        plural = names.get("plural", "")
        namespaced = spec.get("scope", "Namespaced") == "Namespaced"
        for ver in spec.get("versions", []):
            if ver.get("served", True):
                api_version = f"{group}/{ver['name']}"
                self.register_kind(api_version, kind, plural, namespaced)
                schema = (ver.get("schema") or {}).get("openAPIV3Schema")
                if schema:
                    self._cr_schemas[(api_version, kind)] = schema
                if "status" in (ver.get("subresources") or {}):
                    self._status_subresource.add((api_version, kind))

    def lookup_kind(self, api_version: str, kind: str) -> Tuple[str, bool]:
        try:
            return self._kinds[(api_version, kind)]
        except KeyError:
            raise BadRequestError(f"unknown kind {api_version}/{kind}") from None

    def lookup_by_plural(self, api_version: str, plural: str) -> Optional[str]:
        """Reverse lookup used by the HTTP server and CRD-establish polling."""
        with self._lock:
            for (av, kind), (pl, _ns) in self._kinds.items():
                if av == api_version and pl == plural:
                    return kind
        return None

    # -- internal helpers ----------------------------------------------------

    def _validate_cr(self, api_version: str, kind: str, obj: K8sObject) -> None:
        schema = self._cr_schemas.get((api_version, kind))
        if schema is None:
            return
        from .errors import InvalidError

        body = {k: v for k, v in obj.items()
                if k not in ("apiVersion", "kind", "metadata")}
        errs = meta.validate_structural_schema(body, schema, path=kind)
        if errs:
            raise InvalidError(
                f"{kind} is invalid: " + "; ".join(errs[:5])
            )

    def _next_rv(self) -> str:
        with self._lock:
            self._rv_counter += 1
            return str(self._rv_counter)

    def current_rv(self) -> str:
        """The cluster-level resourceVersion a LIST would carry right now."""
        with self._lock:
            return str(self._rv_counter)

    @staticmethod
    def _obj_key(obj: K8sObject) -> Tuple[str, str]:
        return (meta.api_version(obj), meta.kind(obj))

    def _bucket(self, api_version: str, kind: str) -> Dict[Tuple[str, str], K8sObject]:
        self.lookup_kind(api_version, kind)  # validate
        return self._store.setdefault((api_version, kind), {})

    def _notify(self, event_type: str, obj: K8sObject) -> None:
        key = self._obj_key(obj)
        watches = self._watches.get(key)
        history = self._history.get(key)
        if not watches and not self._change_hooks and history is None:
            return  # nobody listening: skip the snapshot copy (hot path)
        snapshot = meta.deep_copy(obj)
        if history is not None:
            try:
                history.append(
                    (int(meta.resource_version(snapshot)), event_type, snapshot)
                )
            except ValueError:
                pass
        for w in watches or ():
            w._deliver(event_type, snapshot)
        for hook in self._change_hooks:
            # hooks are observers (simulated controllers): a failing hook
            # must never fail the API mutation that triggered it
            try:
                hook(event_type, snapshot)
            except Exception:
                logging.getLogger(__name__).exception(
                    "change hook failed for %s %s", event_type, meta.name(obj)
                )

    def _index_pod(self, obj: K8sObject, remove: bool = False) -> None:
        if meta.kind(obj) != "Pod":
            return
        node = obj.get("spec", {}).get("nodeName", "")
        key = (meta.namespace(obj), meta.name(obj))
        if remove:
            bucket = self._pod_node_index.get(node)
            if bucket is not None:
                bucket.discard(key)
        else:
            self._pod_node_index.setdefault(node, set()).add(key)

    def add_change_hook(self, hook: Callable[[str, K8sObject], None]) -> None:
        """Register a callback fired on every mutation (used by simulated
        kubelets / maintenance operators in tests and benchmarks)."""
        with self._lock:
            self._change_hooks.append(hook)

    def _remove_watch(self, watch: Watch) -> None:
        with self._lock:
            lst = self._watches.get(watch._key, [])
            if watch in lst:
                lst.remove(watch)

    # -- CRUD ----------------------------------------------------------------

    def create(self, obj: K8sObject) -> K8sObject:
        obj = meta.deep_copy(obj)
        api_version, kind = self._obj_key(obj)
        with self._lock:
            _, namespaced = self.lookup_kind(api_version, kind)
            md = obj.setdefault("metadata", {})
            ns = md.get("namespace", "")
            if namespaced and not ns:
                md["namespace"] = ns = "default"
            if not namespaced:
                md.pop("namespace", None)
                ns = ""
            name_ = md.get("name", "")
            if not name_:
                gen = md.get("generateName")
                if not gen:
                    raise BadRequestError("metadata.name is required")
                name_ = f"{gen}{uuid.uuid4().hex[:6]}"
                md["name"] = name_
            bucket = self._bucket(api_version, kind)
            if (ns, name_) in bucket:
                raise AlreadyExistsError(f"{kind} {ns}/{name_} already exists")
            if (api_version, kind) in self._status_subresource:
                obj.pop("status", None)  # real apiservers drop it on create
            self._validate_cr(api_version, kind, obj)
            # respect a caller-provided uid (snapshot load, fixtures with
            # pre-wired ownerReferences); assign one otherwise
            md["uid"] = md.get("uid") or str(uuid.uuid4())
            md["resourceVersion"] = self._next_rv()
            md.setdefault("creationTimestamp", _now_iso())
            md.setdefault("generation", 1)
            bucket[(ns, name_)] = obj
            self._index_pod(obj)
            if kind == "CustomResourceDefinition":
                self.register_crd(obj)
                self._establish_crd(obj)
            self._notify("ADDED", obj)
            return meta.deep_copy(obj)

    def get(self, api_version: str, kind: str, name: str, namespace: str = "") -> K8sObject:
        with self._lock:
            _, namespaced = self.lookup_kind(api_version, kind)
            ns = namespace if namespaced else ""
            bucket = self._bucket(api_version, kind)
            obj = bucket.get((ns, name))
            if obj is None:
                raise NotFoundError(f"{kind} {ns}/{name} not found")
            return meta.deep_copy(obj)

    def list(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
    ) -> List[K8sObject]:
        lsel = meta.parse_label_selector(label_selector)
        fsel = meta.parse_field_selector(field_selector)
        with self._lock:
            bucket = self._bucket(api_version, kind)
            candidates = bucket.items()
            if kind == "Pod" and field_selector:
                for fkey, fop, fval in fsel._reqs:
                    if fkey == "spec.nodeName" and fop == "=":
                        keys = self._pod_node_index.get(fval, set())
                        candidates = [(k, bucket[k]) for k in keys if k in bucket]
                        break
            out = []
            for (ns, _name), obj in candidates:
                if namespace is not None and namespace != "" and ns != namespace:
                    continue
                if label_selector and not lsel.matches_object(obj):
                    continue
                if field_selector and not fsel.matches_object(obj):
                    continue
                out.append(meta.deep_copy(obj))
            out.sort(key=lambda o: (meta.namespace(o), meta.name(o)))
            return out

    def list_with_meta(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
    ) -> Tuple[List[K8sObject], str]:
        """LIST plus the list-level ``metadata.resourceVersion`` a watch can
        be anchored at (the reflector's LIST-then-WATCH(rv) contract)."""
        with self._lock:
            items = self.list(
                api_version, kind, namespace=namespace,
                label_selector=label_selector, field_selector=field_selector,
            )
            return items, str(self._rv_counter)

    def list_paged(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
        limit: int = 0,
        continue_token: str = "",
    ) -> Tuple[List[K8sObject], str, str]:
        """Chunked LIST (``limit``/``continue`` — apiserver pagination).

        Returns ``(items, list_rv, next_continue)``; an empty
        ``next_continue`` means the list is complete.  The continue token
        encodes the last returned (namespace, name): like a real apiserver
        the second chunk continues *after* that key in (ns, name) order.
        (A real apiserver additionally serves continues from an etcd
        snapshot; here later chunks see current state, which client-go
        tolerates — documented in docs/testing.md.)
        """
        import base64

        with self._lock:
            items = self.list(
                api_version, kind, namespace=namespace,
                label_selector=label_selector, field_selector=field_selector,
            )
            rv = str(self._rv_counter)
            if continue_token:
                try:
                    last = json.loads(
                        base64.urlsafe_b64decode(continue_token.encode()).decode()
                    )
                    last_key = (last["ns"], last["name"])
                except Exception:
                    raise BadRequestError(
                        "invalid continue token"
                    ) from None
                items = [
                    o for o in items
                    if (meta.namespace(o), meta.name(o)) > last_key
                ]
            next_token = ""
            if limit and len(items) > limit:
                items = items[:limit]
                tail = items[-1]
                next_token = base64.urlsafe_b64encode(json.dumps({
                    "ns": meta.namespace(tail), "name": meta.name(tail),
                }).encode()).decode()
            return items, rv, next_token

    def update(self, obj: K8sObject) -> K8sObject:
        obj = meta.deep_copy(obj)
        api_version, kind = self._obj_key(obj)
        with self._lock:
            _, namespaced = self.lookup_kind(api_version, kind)
            ns = meta.namespace(obj) if namespaced else ""
            name_ = meta.name(obj)
            bucket = self._bucket(api_version, kind)
            stored = bucket.get((ns, name_))
            if stored is None:
                raise NotFoundError(f"{kind} {ns}/{name_} not found")
            rv = meta.resource_version(obj)
            if rv and rv != meta.resource_version(stored):
                raise ConflictError(
                    f"{kind} {ns}/{name_}: resourceVersion {rv} is stale"
                )
            self._validate_cr(api_version, kind, obj)
            if (api_version, kind) in self._status_subresource:
                # main-resource writes cannot change status
                if "status" in stored:
                    obj["status"] = meta.deep_copy(stored["status"])
                else:
                    obj.pop("status", None)
            # Immutable server-side fields carry over.
            obj["metadata"]["uid"] = stored["metadata"]["uid"]
            obj["metadata"]["creationTimestamp"] = stored["metadata"]["creationTimestamp"]
            if "deletionTimestamp" in stored["metadata"]:
                obj["metadata"]["deletionTimestamp"] = stored["metadata"]["deletionTimestamp"]
            obj["metadata"]["resourceVersion"] = self._next_rv()
            # generation bumps only on non-status content changes (real
            # apiserver semantics for resources with a status subresource;
            # harmless approximation for the rest)
            def _gen_view(o):
                return {k: v for k, v in o.items()
                        if k not in ("metadata", "status")}
            gen = stored["metadata"].get("generation", 1)
            if _gen_view(obj) != _gen_view(stored):
                gen += 1
            obj["metadata"]["generation"] = gen
            self._index_pod(stored, remove=True)
            bucket[(ns, name_)] = obj
            self._index_pod(obj)
            if self._finalize_if_ready(api_version, kind, ns, name_):
                # deletion completed by this update: only DELETED was emitted
                return meta.deep_copy(obj)
            self._notify("MODIFIED", obj)
            return meta.deep_copy(obj)

    def patch(
        self,
        api_version: str,
        kind: str,
        name: str,
        patch: K8sObject,
        namespace: str = "",
    ) -> K8sObject:
        """Apply an RFC 7386 JSON merge patch.  If the patch carries
        ``metadata.resourceVersion`` it acts as an optimistic lock (the
        shared-requestor protocol relies on this — reference
        upgrade_requestor.go:320-368)."""
        with self._lock:
            _, namespaced = self.lookup_kind(api_version, kind)
            ns = namespace if namespaced else ""
            bucket = self._bucket(api_version, kind)
            stored = bucket.get((ns, name))
            if stored is None:
                raise NotFoundError(f"{kind} {ns}/{name} not found")
            patch_rv = (
                patch.get("metadata", {}).get("resourceVersion")
                if isinstance(patch.get("metadata"), dict)
                else None
            )
            if patch_rv and patch_rv != meta.resource_version(stored):
                raise ConflictError(
                    f"{kind} {ns}/{name}: resourceVersion {patch_rv} is stale"
                )
            if (api_version, kind) in self._status_subresource and \
                    isinstance(patch, dict) and "status" in patch:
                patch = {k: v for k, v in patch.items() if k != "status"}
            old_node = stored.get("spec", {}).get("nodeName", "") if kind == "Pod" else None
            if (api_version, kind) in self._cr_schemas:
                preview = meta.deep_copy(stored)
                meta.json_merge_patch(preview, patch)
                self._validate_cr(api_version, kind, preview)
            meta.json_merge_patch(stored, patch)
            stored["metadata"]["name"] = name  # patches cannot rename
            stored["metadata"]["resourceVersion"] = self._next_rv()
            if kind == "Pod" and stored.get("spec", {}).get("nodeName", "") != old_node:
                self._pod_node_index.get(old_node, set()).discard((ns, name))
                self._index_pod(stored)
            if self._finalize_if_ready(api_version, kind, ns, name):
                return meta.deep_copy(stored)
            self._notify("MODIFIED", stored)
            return meta.deep_copy(stored)

    def patch_status(
        self,
        api_version: str,
        kind: str,
        name: str,
        status: K8sObject,
        namespace: str = "",
    ) -> K8sObject:
        """The /status subresource: merges ONLY ``.status`` (and, for kinds
        with the subresource declared, is the only way to change it)."""
        with self._lock:
            _, namespaced = self.lookup_kind(api_version, kind)
            ns = namespace if namespaced else ""
            bucket = self._bucket(api_version, kind)
            stored = bucket.get((ns, name))
            if stored is None:
                raise NotFoundError(f"{kind} {ns}/{name} not found")
            merged = meta.deep_copy(stored)
            meta.json_merge_patch(merged, {"status": status})
            self._validate_cr(api_version, kind, merged)
            if "status" in merged:
                stored["status"] = merged["status"]
            else:
                stored.pop("status", None)
            stored["metadata"]["resourceVersion"] = self._next_rv()
            self._notify("MODIFIED", stored)
            return meta.deep_copy(stored)

    def delete(self, api_version: str, kind: str, name: str, namespace: str = "") -> None:
        with self._lock:
            _, namespaced = self.lookup_kind(api_version, kind)
            ns = namespace if namespaced else ""
            bucket = self._bucket(api_version, kind)
            stored = bucket.get((ns, name))
            if stored is None:
                raise NotFoundError(f"{kind} {ns}/{name} not found")
            finalizers = stored.get("metadata", {}).get("finalizers") or []
            if finalizers:
                if "deletionTimestamp" not in stored["metadata"]:
                    stored["metadata"]["deletionTimestamp"] = _now_iso()
                    stored["metadata"]["resourceVersion"] = self._next_rv()
                    self._notify("MODIFIED", stored)
                return
            self._index_pod(stored, remove=True)
            del bucket[(ns, name)]
            # deletion gets its own resourceVersion (real apiservers do this;
            # required so an RV-anchored watch resume replays the delete)
            stored["metadata"]["resourceVersion"] = self._next_rv()
            self._notify("DELETED", stored)

    def _finalize_if_ready(self, api_version: str, kind: str, ns: str, name: str) -> bool:
        """Remove an object whose deletion was pending once finalizers empty.
        Returns True if the object was finalized (DELETED emitted)."""
        bucket = self._bucket(api_version, kind)
        stored = bucket.get((ns, name))
        if stored is None:
            return False
        md = stored.get("metadata", {})
        if "deletionTimestamp" in md and not (md.get("finalizers") or []):
            self._index_pod(stored, remove=True)
            del bucket[(ns, name)]
            stored["metadata"]["resourceVersion"] = self._next_rv()
            self._notify("DELETED", stored)
            return True
        return False

    # -- pods ----------------------------------------------------------------

    def evict_pod(self, name: str, namespace: str) -> None:
        """Eviction API: enforces PodDisruptionBudgets like a real apiserver
        (429 when the eviction would violate a budget), then deletes
        immediately (no kubelet, like envtest)."""
        with self._lock:
            pod = self._bucket("v1", "Pod").get((namespace, name))
            if pod is None:
                raise NotFoundError(f"Pod {namespace}/{name} not found")
            self._check_disruption_budgets(pod, namespace)
            self.delete("v1", "Pod", name, namespace)

    @staticmethod
    def _pod_healthy(pod: K8sObject) -> bool:
        if pod.get("status", {}).get("phase") != "Running":
            return False
        statuses = pod.get("status", {}).get("containerStatuses") or []
        return bool(statuses) and all(c.get("ready") for c in statuses)

    def _check_disruption_budgets(self, pod: K8sObject, namespace: str) -> None:
        from .meta import match_labels_selector

        pod_labels = pod.get("metadata", {}).get("labels", {}) or {}
        for (ns, _), pdb in self._bucket("policy/v1", "PodDisruptionBudget").items():
            if ns != namespace:
                continue
            sel = match_labels_selector(
                pdb.get("spec", {}).get("selector", {}).get("matchLabels", {})
            )
            if not sel.matches(pod_labels):
                continue
            matched = [
                p for (pns, _), p in self._bucket("v1", "Pod").items()
                if pns == namespace and sel.matches(
                    p.get("metadata", {}).get("labels", {}) or {})
            ]
            healthy = sum(1 for p in matched if self._pod_healthy(p))
            spec = pdb.get("spec", {})
            # Real PDBs budget against the owning controller's scale
            # (status.expectedPods, maintained by the disruption controller);
            # honour it when set, else fall back to the live matched count.
            expected = pdb.get("status", {}).get("expectedPods") or len(matched)
            min_available = spec.get("minAvailable")
            max_unavailable = spec.get("maxUnavailable")
            after = healthy - (1 if self._pod_healthy(pod) else 0)
            blocked = False
            if min_available is not None:
                need = _scaled(min_available, expected, round_up=False)
                blocked = after < need
            elif max_unavailable is not None:
                allowed = _scaled(max_unavailable, expected, round_up=False)
                blocked = (expected - after) > allowed
            if blocked:
                err = ApiError(
                    f"Cannot evict pod as it would violate the pod's "
                    f"disruption budget {meta.name(pdb)}"
                )
                err.code = 429
                raise err

    # -- watches -------------------------------------------------------------

    def _history_floor(self, key: Tuple[str, str]) -> Optional[int]:
        """Lowest RV from which this kind's event history is complete, or
        None if history has never been recorded."""
        start = self._history_start.get(key)
        if start is None:
            return None
        hist = self._history[key]
        if len(hist) == hist.maxlen:
            # ring evicted events: complete only from the oldest retained - 1
            return hist[0][0] - 1
        return start

    def watch(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        resource_version: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
        send_initial_events: bool = False,
    ) -> Watch:
        """Open a watch stream with Kubernetes resourceVersion semantics:

        - ``resource_version=None`` — live events from now on (the legacy
          in-process behavior; real apiservers treat an unset RV as "list
          current state then watch", which :meth:`list_with_meta` + an
          anchored watch composes for the informer).
        - ``"0"`` — synthetic ``ADDED`` for every currently-stored matching
          object, then live events.
        - any other value — replay retained history with RV strictly greater
          than the given value, then live; raises
          :class:`~k8s_operator_libs_amd.core.errors.GoneError` (410) when
          the requested RV has fallen out of the bounded history window.
        """
        from .errors import GoneError

        with self._lock:
            self.lookup_kind(api_version, kind)
            key = (api_version, kind)
            # first watch on a kind turns its resume history on
            if key not in self._history:
                self._history[key] = collections.deque(maxlen=self.WATCH_HISTORY)
                self._history_start[key] = self._rv_counter
            w = Watch(self, key, namespace=namespace,
                      label_selector=label_selector,
                      field_selector=field_selector)
            if send_initial_events:
                # WatchList protocol (KEP-3157, beta in 1.32): stream the
                # current state as synthetic ADDEDs, then a BOOKMARK whose
                # annotation k8s.io/initial-events-end marks the consistent
                # snapshot point; live events follow losslessly (we hold
                # the cluster lock throughout registration).
                for obj in self._store.get(key, {}).values():
                    w._deliver("ADDED", meta.deep_copy(obj))
                w.events.put(("BOOKMARK", {
                    "kind": kind, "apiVersion": api_version,
                    "metadata": {
                        "resourceVersion": str(self._rv_counter),
                        "annotations": {"k8s.io/initial-events-end": "true"},
                    },
                }))
                self._watches.setdefault(key, []).append(w)
                return w
            if resource_version == "0":
                for obj in self._store.get(key, {}).values():
                    w._deliver("ADDED", meta.deep_copy(obj))
            elif resource_version not in (None, ""):
                try:
                    rv_int = int(resource_version)
                except ValueError:
                    raise BadRequestError(
                        f"invalid resourceVersion {resource_version!r}"
                    ) from None
                floor = self._history_floor(key)
                if floor is None or rv_int < floor:
                    raise GoneError(
                        f"too old resource version: {resource_version} "
                        f"(history starts at {floor})"
                    )
                for ev_rv, ev_type, snapshot in self._history[key]:
                    if ev_rv > rv_int:
                        w._deliver(ev_type, snapshot)
            self._watches.setdefault(key, []).append(w)
            return w

    # -- CRD establishment ---------------------------------------------------

    def _establish_crd(self, crd: K8sObject) -> None:
        """A real apiserver establishes CRDs asynchronously; we mark the
        Established condition immediately (discovery still exercises the
        polling path through :meth:`lookup_by_plural`)."""
        crd.setdefault("status", {})["conditions"] = [
            {"type": "Established", "status": "True", "reason": "InitialNamesAccepted"}
        ]

    # -- snapshot dump/load (debugging, deterministic replays) ----------------

    def dump(self) -> dict:
        """Serializable snapshot of every stored object (plus the dynamic
        kind registry), for bug reports and replay tests."""
        with self._lock:
            return {
                "kinds": [
                    {"apiVersion": av, "kind": k, "plural": pl, "namespaced": ns}
                    for (av, k), (pl, ns) in sorted(self._kinds.items())
                ],
                "objects": [meta.deep_copy(o) for b in self._store.values()
                            for o in b.values()],
            }

    @classmethod
    def load(cls, snapshot: dict) -> "FakeCluster":
        """Rebuild a cluster from :meth:`dump` output.  resourceVersions are
        reassigned (monotonic) but relative object content is preserved."""
        cluster = cls()
        for k in snapshot.get("kinds", []):
            cluster.register_kind(k["apiVersion"], k["kind"], k["plural"],
                                  k["namespaced"])
        # CRDs first so schema/subresource registration precedes their CRs;
        # then restore statuses through the subresource for isolated kinds
        # (create strips them, like a real apiserver)
        objs = sorted(
            snapshot.get("objects", []),
            key=lambda o: 0 if meta.kind(o) == "CustomResourceDefinition" else 1,
        )
        for obj in objs:
            obj = meta.deep_copy(obj)
            obj.get("metadata", {}).pop("resourceVersion", None)
            status = obj.get("status")
            created = cluster.create(obj)
            key = (meta.api_version(obj), meta.kind(obj))
            if status is not None and key in cluster._status_subresource:
                cluster.patch_status(
                    meta.api_version(obj), meta.kind(obj),
                    meta.name(created), status, meta.namespace(created),
                )
        return cluster

    # -- convenience for tests/benchmarks ------------------------------------

    def object_count(self) -> int:
        with self._lock:
            return sum(len(b) for b in self._store.values())


def _scaled(value, total: int, round_up: bool) -> int:
    from ..api.upgrade.v1alpha1 import IntOrString

    return IntOrString.scaled_value(value, total, round_up)


def _now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
