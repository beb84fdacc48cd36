"""Informer-style cached client.

controller-runtime clients read from a watch-backed cache; writes go to the
apiserver and the cache catches up asynchronously.  That staleness window is
why the reference's NodeUpgradeStateProvider polls after every label patch
(node_upgrade_state_provider.go:92-117).  This module provides the same
architecture natively:

- :class:`CachedClient` wraps any :class:`~k8s_operator_libs_amd.core.client.Client`
  that supports ``watch``: reads (get/list) are served from in-memory stores
  kept current by one watch thread per kind; writes pass through to the
  delegate.
- An optional ``sync_delay`` injects artificial cache lag for tests, proving
  the provider's patch-then-confirm barrier converges rather than
  double-firing transitions.

``wait_for_cache_sync`` blocks until the initial LIST of every informer has
been applied (controller-runtime ``WaitForCacheSync`` analogue).
"""

from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional, Tuple

from . import meta
from .client import Client
from .errors import NotFoundError
from .meta import K8sObject


class _Informer:
    #: reconnect backoff bounds (seconds) — client-go reflector-style
    BACKOFF_BASE = 0.1
    BACKOFF_MAX = 5.0

    #: bootstrap via the WatchList protocol (KEP-3157: sendInitialEvents
    #: streams the initial state through the watch, ending with a BOOKMARK
    #: annotated k8s.io/initial-events-end) instead of a separate LIST —
    #: one less round trip and no list-sized memory spike.  Falls back to
    #: LIST-then-WATCH(rv) automatically when the server rejects it.
    USE_WATCH_LIST = True

    def __init__(self, delegate: Client, api_version: str, kind: str,
                 sync_delay: float = 0.0) -> None:
        self.api_version = api_version
        self.kind = kind
        self._delegate = delegate
        self._sync_delay = sync_delay
        self._store: Dict[Tuple[str, str], K8sObject] = {}
        self._lock = threading.RLock()
        # notified on every store mutation: RV-aware barriers block here
        # instead of polling (see CachedClient.wait_for_resource_version)
        self._changed = threading.Condition(self._lock)
        self._synced = threading.Event()
        self._stop = threading.Event()
        self._watch = None
        self._thread: Optional[threading.Thread] = None
        # last resourceVersion this informer has processed: the anchor for
        # lossless watch reconnects (client-go reflector lastSyncResourceVersion)
        self._last_rv: Optional[str] = None

    def start(self) -> None:
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _relist(self) -> None:
        """Full LIST replacing the store and re-anchoring ``_last_rv``
        (initial sync and 410-Gone recovery)."""
        objs, rv = self._delegate.list_with_meta(self.api_version, self.kind)
        with self._lock:
            self._store = {
                (meta.namespace(obj), meta.name(obj)): obj for obj in objs
            }
            self._last_rv = rv
            self._changed.notify_all()

    def _open_watch(self):
        """WATCH anchored at the last processed resourceVersion; a delegate
        without RV support (``_last_rv`` None) gets a live-only watch, in
        which case the caller must LIST *after* opening to close the gap."""
        return self._delegate.watch(
            self.api_version, self.kind, resource_version=self._last_rv
        )

    def _bootstrap_watch_list(self) -> bool:
        """WatchList bootstrap: consume initial ADDEDs until the
        initial-events-end bookmark, then keep the same stream live.
        Returns False (cleanly) when the delegate/server lacks support."""
        try:
            w = self._delegate.watch(self.api_version, self.kind,
                                     send_initial_events=True)
        except TypeError:
            return False  # delegate without the parameter
        except Exception:
            return False  # server rejected sendInitialEvents
        store: Dict[Tuple[str, str], K8sObject] = {}
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline and not self._stop.is_set():
            item = w.next(timeout=0.5)
            if item is None:
                alive = getattr(w, "alive", None)
                if alive is not None and not alive():
                    break
                continue
            event_type, obj = item
            if event_type == "BOOKMARK":
                md = (obj or {}).get("metadata", {})
                if (md.get("annotations") or {}).get(
                        "k8s.io/initial-events-end") == "true":
                    with self._lock:
                        self._store = store
                        self._last_rv = md.get("resourceVersion")
                        self._changed.notify_all()
                    self._watch = w
                    return True
                continue
            if event_type == "ERROR":
                break
            key = (meta.namespace(obj), meta.name(obj))
            if event_type == "DELETED":
                store.pop(key, None)
            else:
                store[key] = obj
        w.stop()
        return False

    def _run(self) -> None:
        from .errors import GoneError

        backoff = self.BACKOFF_BASE
        # reflector bootstrap: WatchList when available, else LIST (capture
        # list RV) -> WATCH(from that RV).  With an RV-anchored delegate
        # nothing can be lost in between; with a live-only delegate,
        # watch-then-relist covers the gap instead.
        while not self._stop.is_set():
            try:
                if self.USE_WATCH_LIST and self._bootstrap_watch_list():
                    break
                self._relist()
                if self._last_rv is None:
                    # legacy live-only delegate: open watch first, then
                    # relist so events during the LIST aren't dropped
                    self._watch = self._open_watch()
                    self._relist()
                else:
                    self._watch = self._open_watch()
                break
            except Exception:
                self._stop.wait(backoff)
                backoff = min(backoff * 2, self.BACKOFF_MAX)
        self._synced.set()
        backoff = self.BACKOFF_BASE
        while not self._stop.is_set():
            item = self._watch.next(timeout=0.2)
            if item is None:
                alive = getattr(self._watch, "alive", None)
                if alive is not None and not alive():
                    # stream dropped: re-watch from the last processed RV —
                    # only a 410 (resume window expired) forces a full relist
                    self._watch.stop()
                    try:
                        self._watch = self._open_watch()
                        if self._last_rv is None:
                            self._relist()
                        backoff = self.BACKOFF_BASE
                    except GoneError:
                        try:
                            self._relist()
                            self._watch = self._open_watch()
                            backoff = self.BACKOFF_BASE
                        except Exception:
                            self._stop.wait(backoff)
                            backoff = min(backoff * 2, self.BACKOFF_MAX)
                    except Exception:
                        self._stop.wait(backoff)
                        backoff = min(backoff * 2, self.BACKOFF_MAX)
                continue
            event_type, obj = item
            if event_type == "BOOKMARK":
                rv = (obj or {}).get("metadata", {}).get("resourceVersion")
                if rv:
                    self._last_rv = rv
                continue
            if event_type == "ERROR":
                # Kubernetes signals an expired watch as an in-stream ERROR
                # Status (code 410): relist + re-watch from the fresh RV
                code = (obj or {}).get("code")
                try:
                    if code == 410:
                        self._relist()
                    self._watch.stop()
                    self._watch = self._open_watch()
                except Exception:
                    self._stop.wait(backoff)
                    backoff = min(backoff * 2, self.BACKOFF_MAX)
                continue
            if self._sync_delay:
                time.sleep(self._sync_delay)
            key = (meta.namespace(obj), meta.name(obj))
            with self._lock:
                rv = meta.resource_version(obj)
                if event_type == "DELETED":
                    self._store.pop(key, None)
                    self._last_rv = rv or self._last_rv
                    self._changed.notify_all()
                else:
                    current = self._store.get(key)
                    # resourceVersions are monotonic ints in this stack;
                    # never regress the cache on out-of-order delivery
                    if current is not None:
                        try:
                            if int(rv) < int(meta.resource_version(current)):
                                continue
                        except ValueError:
                            pass
                    self._store[key] = obj
                    self._last_rv = rv or self._last_rv
                    self._changed.notify_all()

    def stop(self) -> None:
        self._stop.set()
        if self._watch is not None:
            self._watch.stop()
        if self._thread is not None:
            self._thread.join(timeout=5)

    def wait_sync(self, timeout: float = 10.0) -> bool:
        return self._synced.wait(timeout)

    def wait_for_rv(self, name: str, namespace: str, rv: str,
                    timeout: float) -> bool:
        """Block until the cached copy of (namespace, name) has
        resourceVersion >= ``rv``, or the object is absent from the cache
        (deleted), or the timeout elapses.  Event-driven via the informer's
        condition variable — no polling."""
        try:
            target = int(rv)
        except (TypeError, ValueError):
            return False
        key = (namespace, name)
        deadline = time.monotonic() + timeout

        def caught_up() -> bool:
            obj = self._store.get(key)
            if obj is None:
                return True  # deleted (or never seen): nothing newer coming
            try:
                return int(meta.resource_version(obj)) >= target
            except ValueError:
                return False

        with self._changed:
            while not caught_up():
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    return False
                self._changed.wait(remaining)
            return True

    def get(self, name: str, namespace: str) -> K8sObject:
        with self._lock:
            obj = self._store.get((namespace, name))
            if obj is None:
                raise NotFoundError(f"{self.kind} {namespace}/{name} not found (cache)")
            return meta.deep_copy(obj)

    def list(self, namespace: Optional[str], label_selector: str,
             field_selector: str) -> List[K8sObject]:
        lsel = meta.parse_label_selector(label_selector)
        fsel = meta.parse_field_selector(field_selector)
        out = []
        with self._lock:
            for (ns, _), obj in self._store.items():
                if namespace not in (None, "") and ns != namespace:
                    continue
                if label_selector and not lsel.matches_object(obj):
                    continue
                if field_selector and not fsel.matches_object(obj):
                    continue
                out.append(meta.deep_copy(obj))
        out.sort(key=lambda o: (meta.namespace(o), meta.name(o)))
        return out


class CachedClient(Client):
    """Read-from-cache, write-through client."""

    def __init__(self, delegate: Client, sync_delay: float = 0.0) -> None:
        self._delegate = delegate
        self._sync_delay = sync_delay
        self._informers: Dict[Tuple[str, str], _Informer] = {}
        self._lock = threading.Lock()

    # expose the underlying cluster when the delegate has one (tests)
    @property
    def cluster(self):
        return getattr(self._delegate, "cluster", None)

    def _informer_for(self, api_version: str, kind: str) -> _Informer:
        key = (api_version, kind)
        with self._lock:
            inf = self._informers.get(key)
            if inf is None:
                inf = _Informer(self._delegate, api_version, kind, self._sync_delay)
                self._informers[key] = inf
                inf.start()
        inf.wait_sync()
        return inf

    def wait_for_resource_version(self, api_version: str, kind: str,
                                  name: str, namespace: str, rv: str,
                                  timeout: float = 10.0) -> bool:
        """Block until this cache has caught up to ``rv`` for the given
        object (event-driven).  This is the fast path of the state
        provider's patch-then-confirm barrier: the patch response's
        resourceVersion is the exact point the cache must reach before the
        next reconcile may trust its reads."""
        return self._informer_for(api_version, kind).wait_for_rv(
            name, namespace, rv, timeout
        )

    def wait_for_cache_sync(self, timeout: float = 10.0) -> bool:
        with self._lock:
            informers = list(self._informers.values())
        return all(inf.wait_sync(timeout) for inf in informers)

    def stop(self) -> None:
        with self._lock:
            informers = list(self._informers.values())
            self._informers.clear()
        for inf in informers:
            inf.stop()

    # -- reads: cache --------------------------------------------------------

    def get(self, api_version, kind, name, namespace=""):
        return self._informer_for(api_version, kind).get(name, namespace)

    def list(self, api_version, kind, namespace=None, label_selector="", field_selector=""):
        return self._informer_for(api_version, kind).list(
            namespace, label_selector, field_selector
        )

    # -- writes: pass-through ------------------------------------------------

    def create(self, obj):
        return self._delegate.create(obj)

    def update(self, obj):
        return self._delegate.update(obj)

    def patch(self, api_version, kind, name, patch, namespace=""):
        return self._delegate.patch(api_version, kind, name, patch, namespace)

    def delete(self, api_version, kind, name, namespace="", grace_period_seconds=None):
        try:
            self._delegate.delete(api_version, kind, name, namespace,
                                  grace_period_seconds=grace_period_seconds)
        except TypeError:
            self._delegate.delete(api_version, kind, name, namespace)

    def evict_pod(self, name, namespace):
        self._delegate.evict_pod(name, namespace)

    def patch_status(self, api_version, kind, name, status, namespace=""):
        return self._delegate.patch_status(api_version, kind, name, status,
                                           namespace)

    def watch(self, api_version, kind, namespace=None, resource_version=None,
              label_selector=""):
        return self._delegate.watch(
            api_version, kind, namespace=namespace,
            resource_version=resource_version, label_selector=label_selector,
        )

    def list_with_meta(self, api_version, kind, namespace=None,
                       label_selector="", field_selector=""):
        return self._delegate.list_with_meta(
            api_version, kind, namespace=namespace,
            label_selector=label_selector, field_selector=field_selector,
        )
