"""Driver-name registry, key getters, and concurrency primitives.

Capability parity with the reference's ``pkg/upgrade/util.go``:

- a package-global driver name with validation (``util.go:91-99``);
- key-getter functions resolving the ``amd.com/<driver>-...`` label and
  annotation keys (``util.go:106-155``);
- a thread-safe :class:`StringSet` used as an in-progress guard by the async
  drain / pod managers (``util.go:29-70``);
- a per-key :class:`KeyedMutex` serializing node label/annotation mutations
  (``util.go:72-89``).

The drain / pod / wait managers run one worker thread per node (the Python
analogue of the reference's goroutines); these primitives make that safe.
"""

from __future__ import annotations

import re
import threading
from typing import Iterator

from . import consts

_DRIVER_NAME_LOCK = threading.Lock()
_driver_name = "amdgpu"

# Label keys must be valid Kubernetes label name components once formatted;
# restrict driver names the same way the reference's consumers do.
_DRIVER_NAME_RE = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")


class InvalidDriverNameError(ValueError):
    pass


def set_driver_name(name: str) -> None:
    """Set the package-global driver name (``util.go:96-99``).

    Must be called once at operator startup before the state manager is built.
    Typical AMD values: ``amdgpu`` (kernel driver / dkms), ``rocm`` (user-space
    stack), ``anic`` (AMD/Pensando NIC stack — the xGMI/IF NIC analogue of the
    reference's OFED path).
    """
    if not _DRIVER_NAME_RE.match(name):
        raise InvalidDriverNameError(
            f"invalid driver name {name!r}: must match {_DRIVER_NAME_RE.pattern}"
        )
    global _driver_name
    with _DRIVER_NAME_LOCK:
        _driver_name = name


def get_driver_name() -> str:
    with _DRIVER_NAME_LOCK:
        return _driver_name


# -- key getters (util.go:106-155) ------------------------------------------

def get_upgrade_state_label_key() -> str:
    return consts.UPGRADE_STATE_LABEL_KEY_FMT.format(get_driver_name())


def get_upgrade_skip_node_label_key() -> str:
    return consts.UPGRADE_SKIP_NODE_LABEL_KEY_FMT.format(get_driver_name())


def get_upgrade_skip_drain_pod_selector() -> str:
    key = consts.UPGRADE_SKIP_DRAIN_POD_SELECTOR_FMT.format(get_driver_name())
    return f"{key}!=true"


def get_upgrade_wait_for_safe_driver_load_annotation_key() -> str:
    return consts.UPGRADE_WAIT_FOR_SAFE_DRIVER_LOAD_ANNOTATION_KEY_FMT.format(
        get_driver_name()
    )


def get_upgrade_initial_state_annotation_key() -> str:
    return consts.UPGRADE_INITIAL_STATE_ANNOTATION_KEY_FMT.format(get_driver_name())


def get_wait_for_pod_completion_start_time_annotation_key() -> str:
    return (
        consts.UPGRADE_WAIT_FOR_POD_COMPLETION_START_TIME_ANNOTATION_KEY_FMT.format(
            get_driver_name()
        )
    )


def get_validation_start_time_annotation_key() -> str:
    return consts.UPGRADE_VALIDATION_START_TIME_ANNOTATION_KEY_FMT.format(
        get_driver_name()
    )


def get_upgrade_requested_annotation_key() -> str:
    return consts.UPGRADE_REQUESTED_ANNOTATION_KEY_FMT.format(get_driver_name())


def get_upgrade_requestor_mode_annotation_key() -> str:
    return consts.UPGRADE_REQUESTOR_MODE_ANNOTATION_KEY_FMT.format(get_driver_name())


def get_event_reason() -> str:
    """Event reason string, e.g. ``AMDGPUDriverUpgrade`` (util.go:157-160)."""
    return f"{get_driver_name().upper()}DriverUpgrade"


# -- concurrency primitives --------------------------------------------------

class StringSet:
    """Thread-safe set of strings (util.go:29-70).

    Used by the drain / pod managers as an "operation already in flight for
    this node" guard so a reconcile tick never schedules the same node twice.
    """

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._items: set[str] = set()

    def add(self, item: str) -> None:
        with self._lock:
            self._items.add(item)

    def add_if_absent(self, item: str) -> bool:
        """Atomically add; return True if the item was newly added."""
        with self._lock:
            if item in self._items:
                return False
            self._items.add(item)
            return True

    def remove(self, item: str) -> None:
        with self._lock:
            self._items.discard(item)

    def has(self, item: str) -> bool:
        with self._lock:
            return item in self._items

    def clear(self) -> None:
        with self._lock:
            self._items.clear()

    def __len__(self) -> int:
        with self._lock:
            return len(self._items)

    def __iter__(self) -> Iterator[str]:
        with self._lock:
            return iter(sorted(self._items))


class KeyedMutex:
    """Per-key mutual exclusion (util.go:72-89).

    ``lock(key)`` returns a context manager; the state provider uses one lock
    per node name so concurrent workers serialize their label/annotation
    mutations per node without a global lock.
    """

    def __init__(self) -> None:
        self._guard = threading.Lock()
        self._locks: dict[str, threading.Lock] = {}

    def _get(self, key: str) -> threading.Lock:
        with self._guard:
            lock = self._locks.get(key)
            if lock is None:
                lock = threading.Lock()
                self._locks[key] = lock
            return lock

    def lock(self, key: str) -> "_KeyedLockCtx":
        return _KeyedLockCtx(self._get(key))

    def acquire(self, key: str) -> None:
        self._get(key).acquire()

    def release(self, key: str) -> None:
        self._get(key).release()


class _KeyedLockCtx:
    def __init__(self, lock: threading.Lock) -> None:
        self._lock = lock

    def __enter__(self) -> None:
        self._lock.acquire()

    def __exit__(self, *exc: object) -> None:
        self._lock.release()
