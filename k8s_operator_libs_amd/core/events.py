"""Kubernetes Event recording (``record.EventRecorder`` analogue).

The reference records an Event on every state transition and failure via a
nil-safe helper (``pkg/upgrade/util.go:162-176``,
``node_upgrade_state_provider.go:123-131``).  :class:`EventRecorder` writes
real ``v1`` Event objects through a client; :class:`FakeRecorder` captures
formatted strings for assertions, mirroring client-go's FakeRecorder.
All helpers tolerate a ``None`` recorder.
"""

from __future__ import annotations

import threading
import uuid
from typing import List, Optional

from . import meta
from .client import Client
from .meta import K8sObject

EVENT_TYPE_NORMAL = "Normal"
EVENT_TYPE_WARNING = "Warning"


class EventRecorder:
    def __init__(self, client: Client, namespace: str = "default", component: str = "amd-upgrade") -> None:
        self._client = client
        self._namespace = namespace
        self._component = component

    def event(self, obj: K8sObject, event_type: str, reason: str, message: str) -> None:
        ns = meta.namespace(obj) or self._namespace
        ev = {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {"name": f"{meta.name(obj)}.{uuid.uuid4().hex[:10]}", "namespace": ns},
            "involvedObject": {
                "apiVersion": meta.api_version(obj),
                "kind": meta.kind(obj),
                "name": meta.name(obj),
                "namespace": meta.namespace(obj),
                "uid": meta.uid(obj),
            },
            "type": event_type,
            "reason": reason,
            "message": message,
            "source": {"component": self._component},
            "count": 1,
        }
        try:
            self._client.create(ev)
        except Exception:
            # Event recording is best-effort, never fails the caller.
            pass

    def eventf(self, obj: K8sObject, event_type: str, reason: str, fmt: str, *args: object) -> None:
        self.event(obj, event_type, reason, fmt.format(*args) if args else fmt)


class FakeRecorder:
    """Captures events as ``"<type> <reason> <message>"`` strings."""

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self.events: List[str] = []

    def event(self, obj: K8sObject, event_type: str, reason: str, message: str) -> None:
        with self._lock:
            self.events.append(f"{event_type} {reason} {message}")

    def eventf(self, obj: K8sObject, event_type: str, reason: str, fmt: str, *args: object) -> None:
        self.event(obj, event_type, reason, fmt.format(*args) if args else fmt)


def log_event(recorder: Optional[object], obj: K8sObject, event_type: str, reason: str, message: str) -> None:
    """Nil-safe event helper (util.go:162-176)."""
    if recorder is None or obj is None:
        return
    recorder.event(obj, event_type, reason, message)  # type: ignore[attr-defined]


def log_eventf(recorder: Optional[object], obj: K8sObject, event_type: str, reason: str, fmt: str, *args: object) -> None:
    if recorder is None or obj is None:
        return
    recorder.eventf(obj, event_type, reason, fmt, *args)  # type: ignore[attr-defined]
