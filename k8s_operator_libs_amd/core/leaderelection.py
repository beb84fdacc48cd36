"""Lease-based leader election.

The reference's consumers get leader election from controller-runtime's
manager; operators built on this library get the equivalent here: standard
``coordination.k8s.io/v1 Lease`` semantics (acquire if unheld or expired,
renew while leading, release on stop), so only one operator replica runs the
upgrade state machine at a time.

The state machine itself is safe without election (idempotent, per-node
locks, optimistic concurrency — see tests/test_failure_injection.py's
concurrent-managers test); election avoids the wasted work and event noise
of N replicas reconciling.
"""

from __future__ import annotations

import logging
import threading
import time
import uuid
from typing import Callable, Optional

from .client import Client
from .errors import AlreadyExistsError, ConflictError, NotFoundError

logger = logging.getLogger(__name__)

LEASE_API_VERSION = "coordination.k8s.io/v1"
LEASE_KIND = "Lease"


def _now_iso() -> str:
    # MicroTime precision: second-resolution stamps would make sub-second
    # lease durations appear expired spuriously
    from datetime import datetime, timezone

    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%S.%fZ")


def _parse_iso(ts: str) -> float:
    from datetime import datetime, timezone

    try:
        dt = datetime.strptime(ts, "%Y-%m-%dT%H:%M:%S.%fZ")
    except ValueError:
        try:
            dt = datetime.strptime(ts, "%Y-%m-%dT%H:%M:%SZ")
        except ValueError:
            return 0.0
    return dt.replace(tzinfo=timezone.utc).timestamp()


class LeaderElector:
    """Run ``on_started_leading`` while holding the Lease; renew every
    ``retry_period`` seconds; another candidate takes over once
    ``lease_duration`` elapses without renewal."""

    def __init__(
        self,
        client: Client,
        lease_name: str,
        namespace: str = "default",
        identity: Optional[str] = None,
        lease_duration: float = 15.0,
        retry_period: float = 2.0,
    ) -> None:
        self.client = client
        self.lease_name = lease_name
        self.namespace = namespace
        self.identity = identity or f"{uuid.uuid4().hex[:8]}"
        self.lease_duration = lease_duration
        self.retry_period = retry_period
        self._stop = threading.Event()
        self._leading = threading.Event()
        # The last Lease state THIS elector wrote (from the create/patch
        # RESPONSE).  Renewals key their optimistic lock on it instead of a
        # fresh read: over an informer-backed client a read issued right
        # after our own write can be STALE (the watch event has not landed
        # yet), and renewing against the stale resourceVersion used to 409
        # and fake-demote a healthy leader.
        self._held_lease = None

    # -- lease record handling ----------------------------------------------

    def _try_acquire_or_renew(self) -> bool:
        if self._held_lease is not None:
            if self._renew_held():
                return True
            # fell behind (another writer touched the Lease): fall through
            # to the read-evaluate-acquire path with a fresh view
            self._held_lease = None

        try:
            lease = self.client.get(
                LEASE_API_VERSION, LEASE_KIND, self.lease_name, self.namespace
            )
        except NotFoundError:
            lease = {
                "apiVersion": LEASE_API_VERSION,
                "kind": LEASE_KIND,
                "metadata": {"name": self.lease_name, "namespace": self.namespace},
                "spec": {
                    "holderIdentity": self.identity,
                    "leaseDurationSeconds": int(self.lease_duration),
                    "acquireTime": _now_iso(),
                    "renewTime": _now_iso(),
                    "leaseTransitions": 0,
                },
            }
            try:
                self._held_lease = self.client.create(lease)
                return True
            except AlreadyExistsError:
                return False

        spec = lease.get("spec", {})
        holder = spec.get("holderIdentity")
        renew = _parse_iso(spec.get("renewTime", ""))
        expired = (time.time() - renew) > self.lease_duration
        if holder == self.identity or not holder or expired:
            patch = {
                "metadata": {"resourceVersion": lease["metadata"]["resourceVersion"]},
                "spec": {
                    "holderIdentity": self.identity,
                    "renewTime": _now_iso(),
                },
            }
            if holder != self.identity:
                patch["spec"]["acquireTime"] = _now_iso()
                patch["spec"]["leaseTransitions"] = spec.get("leaseTransitions", 0) + 1
            try:
                # optimistic lock: losing a race means someone else renewed
                self._held_lease = self.client.patch(
                    LEASE_API_VERSION, LEASE_KIND, self.lease_name, patch, self.namespace
                )
                return True
            except ConflictError:
                return False
        return False

    def _renew_held(self) -> bool:
        """Renew against our own last-written resourceVersion (no read)."""
        rv = (self._held_lease.get("metadata") or {}).get("resourceVersion")
        if not rv:
            return False
        try:
            self._held_lease = self.client.patch(
                LEASE_API_VERSION, LEASE_KIND, self.lease_name,
                {"metadata": {"resourceVersion": rv},
                 "spec": {"holderIdentity": self.identity,
                          "renewTime": _now_iso()}},
                self.namespace,
            )
            return True
        except (ConflictError, NotFoundError):
            return False

    def _release(self) -> None:
        try:
            # release against the held state first (same staleness rationale
            # as renewals); fall back to a fresh read
            lease = self._held_lease or self.client.get(
                LEASE_API_VERSION, LEASE_KIND, self.lease_name, self.namespace
            )
            if lease.get("spec", {}).get("holderIdentity") == self.identity:
                self.client.patch(
                    LEASE_API_VERSION, LEASE_KIND, self.lease_name,
                    {"metadata": {"resourceVersion": lease["metadata"]["resourceVersion"]},
                     "spec": {"holderIdentity": None, "renewTime": None}},
                    self.namespace,
                )
        except (NotFoundError, ConflictError):
            pass
        finally:
            self._held_lease = None

    # -- public API -----------------------------------------------------------

    def is_leading(self) -> bool:
        return self._leading.is_set()

    def stop(self) -> None:
        self._stop.set()

    def run(
        self,
        on_started_leading: Callable[[], None],
        on_stopped_leading: Optional[Callable[[], None]] = None,
    ) -> None:
        """Block: campaign, then call ``on_started_leading()`` once leading
        (it should run until it observes lost leadership / stop), renewing in
        a background thread.  Returns when stopped."""
        self._on_stopped_leading = on_stopped_leading
        # fires at most once even though both the renew loop (leadership
        # lost mid-work) and run()'s cleanup path reach it
        self._stopped_fired = threading.Event()
        while not self._stop.is_set():
            if self._try_acquire_or_renew():
                logger.info("leader election: %s acquired %s/%s",
                            self.identity, self.namespace, self.lease_name)
                self._leading.set()
                renewer = threading.Thread(target=self._renew_loop, daemon=True)
                renewer.start()
                try:
                    on_started_leading()
                finally:
                    self._leading.clear()
                    renewer.join(timeout=self.retry_period * 2)
                    self._release()
                    self._fire_stopped_leading()
                return
            self._stop.wait(self.retry_period)

    def _fire_stopped_leading(self) -> None:
        cb = getattr(self, "_on_stopped_leading", None)
        fired = getattr(self, "_stopped_fired", None)
        if cb is None or fired is None or fired.is_set():
            return
        fired.set()
        try:
            cb()
        except Exception:
            logger.exception("on_stopped_leading callback failed")

    def _renew_loop(self) -> None:
        while self._leading.is_set() and not self._stop.is_set():
            if not self._try_acquire_or_renew():
                logger.warning("leader election: %s lost %s/%s",
                               self.identity, self.namespace, self.lease_name)
                self._leading.clear()
                # Leadership lost while the leader's work is still running:
                # fire the callback so the work is STOPPED (without this a
                # demoted replica would keep reconciling — split-brain).
                self._fire_stopped_leading()
                return
            self._stop.wait(self.retry_period)
