"""Node drain engine: the kubectl ``drain.Helper`` analogue.

The reference wraps ``k8s.io/kubectl/pkg/drain`` (reference
``drain_manager.go:76-96``, ``pod_manager.go:136-157``).  This module is a
from-scratch implementation of the same filtering and eviction semantics on
top of this library's :class:`~k8s_operator_libs_amd.core.client.Client`:

- pods already terminating are ignored;
- DaemonSet-managed pods are skipped (``IgnoreAllDaemonSets``) — the driver
  DaemonSet pod itself must survive the drain;
- mirror (static) pods are skipped;
- pods with no controller owner block the drain unless ``force``;
- pods using emptyDir volumes block the drain unless ``delete_emptydir_data``;
- finished pods (Succeeded/Failed) are always deletable;
- an optional label ``pod_selector`` narrows which pods are considered, and an
  optional ``additional_filter`` callable (the operator's PodDeletionFilter)
  further narrows the eviction set.

AMD operators typically pass an ``additional_filter`` matching pods that
request ``amd.com/gpu`` device-plugin resources (see
:func:`gpu_pod_deletion_filter`).
"""

from __future__ import annotations

import logging
import time
from dataclasses import dataclass, field
from typing import Callable, List, Optional

from ..api.upgrade.v1alpha1 import DrainSpec
from ..core import meta
from ..core.client import Client
from ..core.errors import ApiError, NotFoundError
from . import consts

logger = logging.getLogger(__name__)

MIRROR_POD_ANNOTATION = "kubernetes.io/config.mirror"

PodFilter = Callable[[dict], bool]


@dataclass
class PodDeleteList:
    """Outcome of pod-for-deletion selection (drain.PodDeleteList analogue)."""

    pods: List[dict] = field(default_factory=list)
    skipped: List[dict] = field(default_factory=list)
    warnings: List[str] = field(default_factory=list)
    errors: List[str] = field(default_factory=list)


def _is_finished(pod: dict) -> bool:
    return pod.get("status", {}).get("phase") in ("Succeeded", "Failed")


def _is_terminating(pod: dict) -> bool:
    return "deletionTimestamp" in pod.get("metadata", {})


def _is_daemonset_pod(pod: dict) -> bool:
    owner = meta.controller_owner(pod)
    return bool(owner and owner.get("kind") == "DaemonSet")


def _is_mirror_pod(pod: dict) -> bool:
    return MIRROR_POD_ANNOTATION in (pod.get("metadata", {}).get("annotations") or {})


def _has_controller(pod: dict) -> bool:
    return meta.controller_owner(pod) is not None


def _uses_emptydir(pod: dict) -> bool:
    for vol in pod.get("spec", {}).get("volumes", []) or []:
        if "emptyDir" in vol:
            return True
    return False


def pod_requests_resource(pod: dict, resource_prefix: str) -> bool:
    """True if any container requests/limits a resource whose name starts
    with ``resource_prefix`` (e.g. ``amd.com/gpu``)."""
    for container in pod.get("spec", {}).get("containers", []) or []:
        resources = container.get("resources", {}) or {}
        for kind_ in ("requests", "limits"):
            for res_name in (resources.get(kind_) or {}):
                if res_name.startswith(resource_prefix):
                    return True
    return False


def gpu_pod_deletion_filter(pod: dict) -> bool:
    """Default AMD GPU workload filter: pods consuming ``amd.com/gpu*``
    device-plugin resources (the reference's example filter keys on
    ``nvidia.com/gpu`` / ``nvidia.com/mig-*`` — pod_manager_test.go:435-450)."""
    return pod_requests_resource(pod, "amd.com/gpu")


def get_pods_for_deletion(
    client: Client,
    node_name: str,
    *,
    force: bool = False,
    delete_emptydir_data: bool = False,
    ignore_daemonsets: bool = True,
    pod_selector: str = "",
    additional_filter: Optional[PodFilter] = None,
) -> PodDeleteList:
    """Classify every pod on ``node_name`` into deletable / skipped / error,
    mirroring kubectl drain's filter chain."""
    pods = client.list_pods(
        label_selector=pod_selector,
        field_selector=consts.NODE_NAME_FIELD_SELECTOR_FMT.format(node_name),
    )
    out = PodDeleteList()
    for pod in pods:
        pname = f"{meta.namespace(pod)}/{meta.name(pod)}"
        if _is_terminating(pod):
            out.skipped.append(pod)
            continue
        if additional_filter is not None and not additional_filter(pod):
            out.skipped.append(pod)
            continue
        if _is_mirror_pod(pod):
            out.skipped.append(pod)
            out.warnings.append(f"skipping mirror pod {pname}")
            continue
        if _is_daemonset_pod(pod):
            if ignore_daemonsets:
                out.skipped.append(pod)
                out.warnings.append(f"ignoring DaemonSet-managed pod {pname}")
                continue
            out.errors.append(f"cannot delete DaemonSet-managed pod {pname}")
            continue
        if _is_finished(pod):
            out.pods.append(pod)
            continue
        if not _has_controller(pod) and not force:
            out.errors.append(
                f"cannot delete pod not managed by a controller (use force): {pname}"
            )
            continue
        if _uses_emptydir(pod) and not delete_emptydir_data:
            out.errors.append(
                f"cannot delete pod with emptyDir volume (use deleteEmptyDir): {pname}"
            )
            continue
        out.pods.append(pod)
    return out


class DrainError(Exception):
    pass


def delete_or_evict_pods(
    client: Client,
    pods: List[dict],
    *,
    use_eviction: bool = True,
    timeout_seconds: float = 300,
) -> None:
    """Evict (or delete) the given pods and wait until they are gone.

    With the in-memory apiserver eviction completes immediately (no kubelet,
    like envtest); against a real cluster the wait loop polls with backoff up
    to ``timeout_seconds``.
    """
    deadline = time.monotonic() + max(timeout_seconds, 0.001)

    def try_evict(pod) -> bool:
        """True when the evict/delete was accepted; False when a
        PodDisruptionBudget blocked it (429) — retried until the deadline,
        like kubectl drain."""
        try:
            if use_eviction:
                client.evict_pod(meta.name(pod), meta.namespace(pod))
            else:
                client.delete_pod(meta.name(pod), meta.namespace(pod))
            return True
        except NotFoundError:
            return True
        except ApiError as exc:
            if exc.code == 429:
                return False
            raise

    blocked = [pod for pod in pods if not try_evict(pod)]
    interval = 0.001
    remaining = list(pods)
    while remaining:
        # PDB-blocked pods: another replica may have become healthy; retry
        blocked = [pod for pod in blocked if not try_evict(pod)]
        still_there = []
        for pod in remaining:
            try:
                live = client.get("v1", "Pod", meta.name(pod), meta.namespace(pod))
            except NotFoundError:
                continue
            # A replacement pod with the same name but new UID doesn't count.
            if meta.uid(live) == meta.uid(pod):
                still_there.append(pod)
        remaining = still_there
        if not remaining:
            return
        if time.monotonic() >= deadline:
            names = [f"{meta.namespace(p)}/{meta.name(p)}" for p in remaining]
            if blocked:
                names = [f"{n} (PDB-blocked)" if any(
                    meta.name(b) == n.split("/")[-1] for b in blocked) else n
                    for n in names]
            raise DrainError(f"timed out waiting for pods to terminate: {names}")
        time.sleep(interval)
        interval = min(interval * 2, 0.5)


def drain_node(client: Client, node_name: str, spec: DrainSpec) -> None:
    """Full node drain per DrainSpec (cordon is the caller's job).

    Raises :class:`DrainError` if any pod blocks the drain or eviction times
    out — the DrainManager maps that to the upgrade-failed state.
    """
    plist = get_pods_for_deletion(
        client,
        node_name,
        force=spec.force,
        delete_emptydir_data=spec.delete_emptydir_data,
        ignore_daemonsets=True,
        pod_selector=spec.pod_selector,
    )
    if plist.errors:
        raise DrainError("; ".join(plist.errors))
    for warning in plist.warnings:
        logger.debug("drain %s: %s", node_name, warning)
    delete_or_evict_pods(
        client, plist.pods, use_eviction=True, timeout_seconds=spec.timeout_seconds
    )
