"""CRD apply/delete implementation.

Reference call stack (``pkg/crdutil/crdutil.go``):

- ``walk_crd_paths`` — recursive ``*.yaml`` / ``*.yml`` discovery accepting
  files or directories (crdutil.go:126-154);
- ``parse_crds_from_paths`` — multi-document YAML reader that skips non-CRD
  documents (crdutil.go:157-211);
- ``apply_crds`` — create-or-update with conflict retry + ResourceVersion
  copy (crdutil.go:214-249), then ``wait_for_crds`` polls discovery until the
  plural resource is served (crdutil.go:275-319);
- ``delete_crds`` — idempotent delete (crdutil.go:252-272).

Differences from the reference, deliberate:

- the discovery poll uses exponential backoff starting at 5 ms instead of a
  fixed 100 ms interval, cutting CRD-establish latency on fast apiservers
  while keeping the same 10 s ceiling;
- the client is this library's :class:`~k8s_operator_libs_amd.core.client.Client`
  abstraction, so the same code runs against the in-memory apiserver (tests,
  benchmarks) and a real cluster over REST.
"""

from __future__ import annotations

import logging
import os
import time
from typing import Iterable, List

import yaml

from ..core import meta
from ..core.client import Client
from ..core.errors import AlreadyExistsError, ConflictError, NotFoundError
from ..core.meta import K8sObject

logger = logging.getLogger(__name__)

CRD_OPERATION_APPLY = "apply"
CRD_OPERATION_DELETE = "delete"

CRD_API_VERSION = "apiextensions.k8s.io/v1"
CRD_KIND = "CustomResourceDefinition"

# create-or-update conflict retry budget (client-go retry.DefaultRetry analogue)
_CONFLICT_RETRIES = 5
# wait-for-served poll (crdutil.go:284-286 uses 100 ms / 10 s; we back off
# exponentially from 5 ms to the same 10 s deadline)
_WAIT_TIMEOUT_S = 10.0
_WAIT_INITIAL_S = 0.005
_WAIT_MAX_INTERVAL_S = 0.1


class CrdUtilError(Exception):
    pass


def walk_crd_paths(paths: Iterable[str]) -> List[str]:
    """Expand files/directories into a sorted list of YAML file paths
    (crdutil.go:126-154).  Directories are walked recursively."""
    out: List[str] = []
    for path in paths:
        if not os.path.exists(path):
            raise CrdUtilError(f"CRD path does not exist: {path}")
        if os.path.isfile(path):
            out.append(path)
            continue
        for root, _dirs, files in os.walk(path):
            for fname in files:
                if fname.endswith((".yaml", ".yml")):
                    out.append(os.path.join(root, fname))
    return sorted(out)


def parse_crds_from_file(path: str) -> List[K8sObject]:
    """Read one multi-document YAML file, returning only CRD documents
    (crdutil.go:172-211).  Non-CRD documents are skipped with a debug log."""
    crds: List[K8sObject] = []
    with open(path, "r", encoding="utf-8") as fh:
        try:
            docs = list(yaml.safe_load_all(fh))
        except yaml.YAMLError as exc:
            raise CrdUtilError(f"failed to parse YAML from {path}: {exc}") from exc
    for doc in docs:
        if not isinstance(doc, dict) or not doc:
            continue
        if doc.get("kind") != CRD_KIND or doc.get("apiVersion") != CRD_API_VERSION:
            logger.debug("skipping non-CRD document %s/%s in %s",
                         doc.get("apiVersion"), doc.get("kind"), path)
            continue
        crds.append(doc)
    return crds


def parse_crds_from_paths(paths: Iterable[str]) -> List[K8sObject]:
    crds: List[K8sObject] = []
    for path in walk_crd_paths(paths):
        crds.extend(parse_crds_from_file(path))
    return crds


def apply_crds(client: Client, crds: List[K8sObject], wait: bool = True) -> None:
    """Create-or-update each CRD (crdutil.go:214-249), then wait until all are
    served (crdutil.go:275-319)."""
    for crd in crds:
        _apply_one(client, crd)
    if wait:
        wait_for_crds(client, crds)


def _apply_one(client: Client, crd: K8sObject) -> None:
    name = meta.name(crd)
    for attempt in range(_CONFLICT_RETRIES + 1):
        try:
            existing = client.get(CRD_API_VERSION, CRD_KIND, name)
        except NotFoundError:
            try:
                client.create(crd)
                logger.info("created CRD %s", name)
                return
            except AlreadyExistsError:
                continue  # raced with another applier; retry as update
        desired = meta.deep_copy(crd)
        # Carry over the live ResourceVersion so the update is an optimistic
        # replace of the version we just read (crdutil.go:236-243).
        desired.setdefault("metadata", {})["resourceVersion"] = meta.resource_version(existing)
        try:
            client.update(desired)
            logger.info("updated CRD %s", name)
            return
        except ConflictError:
            if attempt == _CONFLICT_RETRIES:
                raise
            time.sleep(0.01 * (attempt + 1))
    # every attempt hit the create/AlreadyExists race: surface it rather
    # than silently skipping the apply
    raise CrdUtilError(
        f"CRD {name}: create kept racing with a concurrent applier "
        f"after {_CONFLICT_RETRIES + 1} attempts"
    )


def delete_crds(client: Client, crds: List[K8sObject]) -> None:
    """Idempotent delete (crdutil.go:252-272)."""
    for crd in crds:
        name = meta.name(crd)
        try:
            client.delete(CRD_API_VERSION, CRD_KIND, name)
            logger.info("deleted CRD %s", name)
        except NotFoundError:
            logger.info("CRD %s already absent", name)


def _served_group_versions(crd: K8sObject) -> List[tuple]:
    spec = crd.get("spec", {})
    group = spec.get("group", "")
    plural = spec.get("names", {}).get("plural", "")
    return [
        (f"{group}/{v['name']}", plural)
        for v in spec.get("versions", [])
        if v.get("served", True)
    ]


def wait_for_crds(client: Client, crds: List[K8sObject], timeout: float = _WAIT_TIMEOUT_S) -> None:
    """Poll until every served group-version of every CRD exposes its plural
    resource (discovery-based readiness, crdutil.go:275-319).  Exponential
    backoff from 5 ms; raises CrdUtilError on timeout."""
    pending = {gv for crd in crds for gv in _served_group_versions(crd)}
    deadline = time.monotonic() + timeout
    interval = _WAIT_INITIAL_S
    while pending:
        served = {gv for gv in pending if _is_served(client, *gv)}
        pending -= served
        if not pending:
            return
        if time.monotonic() >= deadline:
            raise CrdUtilError(f"timed out waiting for CRDs to be served: {sorted(pending)}")
        time.sleep(interval)
        interval = min(interval * 2, _WAIT_MAX_INTERVAL_S)


def _is_served(client: Client, api_version: str, plural: str) -> bool:
    discover = getattr(client, "discover_resource", None)
    if discover is not None:
        return bool(discover(api_version, plural))
    # FakeClient path: consult the cluster's kind registry directly.
    cluster = getattr(client, "cluster", None)
    if cluster is not None:
        return cluster.lookup_by_plural(api_version, plural) is not None
    raise CrdUtilError("client supports neither discovery nor a kind registry")


def process_crds(client: Client, crd_paths: Iterable[str], operation: str) -> int:
    """Top-level entry point (crdutil.go:44-121): parse all CRDs under the
    given paths and apply or delete them.  Returns the number of CRDs
    processed."""
    crds = parse_crds_from_paths(crd_paths)
    if not crds:
        raise CrdUtilError(f"no CRDs found under {list(crd_paths)}")
    if operation == CRD_OPERATION_APPLY:
        apply_crds(client, crds)
    elif operation == CRD_OPERATION_DELETE:
        delete_crds(client, crds)
    else:
        raise CrdUtilError(f"unknown operation {operation!r} (use apply|delete)")
    return len(crds)
