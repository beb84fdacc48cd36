"""The typed ``Client`` interface (controller-runtime ``client.Client`` analogue)
and its in-memory implementation.

Every upgrade manager takes a :class:`Client`; tests and benchmarks inject a
:class:`FakeClient` over a :class:`~k8s_operator_libs_amd.core.fakecluster.FakeCluster`
(the envtest substrate), production injects the httpx
:class:`~k8s_operator_libs_amd.core.restclient.RestClient`.
"""

from __future__ import annotations

import abc
from typing import List, Optional

from .fakecluster import FakeCluster, Watch
from .meta import K8sObject


class Client(abc.ABC):
    """Generic CRUD against the cluster, dict-shaped objects on the wire."""

    @abc.abstractmethod
    def get(self, api_version: str, kind: str, name: str, namespace: str = "") -> K8sObject: ...

    @abc.abstractmethod
    def list(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
    ) -> List[K8sObject]: ...

    @abc.abstractmethod
    def create(self, obj: K8sObject) -> K8sObject: ...

    @abc.abstractmethod
    def update(self, obj: K8sObject) -> K8sObject: ...

    @abc.abstractmethod
    def patch(
        self, api_version: str, kind: str, name: str, patch: K8sObject, namespace: str = ""
    ) -> K8sObject: ...

    @abc.abstractmethod
    def delete(self, api_version: str, kind: str, name: str, namespace: str = "") -> None:
        """Delete an object.  Implementations may accept a
        ``grace_period_seconds`` keyword (pod graceful-deletion override);
        in-memory substrates delete immediately and ignore it."""

    @abc.abstractmethod
    def evict_pod(self, name: str, namespace: str) -> None: ...

    def watch(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        resource_version: Optional[str] = None,
        label_selector: str = "",
    ) -> Watch:
        """Open a watch stream.  ``resource_version`` anchors the stream per
        Kubernetes semantics (None = live-only, "0" = synthetic ADDEDs then
        live, otherwise replay-after-RV with 410
        :class:`~k8s_operator_libs_amd.core.errors.GoneError` when the
        resume window has expired)."""
        raise NotImplementedError("this client does not support watches")

    def list_with_meta(
        self,
        api_version: str,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
    ):
        """LIST returning ``(items, list_resource_version)`` — the anchor for
        a lossless LIST-then-WATCH(rv) reflector loop.  Clients without
        list-level RV support return ``(items, None)`` and the informer falls
        back to live-only watches."""
        return (
            self.list(api_version, kind, namespace=namespace,
                      label_selector=label_selector, field_selector=field_selector),
            None,
        )

    def patch_status(
        self, api_version: str, kind: str, name: str, status: dict, namespace: str = ""
    ) -> K8sObject:
        """Patch only the status subresource (controller-runtime
        ``Status().Patch`` analogue).  Default implementation merges via the
        main resource; clients talking to real apiservers override with the
        ``/status`` endpoint."""
        return self.patch(api_version, kind, name, {"status": status}, namespace)

    # -- typed conveniences used throughout pkg upgrade ----------------------

    def get_node(self, name: str) -> K8sObject:
        return self.get("v1", "Node", name)

    def list_nodes(self, label_selector: str = "") -> List[K8sObject]:
        return self.list("v1", "Node", label_selector=label_selector)

    def list_pods(
        self,
        namespace: Optional[str] = None,
        label_selector: str = "",
        field_selector: str = "",
    ) -> List[K8sObject]:
        return self.list(
            "v1", "Pod", namespace=namespace,
            label_selector=label_selector, field_selector=field_selector,
        )

    def list_daemonsets(
        self, namespace: Optional[str] = None, label_selector: str = ""
    ) -> List[K8sObject]:
        return self.list(
            "apps/v1", "DaemonSet", namespace=namespace, label_selector=label_selector
        )

    def list_controller_revisions(
        self, namespace: Optional[str] = None, label_selector: str = ""
    ) -> List[K8sObject]:
        return self.list(
            "apps/v1", "ControllerRevision",
            namespace=namespace, label_selector=label_selector,
        )

    def delete_pod(self, name: str, namespace: str) -> None:
        self.delete("v1", "Pod", name, namespace)


class FakeClient(Client):
    """Direct in-process client over a :class:`FakeCluster`."""

    def __init__(self, cluster: Optional[FakeCluster] = None) -> None:
        self.cluster = cluster or FakeCluster()

    def get(self, api_version, kind, name, namespace=""):
        return self.cluster.get(api_version, kind, name, namespace)

    def list(self, api_version, kind, namespace=None, label_selector="", field_selector=""):
        return self.cluster.list(
            api_version, kind, namespace=namespace,
            label_selector=label_selector, field_selector=field_selector,
        )

    def create(self, obj):
        return self.cluster.create(obj)

    def update(self, obj):
        return self.cluster.update(obj)

    def patch(self, api_version, kind, name, patch, namespace=""):
        return self.cluster.patch(api_version, kind, name, patch, namespace)

    def delete(self, api_version, kind, name, namespace="", grace_period_seconds=None):
        # in-memory deletion is immediate; grace period has no kubelet to honour
        self.cluster.delete(api_version, kind, name, namespace)

    def evict_pod(self, name, namespace):
        self.cluster.evict_pod(name, namespace)

    def patch_status(self, api_version, kind, name, status, namespace=""):
        return self.cluster.patch_status(api_version, kind, name, status,
                                         namespace)

    def watch(self, api_version, kind, namespace=None, resource_version=None,
              label_selector="", field_selector="", send_initial_events=False):
        return self.cluster.watch(
            api_version, kind, namespace=namespace,
            resource_version=resource_version, label_selector=label_selector,
            field_selector=field_selector,
            send_initial_events=send_initial_events,
        )

    def list_with_meta(self, api_version, kind, namespace=None,
                       label_selector="", field_selector=""):
        return self.cluster.list_with_meta(
            api_version, kind, namespace=namespace,
            label_selector=label_selector, field_selector=field_selector,
        )
