"""Stateful test doubles for the L3 manager interfaces.

Capability parity with the reference's mockery-generated ``pkg/upgrade/mocks``
(CordonManager, DrainManager, NodeUpgradeStateProvider, PodManager,
ValidationManager).  Rather than mechanical method recorders these follow the
reference *test suites'* pattern of stateful fakes — e.g. the mocked
``change_node_upgrade_state`` mutates the label in memory
(upgrade_suit_test.go:114-130) — so state-machine tests run without any
cluster substrate.  Every call is also recorded in ``.calls`` for
assertion, and any method can be overridden with ``side_effect``.
"""

from __future__ import annotations

import threading
from typing import Any, Callable, Dict, List, Optional, Tuple

from ..core import meta
from ..core.meta import K8sObject
from . import consts, util


class _RecorderBase:
    def __init__(self) -> None:
        self.calls: List[Tuple[str, tuple]] = []
        self._lock = threading.Lock()
        self.side_effects: Dict[str, Callable] = {}
        self.failures: Dict[str, Exception] = {}

    def _record(self, method: str, *args: Any) -> None:
        with self._lock:
            self.calls.append((method, args))
        exc = self.failures.get(method)
        if exc is not None:
            raise exc

    def calls_to(self, method: str) -> List[tuple]:
        with self._lock:
            return [a for m, a in self.calls if m == method]


class MockNodeUpgradeStateProvider(_RecorderBase):
    """Mutates labels/annotations on the in-memory node objects."""

    def __init__(self) -> None:
        super().__init__()
        self.nodes: Dict[str, K8sObject] = {}
        self.metrics = None

    def register(self, node: K8sObject) -> K8sObject:
        self.nodes[meta.name(node)] = node
        return node

    def get_node(self, name: str) -> K8sObject:
        self._record("get_node", name)
        return self.nodes[name]

    def change_node_upgrade_state(self, node: K8sObject, new_state: str) -> None:
        self._record("change_node_upgrade_state", meta.name(node), new_state)
        meta.labels(node)[util.get_upgrade_state_label_key()] = new_state
        registered = self.nodes.get(meta.name(node))
        if registered is not None and registered is not node:
            meta.labels(registered)[util.get_upgrade_state_label_key()] = new_state

    def change_node_upgrade_annotation(self, node: K8sObject, key: str, value: str) -> None:
        self._record("change_node_upgrade_annotation", meta.name(node), key, value)
        for target in {id(node): node,
                       id(self.nodes.get(meta.name(node))): self.nodes.get(meta.name(node))}.values():
            if target is None:
                continue
            if value == consts.NULL_STRING:
                meta.annotations(target).pop(key, None)
            else:
                meta.annotations(target)[key] = value


class MockCordonManager(_RecorderBase):
    def cordon(self, node: K8sObject) -> None:
        self._record("cordon", meta.name(node))
        node.setdefault("spec", {})["unschedulable"] = True

    def uncordon(self, node: K8sObject) -> None:
        self._record("uncordon", meta.name(node))
        node.setdefault("spec", {}).pop("unschedulable", None)


class MockDrainManager(_RecorderBase):
    def __init__(self, provider: Optional[MockNodeUpgradeStateProvider] = None,
                 outcome: str = consts.UPGRADE_STATE_POD_RESTART_REQUIRED) -> None:
        super().__init__()
        self.provider = provider
        self.outcome = outcome

    def schedule_nodes_drain(self, config) -> None:
        self._record("schedule_nodes_drain",
                     tuple(meta.name(n) for n in config.nodes))
        if self.provider is not None and config.spec is not None and config.spec.enable:
            for node in config.nodes:
                self.provider.change_node_upgrade_state(node, self.outcome)

    def wait_idle(self, timeout: float = 0.0) -> None:
        self._record("wait_idle")


class MockPodManager(_RecorderBase):
    def __init__(self, provider: Optional[MockNodeUpgradeStateProvider] = None,
                 pod_hashes: Optional[Dict[str, str]] = None,
                 ds_hash: str = "rev") -> None:
        super().__init__()
        self.provider = provider
        self.pod_hashes = pod_hashes or {}
        self.ds_hash = ds_hash
        self.restarted: List[str] = []

    def get_pod_controller_revision_hash(self, pod: K8sObject) -> str:
        self._record("get_pod_controller_revision_hash", meta.name(pod))
        if meta.name(pod) in self.pod_hashes:
            return self.pod_hashes[meta.name(pod)]
        return meta.get_label(pod, "controller-revision-hash") or self.ds_hash

    def get_daemonset_controller_revision_hash(self, ds: K8sObject) -> str:
        self._record("get_daemonset_controller_revision_hash", meta.name(ds))
        return self.ds_hash

    def schedule_pod_eviction(self, config) -> None:
        self._record("schedule_pod_eviction",
                     tuple(meta.name(n) for n in config.nodes))
        if self.provider is not None:
            for node in config.nodes:
                self.provider.change_node_upgrade_state(
                    node, consts.UPGRADE_STATE_POD_RESTART_REQUIRED
                )

    def schedule_pods_restart(self, pods) -> None:
        self._record("schedule_pods_restart", tuple(meta.name(p) for p in pods))
        self.restarted.extend(meta.name(p) for p in pods)

    def schedule_check_on_pod_completion(self, config) -> None:
        self._record("schedule_check_on_pod_completion",
                     tuple(meta.name(n) for n in config.nodes))
        if self.provider is not None:
            for node in config.nodes:
                self.provider.change_node_upgrade_state(
                    node, consts.UPGRADE_STATE_POD_DELETION_REQUIRED
                )

    def handle_timeout_on_pod_completions(self, node, timeout_seconds) -> None:
        self._record("handle_timeout_on_pod_completions", meta.name(node), timeout_seconds)

    @staticmethod
    def is_pod_running_or_pending(pod: K8sObject) -> bool:
        return pod.get("status", {}).get("phase") in ("Running", "Pending")

    def wait_idle(self, timeout: float = 0.0) -> None:
        self._record("wait_idle")


class MockValidationManager(_RecorderBase):
    def __init__(self, result: bool = True, pod_selector: str = "mock") -> None:
        super().__init__()
        self.result = result
        self._pod_selector = pod_selector

    @property
    def pod_selector(self) -> str:
        return self._pod_selector

    def validate(self, node: K8sObject) -> bool:
        self._record("validate", meta.name(node))
        return self.result


class MockSafeDriverLoadManager(_RecorderBase):
    def __init__(self, waiting_nodes: Optional[set] = None) -> None:
        super().__init__()
        self.waiting_nodes = waiting_nodes or set()

    def is_waiting_for_safe_driver_load(self, node: K8sObject) -> bool:
        self._record("is_waiting_for_safe_driver_load", meta.name(node))
        return meta.name(node) in self.waiting_nodes

    def unblock_loading(self, node: K8sObject) -> None:
        self._record("unblock_loading", meta.name(node))
        self.waiting_nodes.discard(meta.name(node))
