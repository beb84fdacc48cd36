"""Requestor upgrade mode: delegate node maintenance to an external operator.

Capability parity with the reference's ``pkg/upgrade/upgrade_requestor.go``.
Instead of cordoning/draining itself, the library creates
``maintenance.amd.com/v1alpha1 NodeMaintenance`` objects and watches their
conditions; an external maintenance operator performs cordon, wait-for-pods
and drain, then marks the object Ready.  Throttling (maxParallelUpgrades) is
the maintenance operator's job in this mode — the reference intentionally
applies no gating here (upgrade_requestor.go:277-319).

Shared-requestor flow: when several operators (e.g. GPU + NIC) share the
default name prefix, the second operator does not create a duplicate object
— it appends its requestor ID to ``spec.additionalRequestors`` with an
optimistic-lock merge patch, and symmetrically removes itself (or deletes the
object if it is the owner) on uncordon (upgrade_requestor.go:320-410).
"""

from __future__ import annotations

import logging
import os
from dataclasses import dataclass, field as dc_field
from typing import List, Optional

from ..api.upgrade.v1alpha1 import DriverUpgradePolicySpec
from ..core import meta
from ..core.errors import AlreadyExistsError, NotFoundError
from ..core.meta import K8sObject
from . import consts, util
from .common_manager import (
    ClusterUpgradeState,
    CommonUpgradeManager,
    NodeUpgradeState,
    is_node_in_requestor_mode,
)

logger = logging.getLogger(__name__)

NODE_MAINTENANCE_API_VERSION = "maintenance.amd.com/v1alpha1"
NODE_MAINTENANCE_KIND = "NodeMaintenance"

# Default pod-eviction filters for the AMD maintenance operator: GPU-operator
# pods consuming device-plugin resources, and the AMD/Pensando NIC (RDMA over
# xGMI/IF fabric) resources — mirror of upgrade_requestor.go:47-50.
MAINTENANCE_OP_EVICTION_GPU = "amd.com/gpu-*"
MAINTENANCE_OP_EVICTION_RDMA = "amd.com/rdma*"
# Default NodeMaintenance name prefix shared by AMD operators.
DEFAULT_NODE_MAINTENANCE_NAME_PREFIX = "amd-operator"

# Condition signalled by the maintenance operator when the node is quiesced.
CONDITION_REASON_READY = "Ready"


class NodeMaintenanceUpgradeDisabledError(Exception):
    pass


@dataclass
class RequestorOptions:
    """(upgrade_requestor.go:68-82)"""

    use_maintenance_operator: bool = False
    requestor_id: str = ""
    namespace: str = "default"
    name_prefix: str = DEFAULT_NODE_MAINTENANCE_NAME_PREFIX
    pod_eviction_filters: List[dict] = dc_field(default_factory=list)


def get_requestor_opts_from_envs() -> RequestorOptions:
    """Read MAINTENANCE_OPERATOR_* env vars (upgrade_requestor.go:527-546)."""
    return RequestorOptions(
        use_maintenance_operator=(
            os.environ.get("MAINTENANCE_OPERATOR_ENABLED") == consts.TRUE_STRING
        ),
        requestor_id=os.environ.get("MAINTENANCE_OPERATOR_REQUESTOR_ID", ""),
        namespace=os.environ.get("MAINTENANCE_OPERATOR_REQUESTOR_NAMESPACE") or "default",
        name_prefix=(
            os.environ.get("MAINTENANCE_OPERATOR_NODE_MAINTENANCE_PREFIX")
            or DEFAULT_NODE_MAINTENANCE_NAME_PREFIX
        ),
    )


def convert_policy_to_maintenance_spec(
    upgrade_policy: Optional[DriverUpgradePolicySpec], opts: RequestorOptions
) -> tuple:
    """Map the upgrade policy onto NodeMaintenance drain/wait specs
    (upgrade_requestor.go:493-523)."""
    if upgrade_policy is None:
        return None, None
    drain_spec: dict = {}
    if upgrade_policy.drain_spec is not None:
        drain_spec = {
            "force": upgrade_policy.drain_spec.force,
            "podSelector": upgrade_policy.drain_spec.pod_selector,
            "timeoutSeconds": upgrade_policy.drain_spec.timeout_seconds,
            "deleteEmptyDir": upgrade_policy.drain_spec.delete_emptydir_data,
        }
    if upgrade_policy.pod_deletion is not None:
        drain_spec["podEvictionFilters"] = list(opts.pod_eviction_filters)
    pod_completion = None
    if upgrade_policy.wait_for_completion is not None:
        pod_completion = {
            "podSelector": upgrade_policy.wait_for_completion.pod_selector,
            "timeoutSeconds": upgrade_policy.wait_for_completion.timeout_seconds,
        }
    return drain_spec or None, pod_completion


def find_status_condition(obj: K8sObject, cond_type: str) -> Optional[dict]:
    for cond in obj.get("status", {}).get("conditions", []) or []:
        if cond.get("type") == cond_type:
            return cond
    return None


class RequestorNodeStateManager:
    """(RequestorNodeStateManagerImpl, upgrade_requestor.go:86-89)"""

    def __init__(self, common: CommonUpgradeManager, opts: RequestorOptions) -> None:
        if not opts.use_maintenance_operator:
            raise NodeMaintenanceUpgradeDisabledError(
                "node maintenance upgrade mode is disabled"
            )
        self.common = common
        self.opts = opts
        self._default_nm_spec: dict = {}

    # -- NodeMaintenance object helpers --------------------------------------

    def get_node_maintenance_name(self, node_name: str) -> str:
        return f"{self.opts.name_prefix}-{node_name}"

    def set_default_node_maintenance(
        self, upgrade_policy: Optional[DriverUpgradePolicySpec]
    ) -> None:
        """(SetDefaultNodeMaintenance, upgrade_requestor.go:161-174)"""
        drain_spec, pod_completion = convert_policy_to_maintenance_spec(
            upgrade_policy, self.opts
        )
        spec = {"requestorID": self.opts.requestor_id}
        if drain_spec is not None:
            spec["drainSpec"] = drain_spec
        if pod_completion is not None:
            spec["waitForPodCompletion"] = pod_completion
        self._default_nm_spec = spec

    def new_node_maintenance(self, node_name: str) -> K8sObject:
        spec = dict(self._default_nm_spec)
        spec["nodeName"] = node_name
        spec["cordon"] = True
        return {
            "apiVersion": NODE_MAINTENANCE_API_VERSION,
            "kind": NODE_MAINTENANCE_KIND,
            "metadata": {
                "name": self.get_node_maintenance_name(node_name),
                "namespace": self.opts.namespace,
            },
            "spec": spec,
        }

    def get_node_maintenance_obj(self, node_name: str) -> Optional[K8sObject]:
        """(upgrade_requestor.go:203-218): None when absent."""
        try:
            return self.common.client.get(
                NODE_MAINTENANCE_API_VERSION, NODE_MAINTENANCE_KIND,
                self.get_node_maintenance_name(node_name), self.opts.namespace,
            )
        except NotFoundError:
            return None

    def _create_node_maintenance(self, node_state: NodeUpgradeState) -> None:
        nm = self.new_node_maintenance(meta.name(node_state.node))
        node_state.node_maintenance = nm
        try:
            self.common.client.create(nm)
        except AlreadyExistsError:
            logger.warning("nodeMaintenance %s already exists", meta.name(nm))

    def _create_or_update_node_maintenance(self, node_state: NodeUpgradeState) -> None:
        """Shared-requestor create/join (upgrade_requestor.go:320-368)."""
        nm = node_state.node_maintenance
        if nm is not None and self.opts.name_prefix == DEFAULT_NODE_MAINTENANCE_NAME_PREFIX:
            spec = nm.get("spec", {})
            if spec.get("requestorID") == self.opts.requestor_id:
                return  # we own it already
            additional = spec.get("additionalRequestors") or []
            if self.opts.requestor_id in additional:
                return
            # join via optimistic-lock merge patch so concurrent operators
            # can't clobber each other's additionalRequestors entries
            self.common.client.patch(
                NODE_MAINTENANCE_API_VERSION, NODE_MAINTENANCE_KIND, meta.name(nm),
                {
                    "metadata": {"resourceVersion": meta.resource_version(nm)},
                    "spec": {"additionalRequestors": additional + [self.opts.requestor_id]},
                },
                meta.namespace(nm),
            )
        else:
            self._create_node_maintenance(node_state)

    def _delete_or_update_node_maintenance(self, node_state: NodeUpgradeState) -> None:
        """Symmetric removal on uncordon (upgrade_requestor.go:370-410)."""
        nm = node_state.node_maintenance
        if nm is None:
            return
        spec = nm.get("spec", {})
        if spec.get("requestorID") == self.opts.requestor_id:
            # we own the object: request deletion (the maintenance operator's
            # finalizer performs the actual removal after uncordon)
            if "deletionTimestamp" not in nm.get("metadata", {}):
                try:
                    self.common.client.delete(
                        NODE_MAINTENANCE_API_VERSION, NODE_MAINTENANCE_KIND,
                        meta.name(nm), meta.namespace(nm),
                    )
                except NotFoundError:
                    pass
            return
        additional = spec.get("additionalRequestors") or []
        if self.opts.requestor_id in additional:
            remaining = [r for r in additional if r != self.opts.requestor_id]
            self.common.client.patch(
                NODE_MAINTENANCE_API_VERSION, NODE_MAINTENANCE_KIND, meta.name(nm),
                {
                    "metadata": {"resourceVersion": meta.resource_version(nm)},
                    "spec": {"additionalRequestors": remaining or None},
                },
                meta.namespace(nm),
            )

    # -- phase processors -----------------------------------------------------

    def process_upgrade_required_nodes(
        self,
        state: ClusterUpgradeState,
        upgrade_policy: DriverUpgradePolicySpec,
    ) -> None:
        """(upgrade_requestor.go:277-319).  NOTE: no maxParallelUpgrades
        gating here — throttling is delegated to the maintenance operator."""
        common = self.common
        self.set_default_node_maintenance(upgrade_policy)
        for node_state in state.nodes_in(consts.UPGRADE_STATE_UPGRADE_REQUIRED):
            node = node_state.node
            if common.is_upgrade_requested(node):
                common.node_state_provider.change_node_upgrade_annotation(
                    node, util.get_upgrade_requested_annotation_key(), consts.NULL_STRING
                )
            if common.skip_node_upgrade(node):
                continue
            self._create_or_update_node_maintenance(node_state)
            common.node_state_provider.change_node_upgrade_annotation(
                node, util.get_upgrade_requestor_mode_annotation_key(), consts.TRUE_STRING
            )
            common.node_state_provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED
            )

    def process_node_maintenance_required_nodes(self, state: ClusterUpgradeState) -> None:
        """Watch for maintenance completion (upgrade_requestor.go:416-452):
        Ready condition -> pod-restart-required; missing object -> recover to
        upgrade-required."""
        common = self.common
        for node_state in state.nodes_in(consts.UPGRADE_STATE_NODE_MAINTENANCE_REQUIRED):
            nm = node_state.node_maintenance
            if nm is None:
                # object vanished (e.g. deleted externally): restart the flow
                common.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_UPGRADE_REQUIRED
                )
                continue
            cond = find_status_condition(nm, CONDITION_REASON_READY)
            if cond is not None and cond.get("reason") == CONDITION_REASON_READY:
                common.node_state_provider.change_node_upgrade_state(
                    node_state.node, consts.UPGRADE_STATE_POD_RESTART_REQUIRED
                )

    def process_uncordon_required_nodes(self, state: ClusterUpgradeState) -> None:
        """(upgrade_requestor.go:454-488): requestor-mode nodes complete here;
        the maintenance operator performs the actual uncordon when its object
        is deleted/released."""
        common = self.common
        for node_state in state.nodes_in(consts.UPGRADE_STATE_UNCORDON_REQUIRED):
            node = node_state.node
            if not is_node_in_requestor_mode(node):
                continue
            common.node_state_provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_DONE
            )
            common.node_state_provider.change_node_upgrade_annotation(
                node, util.get_upgrade_requestor_mode_annotation_key(), consts.NULL_STRING
            )
            self._delete_or_update_node_maintenance(node_state)


# -- watch predicates for consumer operators ---------------------------------

def requestor_id_predicate(requestor_id: str):
    """Event filter: NodeMaintenance objects owned by or shared with this
    requestor (upgrade_requestor.go:93-103)."""

    def pred(obj: K8sObject) -> bool:
        spec = obj.get("spec", {})
        return (
            spec.get("requestorID") == requestor_id
            or requestor_id in (spec.get("additionalRequestors") or [])
        )

    return pred


def condition_changed_predicate(old: Optional[K8sObject], new: Optional[K8sObject]) -> bool:
    """Update-event filter (upgrade_requestor.go:115-159): fire when the
    sorted condition list changed, or on the deletion-flow update where the
    object's finalizers were removed while deletion is pending."""
    if old is None or new is None:
        return True

    def conds(o: K8sObject):
        return sorted(
            (
                {k: c.get(k) for k in ("type", "status", "reason", "message")}
                for c in o.get("status", {}).get("conditions", []) or []
            ),
            key=lambda c: str(c.get("type")),
        )

    if conds(old) != conds(new):
        return True
    old_fin = old.get("metadata", {}).get("finalizers") or []
    new_fin = new.get("metadata", {}).get("finalizers") or []
    if (
        "deletionTimestamp" in new.get("metadata", {})
        and old_fin
        and not new_fin
    ):
        return True
    return False
