"""In-place upgrade mode: the library performs node operations itself.

Capability parity with the reference's ``pkg/upgrade/upgrade_inplace.go``:
``process_upgrade_required_nodes`` moves up to ``upgrades_available`` nodes
into cordon-required (already-cordoned nodes always progress, bypassing the
limit — upgrade_inplace.go:87-97), and ``process_uncordon_required_nodes``
uncordons and completes nodes not owned by requestor mode.
"""

from __future__ import annotations

import logging
from ..api.upgrade.v1alpha1 import DriverUpgradePolicySpec, IntOrString
from ..core import meta
from . import consts, util
from .common_manager import (
    ClusterUpgradeState,
    CommonUpgradeManager,
    is_node_in_requestor_mode,
    is_node_unschedulable,
)

logger = logging.getLogger(__name__)


class InplaceNodeStateManager:
    def __init__(self, common: CommonUpgradeManager) -> None:
        self.common = common

    def process_upgrade_required_nodes(
        self,
        state: ClusterUpgradeState,
        upgrade_policy: DriverUpgradePolicySpec,
    ) -> None:
        """(upgrade_inplace.go:44-109)"""
        common = self.common
        total_nodes = common.get_total_managed_nodes(state)
        in_progress = common.get_upgrades_in_progress(state)
        current_unavailable = common.get_current_unavailable_nodes(state)
        max_unavailable = total_nodes
        if upgrade_policy.max_unavailable is not None:
            max_unavailable = IntOrString.scaled_value(
                upgrade_policy.max_unavailable, total_nodes, round_up=True
            )
        upgrades_available = common.get_upgrades_available(
            state, upgrade_policy.max_parallel_upgrades, max_unavailable
        )
        logger.info(
            "upgrades: in_progress=%d max_parallel=%d slots=%d unavailable=%d "
            "total=%d max_unavailable=%d",
            in_progress, upgrade_policy.max_parallel_upgrades, upgrades_available,
            current_unavailable, total_nodes, max_unavailable,
        )
        for node_state in state.nodes_in(consts.UPGRADE_STATE_UPGRADE_REQUIRED):
            node = node_state.node
            if common.is_upgrade_requested(node):
                # consume the explicit upgrade-requested annotation
                common.node_state_provider.change_node_upgrade_annotation(
                    node, util.get_upgrade_requested_annotation_key(), consts.NULL_STRING
                )
            if common.skip_node_upgrade(node):
                logger.info("node %s marked to skip upgrades", meta.name(node))
                continue
            if upgrades_available <= 0:
                # manually cordoned nodes bypass the limit: they are already
                # unavailable, so upgrading them costs nothing extra
                # (upgrade_inplace.go:87-97)
                if not is_node_unschedulable(node):
                    continue
                logger.debug("node %s already cordoned, progressing", meta.name(node))
            common.node_state_provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_CORDON_REQUIRED
            )
            upgrades_available -= 1

    def process_node_maintenance_required_nodes(self, state: ClusterUpgradeState) -> None:
        """No-op in in-place mode (upgrade_inplace.go:115-120)."""

    def process_uncordon_required_nodes(self, state: ClusterUpgradeState) -> None:
        """(upgrade_inplace.go:124-147)"""
        common = self.common

        def one(node_state):
            node = node_state.node
            if is_node_in_requestor_mode(node):
                # requestor flow owns this node's uncordon
                return
            common.cordon_manager.uncordon(node)
            common.node_state_provider.change_node_upgrade_state(
                node, consts.UPGRADE_STATE_DONE
            )

        common.for_each_node(
            state.nodes_in(consts.UPGRADE_STATE_UNCORDON_REQUIRED), one
        )
