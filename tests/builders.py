"""Shim: fixtures now live in the public testing module."""

from k8s_operator_libs_amd.testing import *  # noqa: F401,F403
from k8s_operator_libs_amd.testing import (  # noqa: F401
    DRIVER_LABELS, DRIVER_NS, DaemonSetBuilder, NodeBuilder,
    NodeMaintenanceBuilder, PodBuilder, driver_pod_for,
    make_controller_revision,
)
