"""Shim: simulated actors now live in the public testing module."""

from k8s_operator_libs_amd.testing import (  # noqa: F401
    SimDaemonSetController, SimMaintenanceOperator,
)
