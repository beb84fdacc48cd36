"""Rolling driver-upgrade state machine for AMD GPU / NIC drivers.

Capability parity with the reference's ``pkg/upgrade``.  Public surface:

- :class:`~k8s_operator_libs_amd.upgrade.state_manager.ClusterUpgradeStateManager`
  — the facade a consumer operator's reconcile drives
  (``build_state`` + ``apply_state``);
- :class:`~k8s_operator_libs_amd.upgrade.common_manager.CommonUpgradeManager`
  — shared phase processors and rolling-window arithmetic;
- the L3 managers (cordon / drain / pod / validation / safe-driver-load) and
  the :class:`~k8s_operator_libs_amd.upgrade.node_state_provider.NodeUpgradeStateProvider`;
- in-place and requestor (maintenance-operator) mode implementations;
- state/key constants on the ``amd.com`` domain and the driver-name registry.
"""

from . import consts, util  # noqa: F401
from .common_manager import (  # noqa: F401
    ClusterUpgradeState,
    CommonUpgradeManager,
    NodeUpgradeState,
    is_node_in_requestor_mode,
    is_node_unschedulable,
    is_orphaned_pod,
)
from .controller import UpgradeController  # noqa: F401
from .cordon_manager import CordonManager  # noqa: F401
from .drain import gpu_pod_deletion_filter, pod_requests_resource  # noqa: F401
from .drain_manager import DrainConfiguration, DrainManager  # noqa: F401
from .inplace import InplaceNodeStateManager  # noqa: F401
from .node_state_provider import NodeUpgradeStateProvider  # noqa: F401
from .pod_manager import PodManager, PodManagerConfig  # noqa: F401
from .requestor import (  # noqa: F401
    RequestorNodeStateManager,
    RequestorOptions,
    get_requestor_opts_from_envs,
)
from .safe_driver_load_manager import SafeDriverLoadManager  # noqa: F401
from .state_manager import (  # noqa: F401
    BuildStateError,
    ClusterUpgradeStateManager,
    StateOptions,
)
from .validation_manager import ValidationManager  # noqa: F401
