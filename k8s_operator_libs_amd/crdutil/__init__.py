"""CustomResourceDefinition lifecycle helpers.

Capability parity with the reference's ``pkg/crdutil/crdutil.go``: apply or
delete CRDs from YAML files and directories, with create-or-update retry on
conflict and a wait-until-served poll.  Intended for Helm pre-install /
pre-delete hook jobs, solving Helm's unmanaged-``crds/``-directory problem
(reference ``pkg/crdutil/README.md:8-15``): unlike Helm, this applies CRD
*updates* on upgrade and can garbage-collect them on chart deletion.
"""

from .crdutil import (  # noqa: F401
    CRD_OPERATION_APPLY,
    CRD_OPERATION_DELETE,
    CrdUtilError,
    apply_crds,
    delete_crds,
    parse_crds_from_paths,
    process_crds,
    wait_for_crds,
    walk_crd_paths,
)
