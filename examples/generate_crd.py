#!/usr/bin/env python3
"""Generate a consumer CRD embedding the driver-upgrade policy schema.

The controller-gen analogue for consumers: produces a complete
``AMDGPUDriver`` CustomResourceDefinition whose
``spec.driverUpgradePolicy`` is the structural OpenAPI schema generated from
:mod:`k8s_operator_libs_amd.api.upgrade.v1alpha1` — so a consuming operator's
CRD always matches the library's policy types.

    python examples/generate_crd.py > amdgpudrivers.yaml
"""

import sys

import yaml

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import openapi_v3_schema


def build_crd() -> dict:
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": "amdgpudrivers.driver.amd.com"},
        "spec": {
            "group": "driver.amd.com",
            "scope": "Cluster",
            "names": {
                "kind": "AMDGPUDriver",
                "listKind": "AMDGPUDriverList",
                "plural": "amdgpudrivers",
                "singular": "amdgpudriver",
                "shortNames": ["agd"],
            },
            "versions": [
                {
                    "name": "v1alpha1",
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                    "schema": {
                        "openAPIV3Schema": {
                            "type": "object",
                            "properties": {
                                "spec": {
                                    "type": "object",
                                    "properties": {
                                        "driverVersion": {"type": "string"},
                                        "image": {"type": "string"},
                                        "driverUpgradePolicy": openapi_v3_schema(),
                                    },
                                },
                                "status": {
                                    "type": "object",
                                    "x-kubernetes-preserve-unknown-fields": True,
                                },
                            },
                        }
                    },
                }
            ],
        },
    }


def main() -> int:
    print(yaml.safe_dump(build_crd(), sort_keys=False))
    return 0


if __name__ == "__main__":
    sys.exit(main())
