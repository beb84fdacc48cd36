#!/usr/bin/env python3
"""Helm-hook CLI: apply or delete CRDs from YAML paths.

Capability parity with the reference's ``examples/apply-crds/main.go:34-61``.
Run as a Helm pre-install/pre-upgrade (apply) or pre-delete (delete) hook::

    python examples/apply_crds.py --crds-path deploy/crds --operation apply
    python examples/apply_crds.py --crds-path a.yaml --crds-path dir/ --operation delete

Against a real cluster this uses in-cluster configuration (service-account
token) or ``$KUBECONFIG``; ``--fake`` runs against an in-memory apiserver
(useful for dry-runs and CI).
"""

import argparse
import logging
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from k8s_operator_libs_amd.core.client import FakeClient
from k8s_operator_libs_amd.crdutil import (
    CRD_OPERATION_APPLY,
    CRD_OPERATION_DELETE,
    process_crds,
)


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument(
        "--crds-path", action="append", required=True, dest="crds_paths",
        help="file or directory containing CRD YAMLs (repeatable)",
    )
    parser.add_argument(
        "--operation", choices=[CRD_OPERATION_APPLY, CRD_OPERATION_DELETE],
        required=True,
    )
    parser.add_argument("--fake", action="store_true", help="use an in-memory apiserver")
    parser.add_argument("-v", "--verbose", action="store_true")
    args = parser.parse_args(argv)

    logging.basicConfig(level=logging.DEBUG if args.verbose else logging.INFO,
                        format="%(asctime)s %(levelname)s %(name)s %(message)s")

    if args.fake:
        client = FakeClient()
    else:
        from k8s_operator_libs_amd.core.restclient import RestClient

        client = RestClient.from_environment()

    n = process_crds(client, args.crds_paths, args.operation)
    logging.info("%s: processed %d CRDs", args.operation, n)
    return 0


if __name__ == "__main__":
    sys.exit(main())
