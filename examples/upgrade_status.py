#!/usr/bin/env python3
"""kubectl-style inspection of driver-upgrade state across the cluster.

    python examples/upgrade_status.py [--driver-name amdgpu] [--watch]

Prints one row per node: upgrade state, schedulability, safe-load/requestor
annotations, and the aggregate counts consumers export as metrics.
"""

from __future__ import annotations

import argparse
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from k8s_operator_libs_amd.upgrade import util


def render(client, driver_name: str) -> str:
    util.set_driver_name(driver_name)
    state_key = util.get_upgrade_state_label_key()
    safe_key = util.get_upgrade_wait_for_safe_driver_load_annotation_key()
    req_key = util.get_upgrade_requestor_mode_annotation_key()
    rows = []
    counts: dict = {}
    for node in client.list_nodes():
        md = node["metadata"]
        state = md.get("labels", {}).get(state_key, "")
        counts[state or "unknown"] = counts.get(state or "unknown", 0) + 1
        flags = []
        if node.get("spec", {}).get("unschedulable"):
            flags.append("cordoned")
        if safe_key in (md.get("annotations") or {}):
            flags.append("safe-load-wait")
        if (md.get("annotations") or {}).get(req_key) == "true":
            flags.append("requestor")
        rows.append((md["name"], state or "<unknown>", ",".join(flags) or "-"))
    out = [f"{'NODE':30} {'UPGRADE-STATE':26} FLAGS"]
    for name, state, flags in sorted(rows):
        out.append(f"{name:30} {state:26} {flags}")
    out.append("")
    out.append("totals: " + "  ".join(f"{k}={v}" for k, v in sorted(counts.items())))
    return "\n".join(out)


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--driver-name", default="amdgpu")
    parser.add_argument("--watch", action="store_true", help="refresh every 2s")
    parser.add_argument("--fake", action="store_true", help="in-memory cluster (demo)")
    args = parser.parse_args(argv)

    if args.fake:
        from k8s_operator_libs_amd.core.client import FakeClient

        client = FakeClient()
    else:
        from k8s_operator_libs_amd.core.restclient import RestClient

        client = RestClient.from_environment()

    while True:
        print(render(client, args.driver_name))
        if not args.watch:
            return 0
        time.sleep(2)
        print("\033[2J\033[H", end="")


if __name__ == "__main__":
    sys.exit(main())
