#!/usr/bin/env python3
"""Example consumer operator: rolling amdgpu driver upgrades on MI355X nodes.

Shows how a GPU operator consumes this library (the reference's consumers are
NVIDIA's GPU Operator / Network Operator; this is the AMD equivalent wired
end-to-end):

- builds a ClusterUpgradeStateManager with pod deletion keyed on
  ``amd.com/gpu`` resources and validation gated on amd-gpu-validator pods,
- runs the reconcile loop (build_state + apply_state) with a requeue
  interval,
- serves Prometheus metrics (reconcile histograms, per-state node gauges)
  on ``--metrics-port``,
- reads requestor-mode configuration from MAINTENANCE_OPERATOR_* env vars.

Run against a real cluster (in-cluster / $KUBECONFIG / $KUBERNETES_MASTER) or
``--demo`` to watch a full rolling upgrade against an in-process mini
apiserver with 8 synthetic MI355X nodes.
"""

from __future__ import annotations

import argparse
import logging
import sys
import threading
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import DriverUpgradePolicySpec
from k8s_operator_libs_amd.upgrade import consts, util
from k8s_operator_libs_amd.upgrade.drain import gpu_pod_deletion_filter
from k8s_operator_libs_amd.upgrade.requestor import get_requestor_opts_from_envs
from k8s_operator_libs_amd.upgrade.state_manager import (
    ClusterUpgradeStateManager,
    StateOptions,
)

log = logging.getLogger("amdgpu-upgrade-operator")


def serve_metrics(manager, port: int):
    from http.server import BaseHTTPRequestHandler, HTTPServer

    class Handler(BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path != "/metrics":
                self.send_response(404)
                self.end_headers()
                return
            body = manager.metrics.render_text().encode()
            self.send_response(200)
            self.send_header("Content-Type", "text/plain; version=0.0.4")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *args):
            pass

    HTTPServer.allow_reuse_address = True  # survive TIME_WAIT on restart
    server = HTTPServer(("127.0.0.1", port), Handler)
    threading.Thread(target=server.serve_forever, daemon=True).start()
    return server


def run_demo(args):
    """Self-contained demo: 8 synthetic MI355X nodes on the mini-apiserver."""
    import bench as bench_mod
    from k8s_operator_libs_amd.core.apiserver import start_apiserver
    from k8s_operator_libs_amd.core.restclient import RestClient

    handle = start_apiserver()
    log.info("mini-apiserver at %s", handle.url)
    client = RestClient(handle.url)

    ds, _cleanup = bench_mod._make_cluster(handle.cluster, args.demo_nodes,
                                           "oldrev", "newrev")
    bench_mod._DsController(handle.cluster, ds, "newrev",
                            bench_mod._driver_labels("inplace"))

    options = None
    if args.demo_requestor:
        from k8s_operator_libs_amd.testing import SimMaintenanceOperator
        from k8s_operator_libs_amd.upgrade.requestor import RequestorOptions

        # the simulated maintenance operator must not evict the validator
        # pods the upgrade gates on (the real one applies podEvictionFilters)
        SimMaintenanceOperator(
            handle.cluster,
            evict_filter=lambda pod: pod["metadata"].get("labels", {})
            .get("app") != "amd-gpu-validator",
        )
        options = StateOptions(requestor=RequestorOptions(
            use_maintenance_operator=True,
            requestor_id="amd.gpu.operator",
            namespace="default",
        ))
        log.info("demo: requestor mode (simulated maintenance operator)")
    manager = (
        ClusterUpgradeStateManager(client, options=options)
        .with_pod_deletion_enabled(gpu_pod_deletion_filter)
        .with_validation_enabled("app=amd-gpu-validator")
    )
    serve_metrics(manager, args.metrics_port)
    policy = DriverUpgradePolicySpec.model_validate({
        "autoUpgrade": True,
        "maxParallelUpgrades": args.max_parallel,
        "maxUnavailable": "50%",
        "podDeletion": {"deleteEmptyDir": True},
        "drain": {"enable": True},
    })

    # demo validator: mark validation pods ready once their node reaches
    # validation-required (a real cluster runs the gpu_health check pod)
    def validator_sim():
        state_key = util.get_upgrade_state_label_key()
        while True:
            for node in handle.cluster.list("v1", "Node"):
                if node["metadata"]["labels"].get(state_key) == \
                        consts.UPGRADE_STATE_VALIDATION_REQUIRED:
                    name = node["metadata"]["name"]
                    try:
                        handle.cluster.patch(
                            "v1", "Pod", f"validator-{name}",
                            {"status": {"containerStatuses": [
                                {"name": "v", "ready": True, "restartCount": 0}]}},
                            bench_mod.DRIVER_NS)
                    except Exception:
                        pass
            time.sleep(0.05)

    threading.Thread(target=validator_sim, daemon=True).start()

    from k8s_operator_libs_amd.upgrade.controller import UpgradeController

    t0 = time.time()
    controller = UpgradeController(
        manager, bench_mod.DRIVER_NS, bench_mod.DRIVER_LABELS, policy,
        resync_seconds=max(args.interval, 0.05),
    )
    ok = controller.run(until_all_done=True, max_reconciles=200)
    controller.stop()
    log.info("demo %s in %.2fs; metrics: curl 127.0.0.1:%d/metrics",
             "completed" if ok else "DID NOT complete", time.time() - t0,
             args.metrics_port)
    p50 = manager.metrics.reconcile_duration.quantile(0.5) * 1000
    log.info("reconcile p50: %.2f ms over %d ticks",
             p50, manager.metrics.reconcile_duration.count)
    handle.stop()
    return 0 if ok else 1


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--namespace", default="amd-gpu-operator")
    parser.add_argument("--driver-label", default="app=amdgpu-driver-daemonset")
    parser.add_argument("--driver-name", default="amdgpu")
    parser.add_argument("--max-parallel", type=int, default=2)
    parser.add_argument("--interval", type=float, default=0.2)
    parser.add_argument("--metrics-port", type=int, default=8080)
    parser.add_argument("--demo", action="store_true")
    parser.add_argument("--demo-nodes", type=int, default=8)
    parser.add_argument("--no-leader-election", action="store_true")
    parser.add_argument("--demo-requestor", action="store_true",
                        help="demo: delegate node quiescing to a simulated maintenance operator")
    args = parser.parse_args(argv)

    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    util.set_driver_name(args.driver_name)

    if args.demo:
        return run_demo(args)

    from k8s_operator_libs_amd.core.restclient import RestClient

    client = RestClient.from_environment()
    key, _, value = args.driver_label.partition("=")
    driver_labels = {key: value}
    requestor = get_requestor_opts_from_envs()
    manager = ClusterUpgradeStateManager(
        client, options=StateOptions(requestor=requestor)
    ).with_pod_deletion_enabled(gpu_pod_deletion_filter).with_validation_enabled(
        "app=amd-gpu-validator"
    )
    serve_metrics(manager, args.metrics_port)
    policy = DriverUpgradePolicySpec.model_validate({
        "autoUpgrade": True,
        "maxParallelUpgrades": args.max_parallel,
        "podDeletion": {"deleteEmptyDir": True},
        "drain": {"enable": True},
    })

    from k8s_operator_libs_amd.core.leaderelection import LeaderElector
    from k8s_operator_libs_amd.upgrade.controller import UpgradeController

    controller = UpgradeController(
        manager, args.namespace, driver_labels, policy,
        resync_seconds=max(args.interval, 1.0),
    )

    # graceful termination: SIGTERM/SIGINT stop the reconcile loop, which
    # releases the Lease on the way out so another replica takes over
    # immediately instead of waiting out the lease duration
    import signal

    def _shutdown(signum, frame):
        log.info("received signal %d: shutting down", signum)
        controller.stop()

    try:
        signal.signal(signal.SIGTERM, _shutdown)
        signal.signal(signal.SIGINT, _shutdown)
    except ValueError:
        pass  # not the main thread (tests drive main() from a worker)

    if args.no_leader_election:
        controller.run()
    else:
        elector = LeaderElector(client, "amd-gpu-operator-upgrade",
                                namespace=args.namespace)
        elector.run(on_started_leading=controller.run,
                    on_stopped_leading=controller.stop)
    return 0


if __name__ == "__main__":
    sys.exit(main())
