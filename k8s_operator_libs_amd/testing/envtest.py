"""envtest: run the library against a REAL kube-apiserver + etcd.

The reference's entire test strategy is envtest — controller-runtime boots a
genuine ``kube-apiserver`` backed by ``etcd`` with no kubelet / scheduler /
controllers (reference ``pkg/upgrade/upgrade_suit_test.go:86-93``,
``Makefile:76-78``).  This module is the native equivalent: it locates the
binaries, boots the control plane, and hands back a base URL + bearer token
the :class:`~k8s_operator_libs_amd.core.restclient.RestClient` can talk to.

Binary discovery order (same conventions as controller-runtime's
setup-envtest):

1. ``$KUBEBUILDER_ASSETS`` — directory containing ``kube-apiserver`` and
   ``etcd`` (what ``setup-envtest use -p path`` prints),
2. ``$TEST_ASSET_KUBE_APISERVER`` / ``$TEST_ASSET_ETCD`` — explicit paths,
3. ``/usr/local/kubebuilder/bin`` — the historic default install dir,
4. ``$PATH``.

Auth uses a static ``--token-auth-file`` bearer token bound to
``system:masters`` (simplest client-agnostic superuser path; envtest proper
uses a signed client cert for the same group), and the service-account
signing keypair is generated with ``openssl`` — both requirements of modern
(v1.20+) apiservers.  Nothing here is AMD- or NVIDIA-specific: it is the
substrate the conformance suite (``tests/test_conformance.py``) runs on when
binaries are available, and skips cleanly when they are not (this build
container has no network and ships neither binary — see docs/testing.md).
"""

from __future__ import annotations

import os
import shutil
import socket
import subprocess
import tempfile
import time
import uuid
from typing import Optional

__all__ = ["find_assets", "EnvtestCluster", "start_envtest"]


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def find_assets() -> Optional[dict]:
    """Locate kube-apiserver and etcd binaries; None when unavailable."""
    candidates = []
    assets = os.environ.get("KUBEBUILDER_ASSETS")
    if assets:
        candidates.append(
            (os.path.join(assets, "kube-apiserver"), os.path.join(assets, "etcd"))
        )
    explicit = (
        os.environ.get("TEST_ASSET_KUBE_APISERVER"),
        os.environ.get("TEST_ASSET_ETCD"),
    )
    if all(explicit):
        candidates.append(explicit)
    candidates.append(
        ("/usr/local/kubebuilder/bin/kube-apiserver", "/usr/local/kubebuilder/bin/etcd")
    )
    path_api = shutil.which("kube-apiserver")
    path_etcd = shutil.which("etcd")
    if path_api and path_etcd:
        candidates.append((path_api, path_etcd))
    for apiserver, etcd in candidates:
        if apiserver and etcd and os.access(apiserver, os.X_OK) and os.access(etcd, os.X_OK):
            return {"kube_apiserver": apiserver, "etcd": etcd}
    return None


class EnvtestCluster:
    """A running kube-apiserver + etcd pair (no kubelet/scheduler), started
    from local binaries.  ``url`` + ``token`` configure a RestClient:

        cluster = start_envtest()
        client = RestClient(cluster.url, token=cluster.token, verify=False)
    """

    #: how long to wait for /readyz (binaries page in slowly on cold start)
    READY_TIMEOUT = 60.0

    def __init__(self, assets: dict) -> None:
        self._assets = assets
        self._dir = tempfile.mkdtemp(prefix="envtest-")
        self._procs: list = []
        self.token = uuid.uuid4().hex
        self.url = ""

    # -- lifecycle -----------------------------------------------------------

    def start(self) -> "EnvtestCluster":
        etcd_client_port = _free_port()
        etcd_peer_port = _free_port()
        api_port = _free_port()
        etcd_dir = os.path.join(self._dir, "etcd-data")
        log_dir = os.path.join(self._dir, "logs")
        os.makedirs(log_dir, exist_ok=True)

        etcd_log = open(os.path.join(log_dir, "etcd.log"), "w")
        self._procs.append((subprocess.Popen([
            self._assets["etcd"],
            "--data-dir", etcd_dir,
            "--listen-client-urls", f"http://127.0.0.1:{etcd_client_port}",
            "--advertise-client-urls", f"http://127.0.0.1:{etcd_client_port}",
            "--listen-peer-urls", f"http://127.0.0.1:{etcd_peer_port}",
            "--initial-advertise-peer-urls", f"http://127.0.0.1:{etcd_peer_port}",
            "--initial-cluster", f"default=http://127.0.0.1:{etcd_peer_port}",
            "--unsafe-no-fsync",  # test-only speedup, like envtest
        ], stdout=etcd_log, stderr=subprocess.STDOUT), etcd_log))

        sa_key = os.path.join(self._dir, "sa.key")
        sa_pub = os.path.join(self._dir, "sa.pub")
        subprocess.run(
            ["openssl", "genrsa", "-out", sa_key, "2048"],
            check=True, capture_output=True,
        )
        subprocess.run(
            ["openssl", "rsa", "-in", sa_key, "-pubout", "-out", sa_pub],
            check=True, capture_output=True,
        )
        token_file = os.path.join(self._dir, "tokens.csv")
        with open(token_file, "w") as fh:
            fh.write(f"{self.token},envtest-admin,envtest-admin,system:masters\n")

        api_log = open(os.path.join(log_dir, "kube-apiserver.log"), "w")
        self._procs.append((subprocess.Popen([
            self._assets["kube_apiserver"],
            "--etcd-servers", f"http://127.0.0.1:{etcd_client_port}",
            "--secure-port", str(api_port),
            "--bind-address", "127.0.0.1",
            "--cert-dir", os.path.join(self._dir, "certs"),  # self-signs
            "--token-auth-file", token_file,
            "--authorization-mode", "RBAC",
            "--service-cluster-ip-range", "10.0.0.0/24",
            "--allow-privileged=true",
            "--disable-admission-plugins", "ServiceAccount",
            "--service-account-issuer", "https://envtest.local",
            "--service-account-key-file", sa_pub,
            "--service-account-signing-key-file", sa_key,
            # same knobs controller-runtime sets for deterministic tests
            "--enable-aggregator-routing=false",
            "--max-mutating-requests-inflight", "400",
            "--max-requests-inflight", "800",
        ], stdout=api_log, stderr=subprocess.STDOUT), api_log))

        self.url = f"https://127.0.0.1:{api_port}"
        self._wait_ready()
        return self

    def _wait_ready(self) -> None:
        import httpx

        deadline = time.monotonic() + self.READY_TIMEOUT
        last_err: Optional[BaseException] = None
        while time.monotonic() < deadline:
            for proc, _log in self._procs:
                if proc.poll() is not None:
                    self.stop()
                    raise RuntimeError(
                        f"envtest process exited rc={proc.returncode}; "
                        f"logs under {self._dir}/logs"
                    )
            try:
                resp = httpx.get(
                    f"{self.url}/readyz", verify=False,
                    headers={"Authorization": f"Bearer {self.token}"},
                    timeout=2.0,
                )
                if resp.status_code == 200:
                    return
                last_err = RuntimeError(f"/readyz -> {resp.status_code}")
            except Exception as exc:  # conn refused while booting
                last_err = exc
            time.sleep(0.25)
        self.stop()
        raise RuntimeError(f"kube-apiserver not ready in {self.READY_TIMEOUT}s: {last_err}")

    def stop(self) -> None:
        for proc, log in self._procs:
            if proc.poll() is None:
                proc.terminate()
                try:
                    proc.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    proc.kill()
                    proc.wait(timeout=5)
            log.close()
        self._procs.clear()
        shutil.rmtree(self._dir, ignore_errors=True)

    def __enter__(self) -> "EnvtestCluster":
        return self

    def __exit__(self, *exc) -> None:
        self.stop()


def start_envtest() -> Optional[EnvtestCluster]:
    """Boot a real control plane if binaries are present; None otherwise."""
    assets = find_assets()
    if assets is None:
        return None
    return EnvtestCluster(assets).start()
