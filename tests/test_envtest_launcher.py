"""envtest launcher plumbing, testable without real binaries.

The real kube-apiserver/etcd path runs via `make test-real-apiserver` where
assets exist; here we pin what CAN be verified offline: binary discovery
order and the launcher's process supervision (a crashing binary surfaces a
RuntimeError naming the log dir instead of hanging)."""

import os
import stat

import pytest

from k8s_operator_libs_amd.testing import envtest


def _fake_bin(path, script="#!/bin/sh\nsleep 60\n"):
    path.write_text(script)
    path.chmod(path.stat().st_mode | stat.S_IXUSR)
    return str(path)


@pytest.fixture
def assets_dir(tmp_path, monkeypatch):
    d = tmp_path / "assets"
    d.mkdir()
    _fake_bin(d / "kube-apiserver")
    _fake_bin(d / "etcd")
    for var in ("KUBEBUILDER_ASSETS", "TEST_ASSET_KUBE_APISERVER",
                "TEST_ASSET_ETCD"):
        monkeypatch.delenv(var, raising=False)
    return d


class TestFindAssets:
    def test_none_offline(self, monkeypatch):
        for var in ("KUBEBUILDER_ASSETS", "TEST_ASSET_KUBE_APISERVER",
                    "TEST_ASSET_ETCD"):
            monkeypatch.delenv(var, raising=False)
        # this image ships neither binary, nor /usr/local/kubebuilder
        assert envtest.find_assets() is None

    def test_kubebuilder_assets_dir(self, assets_dir, monkeypatch):
        monkeypatch.setenv("KUBEBUILDER_ASSETS", str(assets_dir))
        found = envtest.find_assets()
        assert found == {
            "kube_apiserver": str(assets_dir / "kube-apiserver"),
            "etcd": str(assets_dir / "etcd"),
        }

    def test_explicit_test_asset_vars(self, assets_dir, monkeypatch):
        monkeypatch.setenv("TEST_ASSET_KUBE_APISERVER",
                           str(assets_dir / "kube-apiserver"))
        monkeypatch.setenv("TEST_ASSET_ETCD", str(assets_dir / "etcd"))
        found = envtest.find_assets()
        assert found["etcd"] == str(assets_dir / "etcd")

    def test_non_executable_rejected(self, assets_dir, monkeypatch):
        (assets_dir / "kube-apiserver").chmod(0o644)
        monkeypatch.setenv("KUBEBUILDER_ASSETS", str(assets_dir))
        assert envtest.find_assets() is None

    def test_env_dir_beats_path(self, assets_dir, monkeypatch, tmp_path):
        other = tmp_path / "other"
        other.mkdir()
        _fake_bin(other / "kube-apiserver")
        _fake_bin(other / "etcd")
        monkeypatch.setenv("KUBEBUILDER_ASSETS", str(assets_dir))
        monkeypatch.setenv("PATH", f"{other}:{os.environ['PATH']}")
        assert envtest.find_assets()["etcd"] == str(assets_dir / "etcd")


class TestLauncherSupervision:
    def test_crashing_binary_surfaces_error(self, assets_dir, monkeypatch):
        # etcd exits immediately: start() must fail fast with the log dir
        # in the message, not hang for READY_TIMEOUT
        _fake_bin(assets_dir / "etcd", "#!/bin/sh\necho boom >&2\nexit 3\n")
        monkeypatch.setenv("KUBEBUILDER_ASSETS", str(assets_dir))
        cluster = envtest.EnvtestCluster(envtest.find_assets())
        cluster.READY_TIMEOUT = 10.0
        with pytest.raises(RuntimeError, match="exited rc=3|logs under"):
            cluster.start()

    def test_never_ready_times_out_and_cleans_up(self, assets_dir, monkeypatch):
        monkeypatch.setenv("KUBEBUILDER_ASSETS", str(assets_dir))
        cluster = envtest.EnvtestCluster(envtest.find_assets())
        cluster.READY_TIMEOUT = 2.0
        with pytest.raises(RuntimeError, match="not ready"):
            cluster.start()
        # both fake processes were terminated by stop()
        assert cluster._procs == []
