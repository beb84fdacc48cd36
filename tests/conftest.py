import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real AMD GPU (MI355X) and ROCm runtime"
    )


@pytest.fixture
def cluster():
    from k8s_operator_libs_amd.core import FakeCluster

    return FakeCluster()


@pytest.fixture
def client(cluster):
    from k8s_operator_libs_amd.core import FakeClient

    return FakeClient(cluster)


@pytest.fixture(autouse=True)
def _reset_driver_name():
    from k8s_operator_libs_amd.upgrade import util

    util.set_driver_name("amdgpu")
    yield
    util.set_driver_name("amdgpu")


@pytest.fixture(autouse=True)
def _reset_metrics():
    from k8s_operator_libs_amd import metrics

    metrics.reset_default_registry()
    yield
    metrics.reset_default_registry()
