import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests requiring a real MI355X GPU (run via gpurun)")


@pytest.fixture()
def control_plane():
    from kuberay_amd.testing import ControlPlane
    cp = ControlPlane(kubelet_delay=0.01, job_runtime=0.2, poll_seconds=0.05)
    cp.start()
    yield cp
    cp.stop()


@pytest.fixture()
def client(control_plane):
    return control_plane.client
