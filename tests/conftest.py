import sys
from pathlib import Path

import pytest

# Make the in-tree package importable regardless of cwd.
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X (run via gpurun)"
    )


@pytest.fixture
def fake_cluster():
    from k8s_cc_manager_amd.k8s.fakecluster import FakeCluster

    cluster = FakeCluster()
    url = cluster.start()
    yield cluster, url
    cluster.stop()
