import sys
from pathlib import Path

import pytest

# Make the in-tree package importable regardless of cwd.
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X (run via gpurun)"
    )


@pytest.fixture(autouse=True)
def _isolated_event_log(tmp_path, monkeypatch):
    """Keep transition event logs out of /var/lib during tests."""
    monkeypatch.setenv("CC_EVENT_LOG", str(tmp_path / "transitions.jsonl"))


@pytest.fixture
def fake_cluster():
    from k8s_cc_manager_amd.k8s.fakecluster import FakeCluster

    cluster = FakeCluster()
    url = cluster.start()
    yield cluster, url
    cluster.stop()
