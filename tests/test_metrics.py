"""Prometheus metrics: transition observations are recorded."""

import pytest

prometheus_client = pytest.importorskip("prometheus_client")

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.utils.metrics import METRICS


def _sample(metric_name, labels):
    for family in prometheus_client.REGISTRY.collect():
        for sample in family.samples:
            if sample.name == metric_name and all(
                sample.labels.get(k) == v for k, v in labels.items()
            ):
                return sample.value
    return None


def test_transition_metrics_recorded(fake_cluster):
    assert METRICS.enabled
    cluster, url = fake_cluster
    cluster.add_node("node0")
    mgr = CCManager(
        node_name="node0",
        default_mode="on",
        host_cc=True,
        k8s=K8sClient(url),
        backend=MockBackend(num_gpus=2),
        engine=TransitionEngine(),
        config=ManagerConfig(evict_components=False, cordon_node=False),
    )
    before = _sample("cc_transitions_total", {"mode": "on", "outcome": "ok"}) or 0
    assert mgr.apply_mode("on")
    after = _sample("cc_transitions_total", {"mode": "on", "outcome": "ok"})
    assert after == before + 1
    # histogram recorded at least one observation for the whole transition
    count = _sample("cc_transition_seconds_count", {"mode": "on"})
    assert count and count >= 1


def test_attest_failure_metric_increments():
    """A failing attestor must bump cc_attest_failures_total."""
    import pytest

    from k8s_cc_manager_amd.core.transition import TransitionEngine
    from k8s_cc_manager_amd.device.mock import MockBackend
    from k8s_cc_manager_amd.utils.metrics import METRICS

    if not METRICS.enabled:
        pytest.skip("prometheus_client unavailable")

    def bad_attestor(dev):
        raise RuntimeError("injected attestation failure")

    backend = MockBackend(num_gpus=1)
    engine = TransitionEngine(attestor=bad_attestor)
    before = METRICS.attest_failures._value.get()
    report = engine.apply_cc_mode(backend.get_gpus(), backend.get_gpus(), "on")
    assert not report.ok
    assert "attestation" in report.error
    assert METRICS.attest_failures._value.get() == before + 1
