"""Kubernetes Event emission around transitions."""

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import FaultPlan, MockBackend
from k8s_cc_manager_amd.k8s.client import K8sClient

NODE = "node0"


def _mgr(cluster, url, backend):
    return CCManager(
        node_name=NODE,
        default_mode="on",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend,
        engine=TransitionEngine(),
        config=ManagerConfig(evict_components=False, cordon_node=False),
    )


def test_success_events(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node(NODE)
    mgr = _mgr(cluster, url, MockBackend(num_gpus=2))
    assert mgr.apply_mode("on")
    mgr.flush_events()
    reasons = [e["reason"] for e in cluster.k8s_events]
    assert reasons == ["CCTransitionStarted", "CCTransitionSucceeded"]
    done = cluster.k8s_events[-1]
    assert done["type"] == "Normal"
    assert done["involvedObject"] == {
        "kind": "Node", "name": NODE, "apiVersion": "v1"
    }
    assert "2 device(s) reset" in done["message"]


def test_failure_event_is_warning(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node(NODE)
    be = MockBackend(num_gpus=2, faults=FaultPlan(fail_reset=["0000:10:00.0"]))
    mgr = _mgr(cluster, url, be)
    assert not mgr.apply_mode("on")
    mgr.flush_events()
    last = cluster.k8s_events[-1]
    assert last["reason"] == "CCTransitionFailed"
    assert last["type"] == "Warning"
    assert "0000:10:00.0" in last["message"]


def test_event_worker_survives_post_failures(fake_cluster):
    """A failing Events endpoint must not wedge the worker or the
    queue — posts are best-effort observability."""
    cluster, url = fake_cluster
    cluster.add_node(NODE)
    mgr = _mgr(cluster, url, MockBackend(num_gpus=1))

    calls = {"n": 0}

    def failing_create_event(*a, **kw):
        calls["n"] += 1
        raise RuntimeError("injected event failure")

    mgr.k8s.create_event = failing_create_event
    assert mgr.apply_mode("on")  # transition unaffected
    mgr.flush_events(timeout=5.0)
    assert calls["n"] == 2  # both events were attempted
    assert mgr._event_q.unfinished_tasks == 0  # queue fully drained
    # worker is still alive for subsequent events
    assert mgr.apply_mode("off")
    mgr.flush_events(timeout=5.0)
    assert calls["n"] == 4


def test_idempotent_apply_emits_no_transition_events(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node(NODE)
    mgr = _mgr(cluster, url, MockBackend(num_gpus=1, initial_cc_mode="on"))
    assert mgr.apply_mode("on")
    mgr.flush_events()
    assert cluster.k8s_events == []
