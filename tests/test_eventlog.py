"""Transition event log: records written by the manager, readable back."""

import time

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend, FaultPlan
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.utils import eventlog


def _manager(cluster, url, backend):
    return CCManager(
        node_name="node0",
        default_mode="on",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend,
        engine=TransitionEngine(),
        config=ManagerConfig(evict_components=False, cordon_node=False),
    )


def test_transition_recorded(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node("node0")
    mgr = _manager(cluster, url, MockBackend(num_gpus=2))
    assert mgr.apply_mode("on")
    events = eventlog.read_transitions()
    assert len(events) == 1
    e = events[0]
    assert e["node"] == "node0"
    assert e["mode"] == "on"
    assert e["ok"] is True
    assert set(e["phases"]) >= {"stage", "reset", "verify"}
    assert len(e["devices_changed"]) == 2
    assert e["ts"] <= time.time()


def test_failed_transition_recorded_with_error(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node("node0")
    be = MockBackend(num_gpus=2, faults=FaultPlan(fail_reset=["0000:10:00.0"]))
    mgr = _manager(cluster, url, be)
    assert not mgr.apply_mode("on")
    events = eventlog.read_transitions()
    assert events[-1]["ok"] is False
    assert "0000:10:00.0" in events[-1]["error"]


def test_multiple_transitions_append(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node("node0")
    mgr = _manager(cluster, url, MockBackend(num_gpus=1))
    mgr.apply_mode("on")
    mgr.apply_mode("off")
    mgr.apply_mode("devtools")
    events = eventlog.read_transitions()
    assert [e["mode"] for e in events] == ["on", "off", "devtools"]
