"""Seeded chaos: random interleavings of desired-mode flips, unrelated
label churn and event-log compactions against a live manager; the
system must always converge to the last desired mode with consistent
state labels and restored components."""

import random
import threading
import time

import pytest

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.k8s.eviction import COMPONENT_LABELS
from k8s_cc_manager_amd.k8s.fakecluster import FakeCluster
from k8s_cc_manager_amd.labels import (
    CC_MODE_LABEL,
    CC_READY_LABEL,
    CC_STATE_LABEL,
    ready_value_for_state,
)

NODE = "chaos0"
MODES = ["on", "off", "devtools", "ppcie"]


@pytest.mark.parametrize("seed", [7, 1234, 987654, 5150])
def test_chaos_converges(seed):
    rng = random.Random(seed)
    cluster = FakeCluster(event_log_max=300, operator_tick=0.01)
    url = cluster.start()
    cluster.add_node(NODE, labels={n: "true" for n in COMPONENT_LABELS})
    backend = MockBackend(num_gpus=rng.choice([1, 2, 8]))
    mgr = CCManager(
        node_name=NODE,
        default_mode="off",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend,
        engine=TransitionEngine(),
        config=ManagerConfig(
            evict_components=True,
            cordon_node=True,
            eviction_timeout=5.0,
            eviction_poll_interval=0.02,
            watch_timeout_seconds=2,
            reconnect_backoff=0.05,
            readiness_file=f"/tmp/.chaos-{seed}",
        ),
    )
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()

    last_mode = "off"
    for step in range(40):
        op = rng.random()
        if op < 0.45:
            last_mode = rng.choice(MODES)
            cluster.set_node_label(NODE, CC_MODE_LABEL, last_mode)
        elif op < 0.8:
            cluster.set_node_label(NODE, f"churn-{rng.randint(0, 5)}", str(step))
        else:
            cluster.compact()
        time.sleep(rng.uniform(0.0, 0.08))

    # settle: re-assert the final desired mode then wait for convergence
    cluster.set_node_label(NODE, CC_MODE_LABEL, last_mode)
    deadline = time.monotonic() + 30
    while time.monotonic() < deadline:
        labels = cluster.node_labels(NODE)
        if labels.get(CC_STATE_LABEL) == last_mode and not cluster.node_unschedulable(NODE):
            break
        time.sleep(0.05)

    labels = cluster.node_labels(NODE)
    assert labels.get(CC_STATE_LABEL) == last_mode, (seed, labels)
    assert labels.get(CC_READY_LABEL) == ready_value_for_state(last_mode)
    for name in COMPONENT_LABELS:
        assert labels[name] == "true", (name, labels[name])
    assert not cluster.node_unschedulable(NODE)
    # device state consistent with the label
    if last_mode == "ppcie":
        assert all(d.query_fabric_mode() == "on" for d in backend.get_gpus())
    else:
        assert all(d.query_cc_mode() == last_mode for d in backend.get_gpus())

    mgr.stop_event.set()
    t.join(timeout=10)
    cluster.stop()
