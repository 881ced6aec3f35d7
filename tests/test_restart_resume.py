"""Crash-resume semantics: all durable state lives in labels + the
persisted mode store, so a restarted manager recovers without help
(reference model: DaemonSet restart + initial apply, SURVEY.md §5;
the paused-label algebra makes restore derivable from the label alone)."""

import time

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.k8s.eviction import (
    COMPONENT_APP_LABELS,
    COMPONENT_LABELS,
    PAUSED_VALUE,
)
from k8s_cc_manager_amd.labels import CC_STATE_LABEL

NODE = "node0"


def _mgr(cluster, url, backend=None):
    return CCManager(
        node_name=NODE,
        default_mode="on",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend or MockBackend(num_gpus=2),
        engine=TransitionEngine(),
        config=ManagerConfig(
            evict_components=True,
            cordon_node=True,
            eviction_timeout=5.0,
            eviction_poll_interval=0.02,
        ),
    )


def test_restart_after_crash_mid_eviction(fake_cluster):
    """The previous manager died AFTER pausing component labels (and
    cordoning) but BEFORE restoring. A fresh manager's reconcile must
    finish the transition and restore the components to 'true'."""
    cluster, url = fake_cluster
    # crashed state: everything paused, node cordoned
    labels = {name: PAUSED_VALUE for name in COMPONENT_LABELS}
    cluster.add_node(NODE, labels=labels)
    cluster._nodes[NODE]["spec"]["unschedulable"] = True

    mgr = _mgr(cluster, url)
    assert mgr.apply_mode("on")

    out = cluster.node_labels(NODE)
    assert out[CC_STATE_LABEL] == "on"
    for name in COMPONENT_LABELS:
        assert out[name] == "true"  # paused -> true (derivable restore)
    assert not cluster.node_unschedulable(NODE)
    # operator reschedules the component pods
    deadline = time.monotonic() + 3
    while time.monotonic() < deadline:
        if len(cluster.pods_on(NODE)) == len(COMPONENT_APP_LABELS):
            break
        time.sleep(0.02)
    assert len(cluster.pods_on(NODE)) == len(COMPONENT_APP_LABELS)


def test_restart_preserves_custom_component_values(fake_cluster):
    """Custom label values paused by a crashed manager come back as the
    original custom values, not 'true'."""
    cluster, url = fake_cluster
    labels = {name: "true" for name in COMPONENT_LABELS}
    labels[COMPONENT_LABELS[0]] = "custom-flavor_" + PAUSED_VALUE  # crashed mid-pause
    labels[COMPONENT_LABELS[1]] = "false"  # user-disabled stays disabled
    cluster.add_node(NODE, labels=labels)

    mgr = _mgr(cluster, url)
    assert mgr.apply_mode("on")
    out = cluster.node_labels(NODE)
    assert out[COMPONENT_LABELS[0]] == "custom-flavor"
    assert out[COMPONENT_LABELS[1]] == "false"


def test_second_manager_instance_is_idempotent(fake_cluster):
    """A restart after a completed transition re-applies nothing."""
    cluster, url = fake_cluster
    cluster.add_node(NODE, labels={n: "true" for n in COMPONENT_LABELS})
    backend = MockBackend(num_gpus=2)
    assert _mgr(cluster, url, backend).apply_mode("on")
    resets_before = backend.device(0)._reset_attempts

    # "restart": fresh manager over the same backend/cluster
    assert _mgr(cluster, url, backend).apply_mode("on")
    assert backend.device(0)._reset_attempts == resets_before  # no new reset
    assert cluster.node_labels(NODE)[CC_STATE_LABEL] == "on"
