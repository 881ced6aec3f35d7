"""The reconcile hot path's atomic patches: cordon+pause in one
request, restore+uncordon+state in another (profiles/: the optimization
that cut 14.5 API requests/step to 6.9)."""

import collections

import pytest

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend
from k8s_cc_manager_amd.k8s import eviction
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.k8s.eviction import COMPONENT_LABELS, PAUSED_VALUE
from k8s_cc_manager_amd.labels import CC_READY_LABEL, CC_STATE_LABEL

NODE = "node0"


@pytest.fixture
def cluster_url(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node(NODE, labels={n: "true" for n in COMPONENT_LABELS})
    return cluster, url


def test_evict_components_cordons_atomically(cluster_url):
    cluster, url = cluster_url
    k8s = K8sClient(url)
    snapshot = eviction.fetch_component_labels(k8s, NODE)
    assert eviction.evict_components(
        k8s, NODE, cluster.operator_namespace, snapshot,
        timeout=3.0, poll_interval=0.02, cordon=True,
    )
    labels = cluster.node_labels(NODE)
    assert all(labels[n] == PAUSED_VALUE for n in COMPONENT_LABELS)
    assert cluster.node_unschedulable(NODE)


def test_reschedule_uncordons_and_publishes_state_atomically(cluster_url):
    cluster, url = cluster_url
    k8s = K8sClient(url)
    snapshot = eviction.fetch_component_labels(k8s, NODE)
    eviction.evict_components(
        k8s, NODE, cluster.operator_namespace, snapshot,
        timeout=3.0, poll_interval=0.02, cordon=True,
    )
    assert eviction.reschedule_components(
        k8s, NODE, snapshot, uncordon=True,
        extra_labels=eviction.state_label_dict("on"),
    )
    labels = cluster.node_labels(NODE)
    assert all(labels[n] == "true" for n in COMPONENT_LABELS)
    assert labels[CC_STATE_LABEL] == "on"
    assert labels[CC_READY_LABEL] == "true"
    assert not cluster.node_unschedulable(NODE)


def test_reconcile_request_budget(cluster_url):
    """A full reconcile must stay within the measured request budget:
    2 PATCH total (cordon+pause, restore+uncordon+state) and no
    synchronous Event POSTs on the hot path."""
    cluster, url = cluster_url
    k8s = K8sClient(url)
    counts = collections.Counter()
    orig = k8s._request

    def wrapped(method, path, **kw):
        counts[method] += 1
        return orig(method, path, **kw)

    k8s._request = wrapped
    mgr = CCManager(
        node_name=NODE,
        default_mode="on",
        host_cc=True,
        k8s=k8s,
        backend=MockBackend(num_gpus=2),
        engine=TransitionEngine(),
        config=ManagerConfig(
            evict_components=True,
            cordon_node=True,
            eviction_timeout=3.0,
            eviction_poll_interval=0.02,
        ),
    )
    mgr.read_mode_label()
    counts.clear()
    assert mgr.apply_mode("on")
    sync_patches = counts["PATCH"]
    assert sync_patches == 2, counts
    mgr.flush_events()
    # events went through POST, but only after/async
    assert counts["POST"] == 2, counts
    labels = cluster.node_labels(NODE)
    assert labels[CC_STATE_LABEL] == "on"
    assert not cluster.node_unschedulable(NODE)
