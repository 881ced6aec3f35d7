"""Watch-loop edge behavior: flapping labels queue behind a blocking
apply (reference behavior, SURVEY.md §7 hard-part (d)), non-410 ERROR
events count against the error budget, bookmarks advance the RV."""

import threading
import time

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend, MockLatency
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.labels import CC_MODE_LABEL, CC_STATE_LABEL

NODE = "node0"


def _mgr(cluster, url, backend=None):
    return CCManager(
        node_name=NODE,
        default_mode="off",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend or MockBackend(num_gpus=1),
        engine=TransitionEngine(),
        config=ManagerConfig(
            evict_components=False,
            cordon_node=False,
            watch_timeout_seconds=2,
            reconnect_backoff=0.05,
            readiness_file="/tmp/.cc-watch-test-ready",
        ),
    )


def _wait_state(cluster, value, timeout=8.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if cluster.node_labels(NODE).get(CC_STATE_LABEL) == value:
            return True
        time.sleep(0.02)
    return False


def test_flapping_labels_settle_on_last_value(fake_cluster):
    """Rapid flips during a slow transition: the final applied state is
    the LAST label value (intermediate values may be skipped — events
    are processed in order after the blocking apply returns)."""
    cluster, url = fake_cluster
    cluster.add_node(NODE, labels={CC_MODE_LABEL: "off"})
    backend = MockBackend(num_gpus=1, latency=MockLatency(reset=0.05, boot=0.05))
    mgr = _mgr(cluster, url, backend)
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    assert _wait_state(cluster, "off")

    for value in ("on", "off", "devtools", "on", "devtools"):
        cluster.set_node_label(NODE, CC_MODE_LABEL, value)
        time.sleep(0.01)
    assert _wait_state(cluster, "devtools"), cluster.node_labels(NODE)
    # devices landed on the final mode
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        if backend.modes()["0000:10:00.0"] == "devtools":
            break
        time.sleep(0.02)
    assert backend.modes()["0000:10:00.0"] == "devtools"
    mgr.stop_event.set()
    t.join(timeout=5)


def test_bookmark_events_ignored_but_rv_advances(fake_cluster):
    """BOOKMARK events must not trigger a transition."""
    cluster, url = fake_cluster
    cluster.add_node(NODE, labels={CC_MODE_LABEL: "off"})
    mgr = _mgr(cluster, url)
    applied = []
    original = mgr.apply_mode
    mgr.apply_mode = lambda m: applied.append(m) or original(m)
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    assert _wait_state(cluster, "off")
    n_applies = len(applied)
    # unrelated label churn produces MODIFIED events with unchanged
    # cc.mode -> no further applies
    for i in range(5):
        cluster.set_node_label(NODE, "unrelated", str(i))
    time.sleep(0.5)
    assert len(applied) == n_applies
    mgr.stop_event.set()
    t.join(timeout=5)
