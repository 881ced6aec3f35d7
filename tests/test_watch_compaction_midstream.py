"""A watch whose cursor is trimmed away MID-STREAM must get a 410 ERROR
event (not silently miss events) and the manager must resync."""

import threading
import time

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend
from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.labels import CC_MODE_LABEL, CC_STATE_LABEL

NODE = "node0"


def test_midstream_compaction_emits_410():
    from k8s_cc_manager_amd.k8s.fakecluster import FakeCluster

    cluster = FakeCluster(event_log_max=100)
    url = cluster.start()
    cluster.add_node(NODE)
    k8s = K8sClient(url)
    rv = k8s.get_node(NODE)["metadata"]["resourceVersion"]

    seen = []

    def watcher():
        # deliberately SLOW consumer: backpressure lets the server-side
        # event log trim past this watcher's cursor
        for event in k8s.watch_node(NODE, resource_version=rv, timeout_seconds=30):
            seen.append(event)
            if event.get("type") == "ERROR":
                return
            time.sleep(0.005)

    t = threading.Thread(target=watcher)
    t.start()
    time.sleep(0.2)
    # Burst enough churn BYTES that the server-side handler must block
    # on the socket (the consumer is slower than the producer): kernel
    # socket buffers can absorb hundreds of KB, so the burst has to be
    # megabytes — a small burst fits in the buffers, the handler never
    # falls behind the trim, and no 410 is provoked (seen as a flake on
    # fast boxes).
    pad = "x" * 120
    for i in range(30000):
        cluster.set_node_label(NODE, "churn", f"{pad}{i}")
    t.join(timeout=60)
    cluster.stop()
    assert any(
        e.get("type") == "ERROR" and e["object"].get("code") == 410 for e in seen
    ), f"no 410 in {len(seen)} events"


def test_manager_survives_midstream_compaction():
    from k8s_cc_manager_amd.k8s.fakecluster import FakeCluster

    cluster = FakeCluster(event_log_max=200)
    url = cluster.start()
    cluster.add_node(NODE, labels={CC_MODE_LABEL: "off"})
    backend = MockBackend(num_gpus=1)
    mgr = CCManager(
        node_name=NODE,
        default_mode="off",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend,
        engine=TransitionEngine(),
        config=ManagerConfig(
            evict_components=False,
            cordon_node=False,
            watch_timeout_seconds=5,
            reconnect_backoff=0.05,
            readiness_file="/tmp/.cc-compact-test",
        ),
    )
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        if cluster.node_labels(NODE).get(CC_STATE_LABEL) == "off":
            break
        time.sleep(0.02)

    # churn past the trim boundary, then flip the mode: the 410-resync
    # path must pick up the new desired mode
    for i in range(400):
        cluster.set_node_label(NODE, "churn", str(i))
    cluster.set_node_label(NODE, CC_MODE_LABEL, "devtools")
    deadline = time.monotonic() + 15
    while time.monotonic() < deadline:
        if cluster.node_labels(NODE).get(CC_STATE_LABEL) == "devtools":
            break
        time.sleep(0.05)
    assert cluster.node_labels(NODE).get(CC_STATE_LABEL) == "devtools"
    mgr.stop_event.set()
    t.join(timeout=10)
    cluster.stop()
