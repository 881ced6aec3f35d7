"""Intra-process race safety: concurrent apply_mode calls serialize on
the manager's transition lock (SURVEY.md §5 — the concurrent design
needs a fabric-wide lock the reference's single thread never did)."""

import threading

from k8s_cc_manager_amd.core.manager import CCManager, ManagerConfig
from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend, MockLatency
from k8s_cc_manager_amd.k8s.client import K8sClient


def test_concurrent_apply_mode_serializes(fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node("node0")
    backend = MockBackend(num_gpus=2, latency=MockLatency(reset=0.02))
    in_transition = []
    overlap = []

    class ObservingEngine(TransitionEngine):
        def apply_cc_mode(self, all_devices, gpus, mode):
            if in_transition:
                overlap.append(mode)
            in_transition.append(mode)
            try:
                return super().apply_cc_mode(all_devices, gpus, mode)
            finally:
                in_transition.pop()

    mgr = CCManager(
        node_name="node0",
        default_mode="on",
        host_cc=True,
        k8s=K8sClient(url),
        backend=backend,
        engine=ObservingEngine(),
        config=ManagerConfig(evict_components=False, cordon_node=False),
    )

    threads = [
        threading.Thread(target=mgr.apply_mode, args=(m,))
        for m in ("on", "off", "devtools", "on")
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert overlap == [], f"transitions overlapped: {overlap}"
    # final state is one of the requested modes, consistently applied
    final = {d.query_cc_mode() for d in backend.get_gpus()}
    assert len(final) == 1 and final.pop() in ("on", "off", "devtools")
