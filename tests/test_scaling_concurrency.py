"""Weak-scaling shape of the concurrent engine: transitioning 8 mock
GPUs must cost ~the same wall time as 1 (the reference's serial loops
are ~linear in GPU count — SURVEY.md §6)."""

import time

from k8s_cc_manager_amd.core.transition import TransitionEngine
from k8s_cc_manager_amd.device.mock import MockBackend, MockLatency


def _transition_wall(num_gpus: int) -> float:
    lat = MockLatency(reset=0.02, boot=0.03)
    be = MockBackend(num_gpus=num_gpus, latency=lat)
    devices, _ = be.find_devices()
    engine = TransitionEngine()
    t0 = time.monotonic()
    report = engine.apply_cc_mode(devices, be.get_gpus(), "on")
    wall = time.monotonic() - t0
    assert report.ok
    return wall


def test_weak_scaling_flat_1_to_8():
    w1 = _transition_wall(1)
    w8 = _transition_wall(8)
    # serial would be ~8x; concurrent must stay well under that even
    # on a loaded CI box (xdist runs tests in parallel)
    assert w8 < 4.0 * w1, f"w1={w1:.3f}s w8={w8:.3f}s"


def test_16_gpus_still_bounded():
    w16 = _transition_wall(16)
    assert w16 < 1.0, f"16-GPU concurrent transition took {w16:.3f}s"
