"""Device-contract tests against the mock backend."""

import pytest

from k8s_cc_manager_amd.device.contract import (
    BootTimeoutError,
    ResetError,
)
from k8s_cc_manager_amd.device.mock import FaultPlan, MockBackend, MockLatency


def test_enumeration():
    be = MockBackend(num_gpus=8)
    devices, count = be.find_devices()
    assert count == 8
    assert len(be.get_gpus()) == 8
    assert be.get_fabric_switches() == []  # xGMI is p2p: no switch devices
    assert len(be.get_cc_capable_gpus()) == 8
    assert len({d.bdf for d in devices}) == 8


def test_staged_mode_applies_on_reset():
    be = MockBackend(num_gpus=1)
    dev = be.device(0)
    assert dev.query_cc_mode() == "off"
    dev.set_cc_mode("on")
    # staged, not yet applied
    assert dev.query_cc_mode() == "off"
    dev.reset()
    dev.wait_for_boot()
    assert dev.query_cc_mode() == "on"


def test_fabric_mode_staging():
    be = MockBackend(num_gpus=2)
    for dev in be.get_gpus():
        dev.set_fabric_mode("on")
        dev.reset()
        dev.wait_for_boot()
        assert dev.query_fabric_mode() == "on"


def test_reset_fault_injection():
    be = MockBackend(num_gpus=2, faults=FaultPlan(fail_reset=["0000:10:00.0"]))
    dev = be.device(0)
    with pytest.raises(ResetError):
        dev.reset()
    # other devices unaffected
    be.device(1).reset()


def test_boot_hang_injection():
    be = MockBackend(num_gpus=1, faults=FaultPlan(hang_boot=["0000:10:00.0"]))
    dev = be.device(0)
    dev.reset()
    with pytest.raises(BootTimeoutError):
        dev.wait_for_boot(timeout=0.05)


def test_verify_fault_mode_does_not_latch():
    be = MockBackend(num_gpus=1, faults=FaultPlan(fail_cc_verify=["0000:10:00.0"]))
    dev = be.device(0)
    dev.set_cc_mode("on")
    dev.reset()
    dev.wait_for_boot()
    assert dev.query_cc_mode() == "off"  # did not latch


def test_flaky_reset_recovers():
    be = MockBackend(num_gpus=1, faults=FaultPlan(flaky_resets=1))
    dev = be.device(0)
    with pytest.raises(ResetError):
        dev.reset()
    dev.reset()  # second attempt succeeds


def test_latency_envelope():
    lat = MockLatency(query=0.0, stage=0.0, reset=0.01, boot=0.02)
    be = MockBackend(num_gpus=1, latency=lat)
    dev = be.device(0)
    import time

    t0 = time.monotonic()
    dev.set_cc_mode("on")
    dev.reset()
    dev.wait_for_boot()
    assert time.monotonic() - t0 >= 0.03
    assert dev.query_cc_mode() == "on"
