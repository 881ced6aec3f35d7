"""DeviceExecutor unit tests: aggregation, partition order, single-worker path."""

import threading

import pytest

from k8s_cc_manager_amd.device.mock import MockBackend
from k8s_cc_manager_amd.parallel.executor import DeviceExecutor, PerDeviceError


def test_run_collects_all_results():
    be = MockBackend(num_gpus=4)
    devices, _ = be.find_devices()
    ex = DeviceExecutor()
    out = ex.run("query", devices, lambda d: d.query_cc_mode())
    assert set(out) == {d.bdf for d in devices}
    assert all(v == "off" for v in out.values())


def test_run_aggregates_every_failure():
    be = MockBackend(num_gpus=4)
    devices, _ = be.find_devices()
    bad = {devices[1].bdf, devices[3].bdf}

    def fn(d):
        if d.bdf in bad:
            raise RuntimeError(f"boom {d.bdf}")
        return "ok"

    with pytest.raises(PerDeviceError) as ei:
        DeviceExecutor().run("phase-x", devices, fn)
    err = ei.value
    assert err.phase == "phase-x"
    assert set(err.errors) == bad
    # message names every failing device
    for bdf in bad:
        assert bdf in str(err)


def test_partition_preserves_device_order():
    be = MockBackend(num_gpus=6)
    devices, _ = be.find_devices()
    yes, no = DeviceExecutor().partition(
        "p", devices, lambda d: devices.index(d) % 2 == 0
    )
    assert [devices.index(d) for d in yes] == [0, 2, 4]
    assert [devices.index(d) for d in no] == [1, 3, 5]


def test_empty_device_list():
    assert DeviceExecutor().run("p", [], lambda d: 1) == {}


def test_single_device_runs_inline():
    """One device must not spawn a pool (runs on the calling thread)."""
    be = MockBackend(num_gpus=1)
    devices, _ = be.find_devices()
    caller = threading.current_thread().name
    seen = []
    DeviceExecutor().run("p", devices, lambda d: seen.append(threading.current_thread().name))
    assert seen == [caller]


def test_concurrency_actually_parallel():
    be = MockBackend(num_gpus=8)
    devices, _ = be.find_devices()
    barrier = threading.Barrier(8, timeout=10)

    def fn(d):
        barrier.wait()  # deadlocks unless all 8 run concurrently
        return True

    out = DeviceExecutor().run("p", devices, fn)
    assert len(out) == 8


def test_in_process_fabric_barrier_synchronizes_two_engines():
    """Two in-process hive managers sharing one FabricBarrier cannot
    cross the stage->reset seam independently."""
    import time

    from k8s_cc_manager_amd.core.transition import TransitionEngine
    from k8s_cc_manager_amd.parallel.fabric import FabricBarrier

    barrier = FabricBarrier(parties=2)
    backends = [MockBackend(num_gpus=1), MockBackend(num_gpus=1)]
    engines = [TransitionEngine(barrier=barrier) for _ in backends]
    results = [None, None]

    def run(i):
        devices, _ = backends[i].find_devices()
        results[i] = engines[i].apply_fabric_mode(devices)

    t0 = threading.Thread(target=run, args=(0,))
    t1 = threading.Thread(target=run, args=(1,))
    t0.start()
    time.sleep(0.05)  # stagger: the barrier must hold engine 0 back
    t1.start()
    t0.join(timeout=30)
    t1.join(timeout=30)
    assert results[0].ok and results[1].ok
    assert backends[0].device(0).query_fabric_mode() == "on"
    assert backends[1].device(0).query_fabric_mode() == "on"


def test_fabric_barrier_mixed_initial_state_no_deadlock():
    """One participant's hive is already fabric-on, the other's is off:
    barrier counts still match (unconditional seam waits)."""
    import time

    from k8s_cc_manager_amd.core.transition import TransitionEngine
    from k8s_cc_manager_amd.parallel.fabric import FabricBarrier

    barrier = FabricBarrier(parties=2)
    be_on = MockBackend(num_gpus=1, initial_fabric_mode="on")
    be_off = MockBackend(num_gpus=1)
    backends = [be_on, be_off]
    engines = [TransitionEngine(barrier=barrier) for _ in backends]
    results = [None, None]

    def run(i):
        devices, _ = backends[i].find_devices()
        results[i] = engines[i].apply_fabric_mode(devices)

    threads = [threading.Thread(target=run, args=(i,)) for i in range(2)]
    t0 = time.monotonic()
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert time.monotonic() - t0 < 25, "barrier deadlock"
    assert results[0].ok and results[1].ok
    assert all(b.device(0).query_fabric_mode() == "on" for b in backends)
