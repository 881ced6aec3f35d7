"""Worker for test_dist_fabric_barrier_world2 (launched by torchrun)."""

import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch.distributed as dist  # noqa: E402

from k8s_cc_manager_amd.core.transition import TransitionEngine  # noqa: E402
from k8s_cc_manager_amd.device.mock import MockBackend  # noqa: E402
from k8s_cc_manager_amd.parallel.fabric import DistFabricBarrier  # noqa: E402


def main() -> None:
    dist.init_process_group("gloo")
    barrier = DistFabricBarrier()
    assert barrier.active
    be = MockBackend(num_gpus=1)
    devices, _ = be.find_devices()
    engine = TransitionEngine(barrier=barrier)
    report = engine.apply_cc_mode(devices, devices, "on")
    assert report.ok, report.error
    assert be.device(0).query_cc_mode() == "on"
    # fabric-wide (ppcie) transition across ranks: the xGMI-hive
    # stage-all/reset-all seam synchronizes via the same barrier
    report = engine.apply_fabric_mode(devices)
    assert report.ok, report.error
    assert be.device(0).query_fabric_mode() == "on"
    dist.barrier()
    dist.destroy_process_group()
    print("RANK_OK", os.environ.get("RANK"))


if __name__ == "__main__":
    main()
