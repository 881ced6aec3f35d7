"""Fault-injectable fake device backend.

The reference has no test backend at all (no tests exist);
this mock is what lets the full reconcile (watch -> cordon -> evict ->
4-phase transition -> attest -> uncordon) run on a CPU-only box, with
configurable per-operation latency (so the bench's transition-latency
envelope is meaningful) and injected failures (reset loss, verify
mismatch, boot hang) for the failure-path tests.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from .contract import (
    BootTimeoutError,
    CCDevice,
    DeviceBackend,
    FABRIC_OFF,
    ResetError,
)


@dataclass
class MockLatency:
    """Seconds each device operation takes (synthetic envelope).

    Defaults are near-zero so unit tests are instant; the bench scales
    them up to emulate the real cost structure (stage write ~ms, FLR
    reset and boot-wait dominate — reference cost structure at
    /root/reference/main.py:502-529).
    """

    query: float = 0.0
    stage: float = 0.0
    reset: float = 0.0
    boot: float = 0.0


@dataclass
class FaultPlan:
    """Injected failures, keyed by bdf."""

    #: bdfs whose reset() raises ResetError
    fail_reset: List[str] = field(default_factory=list)
    #: bdfs whose post-reset CC mode readback is wrong
    fail_cc_verify: List[str] = field(default_factory=list)
    #: bdfs whose post-reset fabric readback is wrong
    fail_fabric_verify: List[str] = field(default_factory=list)
    #: bdfs whose wait_for_boot() never completes
    hang_boot: List[str] = field(default_factory=list)
    #: number of initial reset() calls to fail before succeeding (flaky)
    flaky_resets: int = 0


class MockDevice(CCDevice):
    def __init__(
        self,
        bdf: str,
        name: str = "AMD Instinct MI355X (mock)",
        cc_mode: str = "off",
        fabric_mode: str = FABRIC_OFF,
        cc_capable: bool = True,
        fabric_capable: bool = True,
        latency: Optional[MockLatency] = None,
        faults: Optional[FaultPlan] = None,
    ):
        self.bdf = bdf
        self.name = name
        self._lock = threading.Lock()
        self._cc_mode = cc_mode
        self._fabric_mode = fabric_mode
        self._staged_cc: Optional[str] = None
        self._staged_fabric: Optional[str] = None
        self._cc_capable = cc_capable
        self._fabric_capable = fabric_capable
        self._lat = latency or MockLatency()
        self._faults = faults or FaultPlan()
        self._booted = True
        self._boot_ready_at = 0.0
        self._reset_attempts = 0
        # Telemetry for tests/bench
        self.op_log: List[str] = []

    # -- helpers --------------------------------------------------------
    def _sleep(self, seconds: float) -> None:
        if seconds > 0:
            time.sleep(seconds)

    def _log(self, op: str) -> None:
        self.op_log.append(op)

    # -- classification -------------------------------------------------
    def is_gpu(self) -> bool:
        return True

    @property
    def cc_query_supported(self) -> bool:
        return self._cc_capable

    @property
    def fabric_query_supported(self) -> bool:
        return self._fabric_capable

    # -- CC mode --------------------------------------------------------
    def query_cc_mode(self) -> str:
        self._sleep(self._lat.query)
        with self._lock:
            self._log("query_cc")
            return self._cc_mode

    def set_cc_mode(self, mode: str) -> None:
        self._sleep(self._lat.stage)
        with self._lock:
            self._log(f"stage_cc:{mode}")
            self._staged_cc = mode

    # -- fabric mode ----------------------------------------------------
    def query_fabric_mode(self) -> str:
        self._sleep(self._lat.query)
        with self._lock:
            self._log("query_fabric")
            return self._fabric_mode

    def set_fabric_mode(self, mode: str) -> None:
        self._sleep(self._lat.stage)
        with self._lock:
            self._log(f"stage_fabric:{mode}")
            self._staged_fabric = mode

    # -- lifecycle ------------------------------------------------------
    def reset(self) -> None:
        self._sleep(self._lat.reset)
        with self._lock:
            self._log("reset")
            self._reset_attempts += 1
            if self._faults.flaky_resets >= self._reset_attempts:
                raise ResetError(f"{self.bdf}: injected flaky reset")
            if self.bdf in self._faults.fail_reset:
                raise ResetError(f"{self.bdf}: injected reset failure")
            # Apply staged modes (this is the semantic the real hardware
            # has: the staged register takes effect across FLR).
            if self._staged_cc is not None:
                if self.bdf in self._faults.fail_cc_verify:
                    pass  # mode silently does NOT latch -> verify fails
                else:
                    self._cc_mode = self._staged_cc
                self._staged_cc = None
            if self._staged_fabric is not None:
                if self.bdf in self._faults.fail_fabric_verify:
                    pass
                else:
                    self._fabric_mode = self._staged_fabric
                self._staged_fabric = None
            self._booted = False
            self._boot_ready_at = time.monotonic() + self._lat.boot

    def wait_for_boot(self, timeout: float = 60.0) -> None:
        if self.bdf in self._faults.hang_boot:
            # Simulate a bricked device but respect the caller's deadline.
            time.sleep(min(timeout, 0.2))
            raise BootTimeoutError(f"{self.bdf}: injected boot hang")
        remaining = self._boot_ready_at - time.monotonic()
        if remaining > 0:
            if remaining > timeout:
                time.sleep(timeout)
                raise BootTimeoutError(f"{self.bdf}: boot exceeded {timeout}s")
            time.sleep(remaining)
        with self._lock:
            self._booted = True
            self._log("boot")


class MockBackend(DeviceBackend):
    """A node of ``num_gpus`` mock MI355X devices."""

    def __init__(
        self,
        num_gpus: int = 8,
        latency: Optional[MockLatency] = None,
        faults: Optional[FaultPlan] = None,
        initial_cc_mode: str = "off",
        initial_fabric_mode: str = FABRIC_OFF,
        cc_capable: bool = True,
        fabric_capable: bool = True,
    ):
        self._devices: List[MockDevice] = [
            MockDevice(
                bdf=f"0000:{0x10 + 8 * i:02x}:00.0",
                cc_mode=initial_cc_mode,
                fabric_mode=initial_fabric_mode,
                cc_capable=cc_capable,
                fabric_capable=fabric_capable,
                latency=latency,
                faults=faults,
            )
            for i in range(num_gpus)
        ]

    def find_devices(self) -> Tuple[List[CCDevice], int]:
        return list(self._devices), len(self._devices)

    # test hooks --------------------------------------------------------
    def device(self, idx: int) -> MockDevice:
        return self._devices[idx]

    def modes(self) -> Dict[str, str]:
        return {d.bdf: d._cc_mode for d in self._devices}
