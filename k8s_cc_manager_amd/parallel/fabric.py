"""Fabric-wide (xGMI-hive) transition barriers.

The reference's PPCIe pattern stages mode on ALL devices before any
reset so the NVLink fabric applies consistently (/root/reference/
main.py:319-368). On MI355X the xGMI hive has the same invariant; with
transitions now CONCURRENT (and, in the bench's scaling harness, spread
one-process-per-GPU), the stage/reset seam needs an explicit barrier:

- :class:`FabricBarrier` — in-process (threading) form; the
  DeviceExecutor's phase gather already provides it, this class exists
  so multi-manager tests can share one.
- :class:`DistFabricBarrier` — cross-process form over
  ``torch.distributed`` (gloo on CPU, RCCL on ROCm), used when each
  rank owns one GPU of the hive. torch is imported lazily so the
  control-plane daemon itself never depends on it.

Barrier counts are matched by construction: the engine's phase-1 and
stage->reset seams call wait() UNCONDITIONALLY for every participant
(even one with no devices to reset), so mixed initial states across
participants cannot deadlock the hive.
"""

from __future__ import annotations

import logging
import threading

logger = logging.getLogger(__name__)


class FabricBarrier:
    """Reusable barrier for N in-process participants."""

    def __init__(self, parties: int):
        self.parties = parties
        self._barrier = threading.Barrier(parties)

    def wait(self, timeout: float = 300.0) -> None:
        self._barrier.wait(timeout=timeout)


class DistFabricBarrier:
    """Cross-process barrier over an initialized torch.distributed group.

    No-op when torch.distributed is not initialized or world_size == 1,
    so single-process reconciles pay nothing.
    """

    def __init__(self) -> None:
        try:
            import torch.distributed as dist  # local import by design

            self._dist = dist if dist.is_available() and dist.is_initialized() else None
        except Exception:  # pragma: no cover - torch absent
            self._dist = None

    @property
    def active(self) -> bool:
        return self._dist is not None and self._dist.get_world_size() > 1

    def wait(self, timeout: float = 300.0) -> None:
        if self.active:
            self._dist.barrier()
