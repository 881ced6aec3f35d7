"""Concurrent per-device execution.

The reference transitions devices in sequential Python loops
(/root/reference/main.py:486-529), giving ~linear scaling in GPU count.
On an 8-GPU MI355X node each phase here fans out over a thread pool
(device operations are I/O-bound: sysfs writes, FLR wait, firmware
boot), so the per-GPU cost curve flattens — phase barriers (gather all
futures) preserve the load-bearing stage-all-then-reset-all invariant.
"""

from __future__ import annotations

import concurrent.futures as cf
import logging
from typing import Callable, Dict, List, Sequence, Tuple, TypeVar

from ..device.contract import CCDevice

logger = logging.getLogger(__name__)

T = TypeVar("T")


class PerDeviceError(Exception):
    """One or more devices failed inside a concurrent phase."""

    def __init__(self, phase: str, errors: Dict[str, BaseException]):
        self.phase = phase
        self.errors = errors
        detail = "; ".join(f"{bdf}: {err}" for bdf, err in sorted(errors.items()))
        super().__init__(f"phase {phase!r} failed on {len(errors)} device(s): {detail}")


class DeviceExecutor:
    """Thread pool sized to the node's GPU count (max 16)."""

    def __init__(self, max_workers: int = 16):
        self._max_workers = max_workers

    def run(
        self,
        phase: str,
        devices: Sequence[CCDevice],
        fn: Callable[[CCDevice], T],
    ) -> Dict[str, T]:
        """Apply ``fn`` to every device concurrently; gather all results
        (a barrier). Raises :class:`PerDeviceError` carrying every
        failure — never a partial, silent success."""
        if not devices:
            return {}
        results: Dict[str, T] = {}
        errors: Dict[str, BaseException] = {}
        workers = min(self._max_workers, len(devices))
        if workers == 1:
            for dev in devices:
                try:
                    results[dev.bdf] = fn(dev)
                except BaseException as e:
                    errors[dev.bdf] = e
        else:
            with cf.ThreadPoolExecutor(
                max_workers=workers, thread_name_prefix=f"ccdev-{phase}"
            ) as pool:
                futures = {pool.submit(fn, dev): dev for dev in devices}
                for fut in cf.as_completed(futures):
                    dev = futures[fut]
                    try:
                        results[dev.bdf] = fut.result()
                    except BaseException as e:
                        errors[dev.bdf] = e
        if errors:
            raise PerDeviceError(phase, errors)
        return results

    def partition(
        self,
        phase: str,
        devices: Sequence[CCDevice],
        pred: Callable[[CCDevice], bool],
    ) -> Tuple[List[CCDevice], List[CCDevice]]:
        """Concurrently evaluate a predicate; return (true, false) lists
        in stable device order."""
        verdicts = self.run(phase, devices, pred)
        yes = [d for d in devices if verdicts[d.bdf]]
        no = [d for d in devices if not verdicts[d.bdf]]
        return yes, no
