"""Per-phase transition timers.

The reference measures nothing (no timing anywhere); the
BASELINE metric (CC-mode transition sec/GPU, reconcile GPUs/sec) needs
per-phase wall-clock, so every transition here carries a PhaseTimer.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional, Tuple


class StopWatch:
    def __init__(self) -> None:
        self._t0 = time.monotonic()

    def elapsed(self) -> float:
        return time.monotonic() - self._t0

    def restart(self) -> float:
        now = time.monotonic()
        dt = now - self._t0
        self._t0 = now
        return dt


class PhaseTimer:
    """Records named (phase, seconds) intervals in order."""

    def __init__(self) -> None:
        self.phases: List[Tuple[str, float]] = []
        self._current: Optional[str] = None
        self._t0 = 0.0

    def start(self, phase: str) -> None:
        self.stop()
        self._current = phase
        self._t0 = time.monotonic()

    def stop(self) -> None:
        if self._current is not None:
            self.phases.append((self._current, time.monotonic() - self._t0))
            self._current = None

    def as_dict(self) -> Dict[str, float]:
        out: Dict[str, float] = {}
        for name, dt in self.phases:
            out[name] = out.get(name, 0.0) + dt
        return out

    def total(self) -> float:
        return sum(dt for _, dt in self.phases)
