"""Prometheus metrics for transition latency and reconcile outcomes.

The reference exposes no metrics (SURVEY.md §5 — observability is two
labels + a readiness file); this is a north-star addition. All metric
use is behind this module so prometheus_client stays optional.
"""

from __future__ import annotations

import logging
from typing import Dict, Optional

logger = logging.getLogger(__name__)

try:  # pragma: no cover - exercised implicitly
    from prometheus_client import Counter, Gauge, Histogram, start_http_server

    _HAVE_PROM = True
except Exception:  # pragma: no cover
    _HAVE_PROM = False


class Metrics:
    """Singleton-ish metrics registry; no-ops when prometheus is absent."""

    def __init__(self) -> None:
        self.enabled = _HAVE_PROM
        if not self.enabled:  # pragma: no cover
            return
        self.transitions_total = Counter(
            "cc_transitions_total",
            "CC-mode transitions attempted",
            ["mode", "outcome"],
        )
        self.transition_seconds = Histogram(
            "cc_transition_seconds",
            "Whole-transition wall time (all GPUs of the node)",
            ["mode"],
            buckets=(0.1, 0.5, 1, 2, 5, 10, 20, 40, 80, 160, 320),
        )
        self.phase_seconds = Histogram(
            "cc_transition_phase_seconds",
            "Per-phase wall time inside a transition",
            ["phase"],
            buckets=(0.01, 0.05, 0.1, 0.5, 1, 2, 5, 10, 20, 40, 80),
        )
        self.devices_managed = Gauge(
            "cc_devices_managed", "CC-capable GPUs discovered on the node"
        )
        self.attest_failures = Counter(
            "cc_attest_failures_total", "Post-reset attestation probe failures"
        )

    def observe_transition(
        self, mode: str, ok: bool, seconds: float, phases: Optional[Dict[str, float]] = None
    ) -> None:
        if not self.enabled:
            return
        self.transitions_total.labels(mode=mode, outcome="ok" if ok else "failed").inc()
        self.transition_seconds.labels(mode=mode).observe(seconds)
        for phase, dt in (phases or {}).items():
            self.phase_seconds.labels(phase=phase).observe(dt)

    def serve(self, port: int) -> None:
        if not self.enabled:  # pragma: no cover
            logger.warning("prometheus_client unavailable; metrics disabled")
            return
        start_http_server(port)
        logger.info("metrics endpoint on :%d/metrics", port)


METRICS = Metrics()
