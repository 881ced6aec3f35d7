"""Deep attestation: rocprof counter readback around the probe.

The standard probe (:mod:`.attest`) self-verifies with on-device
checksums. Deep attestation additionally runs the probe under
``rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES`` and asserts the hardware
performance counters actually saw matrix-pipe activity — external,
counter-level evidence that the MFMA units executed (the north-star
requirement: counters read back with rocprof to confirm the device is
live inside the TEE, not just that a kernel claimed success).

Enabled per-transition with ``CC_ATTEST_DEEP=1`` (it costs a subprocess
+ profiler session, ~seconds, so the default gate is the fast probe).
"""

from __future__ import annotations

import csv
import logging
import os
import shutil
import subprocess
import sys
import tempfile
from pathlib import Path

from .attest import AttestationError

logger = logging.getLogger(__name__)

MFMA_COUNTER = "SQ_VALU_MFMA_BUSY_CYCLES"


def rocprof_available() -> bool:
    return shutil.which("rocprofv3") is not None


def _sum_counter_csv(outdir: Path, counter: str) -> float:
    total = 0.0
    seen = False
    for path in outdir.rglob("*counter_collection.csv"):
        with open(path) as f:
            for row in csv.DictReader(f):
                name = row.get("Counter_Name") or row.get("counter_name") or ""
                if name.strip() == counter:
                    seen = True
                    try:
                        total += float(row.get("Counter_Value") or row.get("counter_value") or 0)
                    except ValueError:
                        pass
    if not seen:
        raise AttestationError(
            f"rocprof output in {outdir} has no {counter} rows"
        )
    return total


def deep_attest_device(device_index: int, gemm_dim: int = 1024,
                       timeout: float = 300.0) -> float:
    """Run the probe under rocprofv3; return MFMA busy cycles (>0) or
    raise :class:`AttestationError`."""
    if not rocprof_available():
        raise AttestationError("rocprofv3 not on PATH; deep attestation unavailable")
    repo_root = str(Path(__file__).resolve().parent.parent.parent)
    outdir = Path(tempfile.mkdtemp(prefix="cc-deep-attest-"))
    payload = (
        f"import sys; sys.path.insert(0, {repo_root!r}); "
        "from k8s_cc_manager_amd.ops import attest; "
        f"r = attest.attest_device({device_index}, gemm_dim={gemm_dim}); "
        "assert r.ok"
    )
    cmd = [
        "rocprofv3",
        "--pmc",
        MFMA_COUNTER,
        "-d",
        str(outdir),
        "-o",
        "probe",
        "--output-format",
        "csv",
        "--",
        sys.executable,
        "-c",
        payload,
    ]
    env = dict(os.environ)
    env.setdefault("TMPDIR", "/tmp")
    try:
        proc = subprocess.run(
            cmd,
            capture_output=True,
            text=True,
            timeout=timeout,
            env=env,
            cwd="/tmp",
        )
    except subprocess.TimeoutExpired as e:
        raise AttestationError(f"deep attestation timed out: {e}") from e
    finally:
        pass
    if proc.returncode != 0:
        shutil.rmtree(outdir, ignore_errors=True)
        raise AttestationError(
            f"probe under rocprof failed rc={proc.returncode}: "
            f"{proc.stderr[-500:]}"
        )
    try:
        cycles = _sum_counter_csv(outdir, MFMA_COUNTER)
    finally:
        shutil.rmtree(outdir, ignore_errors=True)
    if cycles <= 0:
        raise AttestationError(
            f"device {device_index}: rocprof saw zero MFMA busy cycles — "
            "matrix pipes did not execute"
        )
    logger.info(
        "deep-attested device %d: %s=%.0f", device_index, MFMA_COUNTER, cycles
    )
    return cycles
