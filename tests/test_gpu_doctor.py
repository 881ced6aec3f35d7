"""Doctor on a real MI355X: full probe verdict."""

import json
import subprocess
import sys
from pathlib import Path

import pytest

torch = pytest.importorskip("torch")

REPO = Path(__file__).resolve().parent.parent
pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs an MI355X"),
]


def test_doctor_attest_on_gpu():
    proc = subprocess.run(
        [sys.executable, "-m", "k8s_cc_manager_amd.doctor", "--attest"],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
    )
    report = json.loads(proc.stdout)
    att = report["attestation"]
    assert att["library_loaded"]
    assert att["hip_device_count"] >= 1
    assert len(att["probes"]) == att["hip_device_count"]
    probe = att["probes"][0]
    assert probe["ok"]
    assert probe["max_abs_err"] == 0.0
    assert probe["fp8_max_abs_err"] == 0.0
    assert report["amdsmi"]["available"]
    assert report["native"]["kfd_version"] is not None
    assert report["native"]["kfd_gpu_nodes"]
    # cc_capable verdict requires the HOST to be TEE-enabled too — on the
    # pool box that is whatever it is; the GPU half must be satisfied.
    # (exit code reflects the full verdict)
