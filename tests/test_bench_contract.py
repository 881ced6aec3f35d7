"""bench.py contract: runs on CPU (mock tier) and emits the JSON line."""

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_cpu_mock(tmp_path):
    out = tmp_path / "bench.json"
    proc = subprocess.run(
        [
            sys.executable,
            str(REPO / "bench.py"),
            "--mock",
            "--gpus",
            "2",
            "--steps",
            "3",
            "--warmup",
            "1",
            "--json-out",
            str(out),
        ],
        capture_output=True,
        text=True,
        timeout=240,
        cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = out.read_text().strip()
    result = json.loads(line)
    assert result["metric"] == "reconcile_gpus_per_sec"
    assert result["n_gpus"] == 2
    assert result["steps"] == 3
    assert result["warmup"] == 1
    assert result["value"] > 0
    assert result["higher_is_better"] is True
    assert result["scaling"] == "weak"
    assert result["data"] == "synthetic"
    assert result["config"]["model"] == "cc-mode-transition"
    assert result["config"]["eviction"] is True
    # the stdout JSON line is parseable too
    json_lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert json_lines and json.loads(json_lines[-1])["value"] > 0
