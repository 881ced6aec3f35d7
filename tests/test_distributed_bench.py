"""Multi-process correctness of the distributed path (gloo, ws=2, CPU).

The driver runs bench.py under torch.distributed.run with one rank per
GPU; this test runs the exact same launch shape on CPU (gloo + mock
tier) so the rank plumbing, the DistFabricBarrier stage/reset seam and
the MAX-over-ranks timing reduction are covered without a GPU.
"""

import json
import os
import socket
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

torch = pytest.importorskip("torch")


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_bench_two_ranks_gloo(tmp_path):
    out = tmp_path / "bench2.json"
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node=2",
            "--master-addr=127.0.0.1",
            f"--master-port={_free_port()}",
            str(REPO / "bench.py"),
            "--mock",
            "--gpus",
            "2",
            "--steps",
            "2",
            "--warmup",
            "1",
            "--json-out",
            str(out),
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
        env=env,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    result = json.loads(out.read_text())
    assert result["n_gpus"] == 2  # world=2 ranks x 1 GPU each
    assert result["value"] > 0
    assert result["steps"] == 2


def test_dist_fabric_barrier_world2():
    """DistFabricBarrier.wait() across 2 gloo ranks inside the
    transition engine (stage-all before reset-all across processes)."""
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node=2",
            "--master-addr=127.0.0.1",
            f"--master-port={_free_port()}",
            str(REPO / "tests" / "_dist_barrier_worker.py"),
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
        env=env,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert proc.stdout.count("RANK_OK") == 2
