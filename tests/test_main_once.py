"""Full CLI end-to-end: python -m k8s_cc_manager_amd --once against the
fake API server via kubeconfig, mock device backend."""

import os
import subprocess
import sys
from pathlib import Path

import yaml

from k8s_cc_manager_amd.k8s.eviction import COMPONENT_LABELS
from k8s_cc_manager_amd.labels import (
    CC_MODE_LABEL,
    CC_READY_LABEL,
    CC_STATE_LABEL,
)

REPO = Path(__file__).resolve().parent.parent


def _kubeconfig(tmp_path, url):
    kc = {
        "apiVersion": "v1",
        "kind": "Config",
        "current-context": "t",
        "contexts": [{"name": "t", "context": {"cluster": "c", "user": "u"}}],
        "clusters": [{"name": "c", "cluster": {"server": url}}],
        "users": [{"name": "u", "user": {}}],
    }
    p = tmp_path / "kubeconfig"
    p.write_text(yaml.safe_dump(kc))
    return p


def test_cli_once_applies_mode(fake_cluster, tmp_path):
    cluster, url = fake_cluster
    labels = {name: "true" for name in COMPONENT_LABELS}
    labels[CC_MODE_LABEL] = "devtools"
    cluster.add_node("clinode", labels=labels)

    env = dict(os.environ)
    env.update(
        {
            "NODE_NAME": "clinode",
            "CC_DEVICE_BACKEND": "mock",
            "EVICT_OPERATOR_COMPONENTS": "true",
            "CORDON_NODE": "true",
            "CC_READINESS_FILE": str(tmp_path / ".ready"),
            "CC_EVENT_LOG": str(tmp_path / "events.jsonl"),
            "PYTHONPATH": str(REPO),
        }
    )
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "k8s_cc_manager_amd",
            "--once",
            "--kubeconfig",
            str(_kubeconfig(tmp_path, url)),
        ],
        capture_output=True,
        text=True,
        timeout=120,
        env=env,
        cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    out_labels = cluster.node_labels("clinode")
    assert out_labels[CC_STATE_LABEL] == "devtools"
    assert out_labels[CC_READY_LABEL] == ""
    # capability label published (mock GPUs are CC-capable; host probe
    # depends on the test host -> value is true or false but present)
    assert out_labels.get("amd.com/gpu.cc.capable") in ("true", "false")
    assert not cluster.node_unschedulable("clinode")


def test_cli_requires_node_name(tmp_path):
    env = dict(os.environ)
    env.pop("NODE_NAME", None)
    env["PYTHONPATH"] = str(REPO)
    proc = subprocess.run(
        [sys.executable, "-m", "k8s_cc_manager_amd", "--once"],
        capture_output=True,
        text=True,
        timeout=60,
        env=env,
        cwd=REPO,
    )
    assert proc.returncode == 1
    assert "NODE_NAME" in proc.stderr + proc.stdout
