"""Notices generator against a synthetic site-packages tree."""

import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
SCRIPT = REPO / "hack" / "gen_third_party_notices.py"


def _mk_dist(sp: Path, name: str, version: str, license_name: str = "",
             license_text: str = "", classifier: str = ""):
    d = sp / f"{name}-{version}.dist-info"
    d.mkdir(parents=True)
    meta = [f"Metadata-Version: 2.1", f"Name: {name}", f"Version: {version}"]
    if license_name:
        meta.append(f"License-Expression: {license_name}")
    if classifier:
        meta.append(f"Classifier: {classifier}")
    (d / "METADATA").write_text("\n".join(meta) + "\n")
    if license_text:
        (d / "LICENSE").write_text(license_text)
    return d


def run(*args):
    return subprocess.run(
        [sys.executable, str(SCRIPT), *args], capture_output=True, text=True
    )


def test_generates_notices(tmp_path):
    sp = tmp_path / "sp"
    _mk_dist(sp, "alpha", "1.0", license_name="Apache-2.0",
             license_text="Apache License 2.0 text here")
    _mk_dist(sp, "beta", "2.1",
             classifier="License :: OSI Approved :: MIT License",
             license_text="MIT text")
    out = tmp_path / "NOTICES.md"
    proc = run("--site-packages", str(sp), "--output", str(out))
    assert proc.returncode == 0, proc.stderr
    text = out.read_text()
    assert "| alpha | 1.0 | Apache-2.0 |" in text
    assert "MIT" in text
    assert "Apache License 2.0 text here" in text


def test_fails_on_unresolved(tmp_path):
    sp = tmp_path / "sp"
    _mk_dist(sp, "mystery", "0.1")  # no license info at all
    proc = run("--site-packages", str(sp), "--output", str(tmp_path / "o.md"))
    assert proc.returncode == 1
    assert "unresolved" in proc.stderr


def test_allow_unresolved_flag(tmp_path):
    sp = tmp_path / "sp"
    _mk_dist(sp, "mystery", "0.1")
    proc = run(
        "--site-packages", str(sp), "--output", str(tmp_path / "o.md"),
        "--allow-unresolved",
    )
    assert proc.returncode == 0


def test_path_escape_fails(tmp_path):
    sp = tmp_path / "sp"
    d = _mk_dist(sp, "evil", "0.1", license_name="MIT")
    (d / "METADATA").write_text(
        "Metadata-Version: 2.1\nName: evil\nVersion: 0.1\n"
        "License-Expression: MIT\nLicense-File: ../../../etc/passwd\n"
    )
    proc = run("--site-packages", str(sp), "--output", str(tmp_path / "o.md"))
    assert proc.returncode == 1
    assert "escapes" in proc.stderr
