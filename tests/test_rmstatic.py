"""Build + behavior tests for the static rm (C11-equivalent component:
shell-free preStop cleanup in the distroless image)."""

import subprocess
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
SRC = REPO / "native" / "rmstatic" / "rm.c"


@pytest.fixture(scope="module")
def rm_bin(tmp_path_factory):
    out = tmp_path_factory.mktemp("rm") / "rm"
    subprocess.run(
        ["gcc", "-static", "-Os", "-o", str(out), str(SRC)],
        check=True,
        capture_output=True,
    )
    return out


def test_remove_file(rm_bin, tmp_path):
    f = tmp_path / "x"
    f.write_text("1")
    assert subprocess.run([rm_bin, str(f)]).returncode == 0
    assert not f.exists()


def test_recursive(rm_bin, tmp_path):
    d = tmp_path / "a" / "b" / "c"
    d.mkdir(parents=True)
    (d / "f").write_text("1")
    (tmp_path / "a" / "g").write_text("2")
    assert subprocess.run([rm_bin, "-rf", str(tmp_path / "a")]).returncode == 0
    assert not (tmp_path / "a").exists()


def test_dir_without_r_fails(rm_bin, tmp_path):
    d = tmp_path / "d"
    d.mkdir()
    assert subprocess.run([rm_bin, str(d)], capture_output=True).returncode == 1
    assert d.exists()


def test_missing_without_f_fails(rm_bin, tmp_path):
    assert (
        subprocess.run([rm_bin, str(tmp_path / "none")], capture_output=True).returncode
        == 1
    )


def test_missing_with_f_ok(rm_bin, tmp_path):
    assert subprocess.run([rm_bin, "-f", str(tmp_path / "none")]).returncode == 0


def test_no_symlink_follow(rm_bin, tmp_path):
    target = tmp_path / "target"
    target.mkdir()
    (target / "keep").write_text("1")
    tree = tmp_path / "tree"
    tree.mkdir()
    (tree / "link").symlink_to(target)
    assert subprocess.run([rm_bin, "-rf", str(tree)]).returncode == 0
    assert not tree.exists()
    assert (target / "keep").exists()  # never descended through the link


def test_static_binary(rm_bin):
    out = subprocess.run(["file", str(rm_bin)], capture_output=True, text=True)
    assert "statically linked" in out.stdout
