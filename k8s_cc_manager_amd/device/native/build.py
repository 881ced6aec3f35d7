"""In-tree build of the _devnative C++ extension (g++ + pybind11)."""

from __future__ import annotations

import logging
import subprocess
import sysconfig
from pathlib import Path

logger = logging.getLogger(__name__)

NATIVE_DIR = Path(__file__).resolve().parent
SRC = NATIVE_DIR / "devnative.cpp"


def ext_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return NATIVE_DIR / f"_devnative{suffix}"


def build(force: bool = False) -> Path:
    import pybind11

    out = ext_path()
    if not force and out.exists() and out.stat().st_mtime >= SRC.stat().st_mtime:
        return out
    cmd = [
        "g++",
        "-O2",
        "-shared",
        "-fPIC",
        "-std=c++17",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        str(SRC),
        "-o",
        str(out),
    ]
    logger.info("building %s", out.name)
    subprocess.run(cmd, check=True, capture_output=True, text=True)
    return out


if __name__ == "__main__":
    print(build(force=True))
