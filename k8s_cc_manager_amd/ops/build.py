"""In-tree build of the gfx950 attestation library.

Plain ``hipcc --offload-arch=gfx950`` (cross-compiles without a GPU);
the resulting .so sits next to this file so it travels with the repo
snapshot to GPU boxes. No JIT cache involvement.
"""

from __future__ import annotations

import logging
import os
import subprocess
from pathlib import Path

logger = logging.getLogger(__name__)

OPS_DIR = Path(__file__).resolve().parent
SRC = OPS_DIR / "attest_kernels.hip"
LIB = OPS_DIR / "libccattest.so"
HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("CC_GPU_ARCH", "gfx950")


def build(force: bool = False) -> Path:
    """Compile the attestation library if the source is newer."""
    if (
        not force
        and LIB.exists()
        and LIB.stat().st_mtime >= SRC.stat().st_mtime
    ):
        return LIB
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-shared",
        "-fPIC",
        "-Wno-unused-value",
        str(SRC),
        "-o",
        str(LIB),
    ]
    logger.info("building %s: %s", LIB.name, " ".join(cmd))
    subprocess.run(cmd, check=True, capture_output=True, text=True)
    return LIB


if __name__ == "__main__":
    build(force=True)
    print(LIB)
