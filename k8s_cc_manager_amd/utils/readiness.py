"""Readiness-file signal consumed by the GPU-operator validator.

Reference behavior: touch the file after the initial mode apply, never
fail the process on errors (/root/reference/main.py:63-78); the pod's
preStop hook deletes it (static rm, Dockerfile.distroless:45-46).
"""

from __future__ import annotations

import logging
import os
from pathlib import Path

logger = logging.getLogger(__name__)

DEFAULT_READINESS_FILE = "/run/amd/validations/.cc-manager-ctr-ready"


def readiness_file_path() -> str:
    return os.environ.get("CC_READINESS_FILE", DEFAULT_READINESS_FILE)


def create_readiness_file(path: str | None = None) -> bool:
    p = Path(path or readiness_file_path())
    try:
        p.parent.mkdir(parents=True, exist_ok=True)
        p.touch()
        logger.info("created readiness file %s", p)
        return True
    except Exception as e:  # never fatal, matching reference main.py:76-78
        logger.warning("could not create readiness file %s: %s", p, e)
        return False


def remove_readiness_file(path: str | None = None) -> None:
    p = Path(path or readiness_file_path())
    try:
        p.unlink(missing_ok=True)
    except Exception as e:  # pragma: no cover
        logger.warning("could not remove readiness file %s: %s", p, e)
