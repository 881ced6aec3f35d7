"""Native device library loader.

Exposes the C++ ``_devnative`` extension (PCI sysfs scan, config-space
reads, FLR, /dev/kfd ioctls). Built in-tree by
:mod:`k8s_cc_manager_amd.device.native.build` (g++ + pybind11, no GPU
needed).
"""

from __future__ import annotations

import importlib
import logging
from typing import Any, Dict, List, Optional, Tuple

logger = logging.getLogger(__name__)

_mod = None


def _load():
    global _mod
    if _mod is None:
        try:
            _mod = importlib.import_module(
                "k8s_cc_manager_amd.device.native._devnative"
            )
        except ImportError as e:
            raise RuntimeError(
                "_devnative extension not built — run "
                "python -m k8s_cc_manager_amd.device.native.build"
            ) from e
    return _mod


def available() -> bool:
    try:
        _load()
        return True
    except RuntimeError:
        return False


def pci_scan(root: str = "/sys/bus/pci/devices") -> List[Dict[str, Any]]:
    return _load().pci_scan(root)


def pci_config_read(bdf: str, offset: int = 0, size: int = 64,
                    root: str = "/sys/bus/pci/devices") -> bytes:
    return _load().pci_config_read(bdf, offset, size, root)


def pci_reset(bdf: str, root: str = "/sys/bus/pci/devices") -> None:
    _load().pci_reset(bdf, root)


def kfd_version(dev_path: str = "/dev/kfd") -> Optional[Tuple[int, int]]:
    return _load().kfd_version(dev_path)


def kfd_topology(root: str = "/sys/class/kfd/kfd/topology/nodes") -> List[Dict[str, Any]]:
    return _load().kfd_topology(root)
