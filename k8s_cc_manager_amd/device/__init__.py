"""L1 — the GPU device layer.

The reference outsources this layer to NVIDIA's ``gpu-admin-tools``
(/root/reference/main.py:37-44); here it is first-party: a typed contract
(:mod:`.contract`), a fault-injectable mock (:mod:`.mock`) and the real
amdsmi/KFD backend (:mod:`.amdsmi_backend`).
"""

from .contract import (  # noqa: F401
    CCDevice,
    CCDeviceError,
    ModeVerifyError,
    ResetError,
    BootTimeoutError,
    DeviceBackend,
    FABRIC_OFF,
    FABRIC_ON,
)


def get_backend(name: str = "auto", **kwargs) -> DeviceBackend:
    """Construct a device backend by name.

    ``mock``   — the fault-injectable fake (CPU-only).
    ``amdsmi`` — the real MI355X backend (amdsmi + sysfs + /dev/kfd).
    ``auto``   — amdsmi when AMD GPUs are present, else mock.
    """
    if name == "mock":
        from .mock import MockBackend

        return MockBackend(**kwargs)
    if name == "amdsmi":
        from .amdsmi_backend import AmdSmiBackend

        return AmdSmiBackend(**kwargs)
    if name == "auto":
        try:
            from .amdsmi_backend import AmdSmiBackend

            be = AmdSmiBackend(**kwargs)
            if be.find_devices()[1] > 0:
                return be
        except Exception:  # pragma: no cover - amdsmi missing/no GPUs
            pass
        from .mock import MockBackend

        return MockBackend(**kwargs)
    raise ValueError(f"unknown device backend: {name!r}")
