from .executor import DeviceExecutor, PerDeviceError  # noqa: F401
from .fabric import FabricBarrier, DistFabricBarrier  # noqa: F401
