from .transition import TransitionEngine, TransitionReport  # noqa: F401
from .manager import CCManager, ManagerConfig, FatalConfigError  # noqa: F401
from .hostprobe import is_host_cc_enabled  # noqa: F401
